"""Native (compiled) logic specs — the fast path for sources/ops/sinks.

These map 1:1 onto the C++ catalog in csrc/engine/native_logic.cpp (CPU)
and csrc/hip/kernels.hip (GPU device functors).  Python callables are the
general path; these are the zero-interpreter hot paths the benchmarks use.
"""
from .operators import NativeLogic

I64, F64, F32, U64, I32, U16, U8 = range(7)


def seq_source(stream_len, n_keys=1, batch=1024, value_offset=0):
    """Deterministic keyed integer sequence: v=1..len, key=v%n_keys, ts=v."""
    return NativeLogic("source", "seq", [], [stream_len, n_keys, batch, value_offset])


def rand_source(stream_len, n_keys=1, batch=1024, seed=42):
    """Random keyed stream: key ~ U[0,n_keys), value ~ U[0,1000)."""
    return NativeLogic("source", "rand", [], [stream_len, n_keys, batch, seed])


def affine_map(col=0, a=1, b=0):
    """x = a*x + b on i64 column `col` (in place)."""
    return NativeLogic("map", "affine_i64", [], [col, a, b])


def rand_source_f(stream_len, n_keys=1, batch=1024, seed=42, dtype="f32"):
    """Random keyed FLOAT stream: value ~ U[0, 1000) in an f32/f64 column
    (round-2 float-path catalog — no GIL on float payloads)."""
    return NativeLogic("source", f"rand_{dtype}", [], [stream_len, n_keys, batch, seed])


def affine_map_f(col=0, a=1.0, b=0.0, dtype="f32"):
    """x = a*x + b on an f32/f64 column (in place, native)."""
    return NativeLogic("map", f"affine_{dtype}", [float(a), float(b)], [col])


def gt_filter_f(col=0, thr=0.0, keep_gt=True, dtype="f32"):
    """keep rows where x > thr (keep_gt=False keeps x <= thr), native."""
    return NativeLogic("filter", f"gt_{dtype}", [float(thr)],
                       [col, 1 if keep_gt else 0])


def keyed_reduce_f(col=0, comb="sum", init=0.0, dtype="f32"):
    """per-key running sum/min/max over a float column (f64 state); emits
    the updated (key, acc) per input (reference reduce.hpp + withInitialState)."""
    ci = {"sum": 0, "min": 1, "max": 2}[comb]
    return NativeLogic("reduce", f"comb_by_key_{dtype}", [float(init)], [col, ci])


def sum_sink_f(col=0, dtype="f32"):
    """accumulate f64 sum of a float column (read back with g.sink_sum_f)."""
    return NativeLogic("sink", f"sum_{dtype}", [], [col])


def mod_filter(col=0, m=2, c=0, keep_eq=False):
    """keep rows where (x % m != c); keep_eq flips to ==."""
    return NativeLogic("filter", "mod_i64", [], [col, m, c, 1 if keep_eq else 0])


def dup_flatmap(k=2):
    """emit each input row k times."""
    return NativeLogic("flatmap", "dup_i64", [], [k])


def keyed_sum_reduce(col=0):
    """per-key running sum; emits the updated (key, acc) per input tuple."""
    return NativeLogic("reduce", "sum_by_key_i64", [], [col])


def sum_sink(col=0):
    """accumulate sum of i64 column into engine accumulator (invariant tests)."""
    return NativeLogic("sink", "sum_i64", [], [col])


def last_per_key_sink(col=0):
    """Sink that sums the LAST value per key at EOS (read back with
    g.sink_sum): exposes final per-key accumulators, e.g. after P_Reduce."""
    return NativeLogic("sink", "last_per_key_i64", [], [col])


def count_sink():
    return NativeLogic("sink", "count", [], [])


def split_mod(col=0):
    """split branch = x % n_branches on i64 column."""
    return NativeLogic("split", "mod_i64", [], [col])


def win_agg(op, col=0):
    """Compiled incremental window combine for the window builders:
    op in {'sum','min','max','count','avg'} over payload column `col`."""
    assert op in ("sum", "min", "max", "count", "avg"), op
    return NativeLogic("win_agg", op, [], [col])
