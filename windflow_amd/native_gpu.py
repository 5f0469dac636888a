"""GPU native logic specs — map onto csrc/engine/gpu_ops.cpp factories and
the gfx950 kernels in csrc/hip/.  comb ids: 0 sum, 1 min, 2 max, 3 count.
Value dtypes: 0 I64, 2 F32, 5 U16(bf16)."""
from .operators import NativeLogic

COMB_SUM, COMB_MIN, COMB_MAX, COMB_COUNT = 0, 1, 2, 3


def gpu_source(stream_len, n_keys=1024, batch=1 << 20, vdt=5, seed=42):
    """Device-resident synthetic source (ts, key, value[vdt])."""
    return NativeLogic("gpu_source", "", [], [stream_len, n_keys, batch, vdt, seed])


def gpu_affine_map(col=0, a=1.0, b=0.0, dtype=5):
    """x = a*x + b on device; dtype picks the kernel (i64/f32/bf16)."""
    spec = {0: 1, 2: 2, 5: 3}[dtype]
    return NativeLogic("gpu_map", "", [float(a), float(b)], [spec, col])


def gpu_square_map(col=0):
    return NativeLogic("gpu_map", "", [0.0, 0.0], [4, col])


def gpu_mod_filter(col=0, m=3, c=0):
    """keep x % m != c (i64)."""
    return NativeLogic("gpu_filter", "", [float(m), float(c)], [1, col])


def gpu_gt_filter(col=0, thr=0.5, dtype=2):
    """keep x > thr (f32) / x >= thr (bf16)."""
    spec = 2 if dtype == 2 else 3
    return NativeLogic("gpu_filter", "", [float(thr), 0.0], [spec, col])


def gpu_jit_map(expr, col=0):
    """Custom device map, runtime-compiled with hiprtc for the local arch
    (parity with the reference's __device__ lambda MAP_GPU).  `expr` is a C
    expression over `v` (the column value, mutable type), `ts` (i64) and
    `key` (u64); the result is written back to the column, e.g.
    gpu_jit_map("v * v + 1.0f", col=0)."""
    return NativeLogic("gpu_jit_map", expr, [], [col])


def gpu_jit_filter(expr, col=0):
    """Custom device predicate (reference FILTER_GPU lambda): keep rows
    where `expr` is true, e.g. gpu_jit_filter("v > 0.5f && key % 2 == 0")."""
    return NativeLogic("gpu_jit_filter", expr, [], [col])


def _expr_list(x):
    if isinstance(x, str):
        return x.split(';')
    return list(x)


def _jit_fold_spec(lift, comb, finalize, identity, cols, acc="f32"):
    """Normalize a user fold into (spec, fparams, iparam-prefix).

    lift/comb/finalize: str (';'-joined C expressions) or list of str.
    lift sees v0..v{len(cols)-1} (in the accumulator type), ts (i64),
    key (u64); comb sees a0.., b0..; finalize sees f0..  identity: one
    float per field.  acc="f64" keeps the accumulator fields (and column
    loads) in double precision — output columns stay F32.
    Parity with the reference's arbitrary __device__ lift/combine functors
    (wf/builders_gpu.hpp:466-620, meta_gpu.hpp).  Combine must be
    associative AND commutative (folds are tree/wave shaped)."""
    lift, comb, fin = _expr_list(lift), _expr_list(comb), _expr_list(finalize)
    identity = [float(v) for v in identity]
    cols = [int(c) for c in cols]
    nf, nout, nc = len(lift), len(fin), len(cols)
    if len(comb) != nf or len(identity) != nf:
        raise ValueError("lift/comb/identity must have the same field count")
    if not 1 <= nc <= 4:
        raise ValueError("1..4 value columns")
    if not 1 <= nout <= 4:
        raise ValueError("1..4 outputs")
    if acc not in ("f32", "f64"):
        raise ValueError("acc must be 'f32' or 'f64'")
    spec = "\x1e".join([";".join(lift), ";".join(comb), ";".join(fin), acc])
    cols4 = (cols + [0, 0, 0, 0])[:4]
    return spec, identity, [nf, nout, nc] + cols4


def gpu_jit_reduce(lift="v0", comb="a0+b0", finalize="f0", identity=(0.0,),
                   cols=(0,), max_keys=1 << 16, acc="f32", dense_keys=False):
    """Per-batch keyed reduction with an ARBITRARY user fold, runtime-
    compiled with hiprtc (reference Reduce_GPU accepts any __device__
    combine, builders_gpu.hpp:350).  Emits one row per distinct key:
    (key, finalize(acc) as F32 columns, ts_max).  acc="f64" for double-
    precision accumulation (large sums, i64/f64 columns)."""
    spec, fp, pre = _jit_fold_spec(lift, comb, finalize, identity, cols, acc)
    return NativeLogic("gpu_jit_reduce", spec, fp,
                       pre + [int(max_keys), 1 if dense_keys else 0])


def gpu_avg_reduce(col=0, max_keys=1 << 16):
    """Keyed AVG: fused (sum, count) fold (see FUTURE.md round 1)."""
    return gpu_jit_reduce(lift="v0;1.0f", comb="a0+b0;a1+b1",
                          finalize="(f1 > 0.0f) ? (f0 / f1) : 0.0f",
                          identity=(0.0, 0.0), cols=(col,), max_keys=max_keys)


def gpu_jit_ffat_windows(win=1000, slide=100, lift="v0", comb="a0+b0",
                         finalize="f0", identity=(0.0,), cols=(0,),
                         max_keys=1 << 16, tb=False, lateness=0,
                         pend_ring_log2=0, invertible=False, acc="f32",
                         dense_keys=False):
    """Keyed sliding window with an ARBITRARY user lift+combine+finalize
    fold (reference Ffat_Windows_GPU arbitrary lift/comb functors,
    ffat_windows_gpu.hpp:60).  Multi-column values (cols), multi-column
    results (finalize list) and multi-field accumulators all supported;
    invertible=True asserts comb is fieldwise + (running-total windows
    instead of P-pane recombines — use for sum/avg-shaped folds)."""
    if win < 1 or slide < 1:
        raise ValueError("window length and slide must be >= 1")
    spec, fp, pre = _jit_fold_spec(lift, comb, finalize, identity, cols, acc)
    ip = pre + [int(max_keys), int(win), int(slide), 1 if tb else 0,
                int(lateness), int(pend_ring_log2), 1 if invertible else 0,
                1 if dense_keys else 0]
    return NativeLogic("gpu_jit_ffat", spec, fp, ip)


def gpu_avg_ffat_windows(win=1000, slide=100, col=0, **kw):
    """Keyed sliding-window AVG: fused (sum, count) pane pairs."""
    # empty (gap) windows fire the reference's default result 0, not 0/0:
    # the count field gates the division
    return gpu_jit_ffat_windows(win, slide, lift="v0;1.0f",
                                comb="a0+b0;a1+b1",
                                finalize="(f1 > 0.0f) ? (f0 / f1) : 0.0f",
                                identity=(0.0, 0.0), cols=(col,),
                                invertible=True, **kw)


def gpu_jit_stateful_map(body, ncols=1, nstate=1, init=(0.0,),
                         max_keys=1 << 16):
    """Keyed stateful map with an ARBITRARY statement body (reference
    stateful MAP_GPU functors, map_gpu.hpp:80-102).  body sees v0..v{n}
    (mutable float, written back in place), ts (i64), key (u64), s0..s{m}
    (mutable f64 per-key state), and runs per tuple in key order, e.g.
    "s0 = s0 + v0; v0 = s0" (running sum)."""
    return NativeLogic("gpu_jit_stateful", body, [float(v) for v in init],
                       [0, int(ncols), int(nstate), int(max_keys)])


def gpu_jit_stateful_filter(body, ncols=1, nstate=1, init=(0.0,),
                            max_keys=1 << 16):
    """Keyed stateful filter: body sets `keep` (int), e.g.
    "keep = v0 != s0; s0 = v0" (consecutive dedup)."""
    return NativeLogic("gpu_jit_stateful", body, [float(v) for v in init],
                       [1, int(ncols), int(nstate), int(max_keys)])


def gpu_jit_split(expr, ncols=1):
    """Per-tuple device split: a JIT-compiled branch expression over
    (v0..v{ncols-1}, ts, key) routes each ROW to one branch, compacted on
    device per branch (reference splitting_emitter_gpu replicates whole
    batches; per-tuple routing is the CPU Splitting_Emitter semantics).
    Rows whose branch id is outside [0, n_branches) are dropped."""
    return NativeLogic("gpu_split", expr, [], [int(ncols)])


def gpu_keyed_running_sum(col=0, max_keys=1 << 16):
    """stateful map: per-key running sum written in place (key-order walk)."""
    return NativeLogic("gpu_map_keyed", "", [0.0, 0.0], [1, col, max_keys])


def gpu_keyed_ema(col=0, alpha=0.9, max_keys=1 << 16):
    """stateful map: x = ema(state, x) per key."""
    return NativeLogic("gpu_map_keyed", "", [float(alpha), 0.0], [2, col, max_keys])


def gpu_keyed_running_count(col=0, max_keys=1 << 16):
    return NativeLogic("gpu_map_keyed", "", [0.0, 0.0], [3, col, max_keys])


def gpu_keyed_dedup(col=0, max_keys=1 << 16):
    """stateful filter: drop consecutive per-key duplicates."""
    return NativeLogic("gpu_filter_keyed", "", [0.0, 0.0], [1, col, max_keys])


def gpu_keyed_every_kth(k=2, col=0, max_keys=1 << 16):
    """stateful filter: keep every k-th tuple per key."""
    return NativeLogic("gpu_filter_keyed", "", [float(k), 0.0], [2, col, max_keys])


def gpu_keyed_reduce(comb=COMB_SUM, col=0, max_keys=1 << 16, dense_keys=False):
    """per-batch keyed reduction -> one (key, agg, ts_max) per distinct key.
    dense_keys: user-asserted integer keys in [0, max_keys) — slot = key,
    skipping the hash probe (~25% of the keyed chain); out-of-range keys
    fail loudly."""
    return NativeLogic("gpu_reduce", "", [],
                       [comb, col, max_keys, 1 if dense_keys else 0])


def gpu_reduce_all(comb=COMB_SUM, col=0):
    """per-batch UNKEYED full reduction -> one (value, ts_max) tuple per
    batch (reference reduce_gpu.hpp:269 thrust::reduce path); deterministic
    two-stage tree, no sort and no atomics."""
    return NativeLogic("gpu_reduce_all", "", [], [comb, col])


def gpu_ffat_windows(comb=COMB_SUM, col=0, win=1000, slide=100,
                     max_keys=1 << 16, use_tree=False, tb=False, lateness=0,
                     pend_ring_log2=0, dense_keys=False):
    """Keyed sliding window over panes (pane = gcd(win, slide)).
    CB (default): windows fire every `slide` tuples per key; use_tree
    selects the FlatFAT-arena path (O(log) combine for large win/slide
    ratios and non-invertible combines).
    TB (tb=True): event-time windows [w*slide, w*slide+win) on ts; panes
    complete at watermark - lateness; pending out-of-order panes live in a
    2^pend_ring_log2 ring per key (default 2^16)."""
    if win < 1 or slide < 1:
        raise ValueError("window length and slide must be >= 1")
    return NativeLogic("gpu_ffat", "", [],
                       [comb, col, win, slide, max_keys, 1 if use_tree else 0,
                        1 if tb else 0, lateness, pend_ring_log2,
                        1 if dense_keys else 0])


def gpu_keyby_exchange():
    """Inter-GPU keyby shuffle: rows bucketed by hash(key) % world on device,
    exchanged with a grouped RCCL send/recv all-to-allv over xGMI.  One
    replica per rank; requires PipeGraph.set_dist(rank, world, rccl_id)."""
    return NativeLogic("gpu_exchange", "", [], [])


def gpu_gram_windows(win, max_keys=1 << 14, dense_keys=False):
    """MFMA windowed Gram aggregator (MI355X extension): per-key tumbling
    windows of `win` tuples over 16 F32 payload columns; the window
    aggregate sum(v * v^T) — online covariance / Gram — is GEMM-shaped and
    runs on the matrix cores (v_mfma_f32_16x16x4_f32).  Emits 16 rows per
    fired window: (key, ts=last tuple, c0=gwid, c1..c16 = one Gram row).
    Output schema must be [I64] + [F32]*16."""
    return NativeLogic("gpu_gram", "", [],
                       [win, max_keys, 1 if dense_keys else 0])


def gpu_count_sink():
    return NativeLogic("gpu_count_sink", "", [], [])


def gpu_to_host():
    return NativeLogic("gpu_to_host", "", [], [])
