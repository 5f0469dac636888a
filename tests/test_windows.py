"""CPU window/join suite vs brute-force numpy oracles.

Mirrors the reference's win_tests / join_tests differential strategy
 but with exact per-window oracles, which the reference
lacks: every operator's fired (key, value) multiset is compared against
a plain recomputation over the generated stream.
"""
import math
from collections import Counter, defaultdict

import numpy as np
import pytest

import windflow_amd as wf
from windflow_amd import native
from windflow_amd.builders import (Keyed_Windows_Builder, Parallel_Windows_Builder,
                                   Paned_Windows_Builder, MapReduce_Windows_Builder,
                                   Ffat_Windows_Builder, Interval_Join_Builder,
                                   Source_Builder, Sink_Builder)


class Collect:
    """Python sink callable collecting (ts, key, c0[, c1]) rows."""

    def __init__(self):
        self.rows = []

    def __call__(self, cols):
        n = len(cols['ts'])
        cs = [cols[f'c{i}'] for i in range(8) if f'c{i}' in cols]
        for i in range(n):
            self.rows.append((int(cols['ts'][i]), int(cols['key'][i]),
                              *[c[i].item() for c in cs]))


def run_graph(win_op, stream_len=3000, n_keys=7, batch=128, src_par=1,
              mode=wf.ExecutionMode.DEFAULT):
    g = wf.PipeGraph("t", mode)
    src = (Source_Builder(native.seq_source(stream_len, n_keys, batch))
           .withParallelism(src_par).withOutputSchema([0])
           .withOutputBatchSize(batch).build())
    mp = g.add_source(src)
    mp.add(win_op)
    col = Collect()
    snk = Sink_Builder(col).withParallelism(1).build()
    mp.add_sink(snk)
    g.run()
    return col.rows


def seq_stream(stream_len, n_keys):
    """key -> list of (ts, value) in arrival order (v=1..N, key=v%K, ts=v)."""
    per = defaultdict(list)
    for v in range(1, stream_len + 1):
        per[v % n_keys].append((v, v))
    return per


AGGS = {"sum": sum, "max": max, "min": min,
        "count": len, "avg": lambda xs: sum(xs) / len(xs)}


def oracle_cb(per, win, slide, agg="sum"):
    """All windows incl. EOS partials: per key, window w = rows [w*s, w*s+win)."""
    f = AGGS[agg]
    out = Counter()
    for k, rows in per.items():
        vals = [v for _, v in rows]
        w = 0
        while w * slide < len(vals):
            seg = vals[w * slide: w * slide + win]
            out[(k, f(seg))] += 1
            w += 1
    return out


def oracle_tb(per, win, slide, agg="sum"):
    """Windows [wS, wS+win) on ts; from the first window containing the
    key's first tuple up to the last containing its max ts."""
    f = AGGS[agg]
    out = Counter()
    for k, rows in per.items():
        ts = [t for t, _ in rows]
        vals = {t: v for t, v in rows}
        t0, tmax = ts[0], ts[-1]
        w0 = max(0, -(-(t0 - win + 1) // slide))
        for w in range(w0, tmax // slide + 1):
            seg = [vals[t] for t in ts if w * slide <= t < w * slide + win]
            out[(k, f(seg) if seg else 0)] += 1
    return out


def got_counter(rows):
    return Counter((k, v) for _, k, v in rows)


# ---------------- keyed windows ----------------
@pytest.mark.parametrize("agg", ["sum", "max", "count"])
def test_keyed_cb(agg):
    win, slide = 40, 10
    rows = run_graph(Keyed_Windows_Builder(func=(agg, 0)).withCBWindows(win, slide)
                     .withOutputSchema([0]).build())
    assert got_counter(rows) == oracle_cb(seq_stream(3000, 7), win, slide, agg)


@pytest.mark.parametrize("agg", ["sum", "min"])
def test_keyed_tb(agg):
    win, slide = 100, 25
    rows = run_graph(Keyed_Windows_Builder(func=(agg, 0)).withTBWindows(win, slide)
                     .withOutputSchema([0]).build())
    assert got_counter(rows) == oracle_tb(seq_stream(3000, 7), win, slide, agg)


def test_keyed_cb_parallel_replicas():
    win, slide = 30, 30  # tumbling
    rows = run_graph(Keyed_Windows_Builder(func=("sum", 0))
                     .withCBWindows(win, slide).withParallelism(3)
                     .withOutputSchema([0]).build())
    assert got_counter(rows) == oracle_cb(seq_stream(3000, 7), win, slide)


def test_keyed_avg_f64():
    win, slide = 50, 50
    rows = run_graph(Keyed_Windows_Builder(func=("avg", 0)).withCBWindows(win, slide)
                     .withOutputSchema([1]).build())
    exp = oracle_cb(seq_stream(3000, 7), win, slide, "avg")
    got = Counter((k, round(v, 9)) for _, k, v in rows)
    assert got == Counter({(k, round(v, 9)): c for (k, v), c in exp.items()})


# ---------------- python (non-incremental) windows ----------------
def test_py_window_fn_matches_native():
    win, slide = 25, 5

    def median(w):
        return float(np.median(w['c0'])) if len(w['c0']) else 0.0

    rows = run_graph(Keyed_Windows_Builder(func=median).withCBWindows(win, slide)
                     .withOutputSchema([1]).build(), stream_len=1500)
    per = seq_stream(1500, 7)
    exp = Counter()
    for k, r in per.items():
        vals = [v for _, v in r]
        w = 0
        while w * slide < len(vals):
            exp[(k, float(np.median(vals[w * slide: w * slide + win])))] += 1
            w += 1
    assert Counter((k, v) for _, k, v in rows) == exp


def test_py_window_tb():
    win, slide = 80, 20

    def total(w):
        return float(w['c0'].sum())

    rows = run_graph(Keyed_Windows_Builder(func=total).withTBWindows(win, slide)
                     .withOutputSchema([0]).build(), stream_len=2000)
    assert got_counter(rows) == oracle_tb(seq_stream(2000, 7), win, slide, "sum")


# ---------------- parallel / paned / mapreduce ----------------
def test_parallel_windows_cb():
    win, slide = 40, 10
    rows = run_graph(Parallel_Windows_Builder(func=("sum", 0))
                     .withCBWindows(win, slide).withParallelism(4)
                     .withOutputSchema([0]).build())
    assert got_counter(rows) == oracle_cb(seq_stream(3000, 7), win, slide)


def test_parallel_windows_tb():
    win, slide = 60, 15
    rows = run_graph(Parallel_Windows_Builder(func=("max", 0))
                     .withTBWindows(win, slide).withParallelism(3)
                     .withOutputSchema([0]).build())
    assert got_counter(rows) == oracle_tb(seq_stream(3000, 7), win, slide, "max")


@pytest.mark.parametrize("par", [1, 3])
def test_paned_windows_tb(par):
    win, slide = 100, 20  # pane = 20
    rows = run_graph(Paned_Windows_Builder(plq_func=("sum", 0))
                     .withTBWindows(win, slide).withParallelism(par)
                     .withOutputSchema([0]).build())
    assert got_counter(rows) == oracle_tb(seq_stream(3000, 7), win, slide)


@pytest.mark.parametrize("par", [1, 3])
def test_paned_windows_cb(par):
    win, slide = 40, 10  # pane = 10 (count-based panes per key)
    rows = run_graph(Paned_Windows_Builder(plq_func=("sum", 0))
                     .withCBWindows(win, slide).withParallelism(par)
                     .withOutputSchema([0]).build())
    assert got_counter(rows) == oracle_cb(seq_stream(3000, 7), win, slide)


def test_mapreduce_windows_cb():
    win, slide = 40, 10
    rows = run_graph(MapReduce_Windows_Builder(map_func=("sum", 0))
                     .withCBWindows(win, slide).withParallelism(3)
                     .withOutputSchema([0]).build())
    assert got_counter(rows) == oracle_cb(seq_stream(3000, 7), win, slide)


def test_mapreduce_windows_tb():
    win, slide = 90, 30
    rows = run_graph(MapReduce_Windows_Builder(map_func=("sum", 0))
                     .withTBWindows(win, slide).withParallelism(2)
                     .withOutputSchema([0]).build())
    assert got_counter(rows) == oracle_tb(seq_stream(3000, 7), win, slide)


# ---------------- FFAT (FlatFAT) ----------------
@pytest.mark.parametrize("agg", ["sum", "max"])
def test_ffat_cb_matches_keyed(agg):
    win, slide = 64, 16
    ff = (Ffat_Windows_Builder(comb=(agg, 0)).withCBWindows(win, slide)
          .withOutputSchema([0]).build())
    rows = run_graph(ff)
    assert got_counter(rows) == oracle_cb(seq_stream(3000, 7), win, slide, agg)


@pytest.mark.parametrize("agg", ["sum", "max"])
def test_ffat_tb_matches_keyed(agg):
    win, slide = 120, 30
    ff = (Ffat_Windows_Builder(comb=(agg, 0)).withTBWindows(win, slide)
          .withOutputSchema([0]).build())
    rows = run_graph(ff)
    assert got_counter(rows) == oracle_tb(seq_stream(3000, 7), win, slide, agg)


def test_ffat_cb_nonpow2_win():
    win, slide = 100, 10
    ff = (Ffat_Windows_Builder(comb=("min", 0)).withCBWindows(win, slide)
          .withOutputSchema([0]).build())
    rows = run_graph(ff, stream_len=5000, n_keys=11)
    assert got_counter(rows) == oracle_cb(seq_stream(5000, 11), win, slide, "min")


# ---------------- interval join ----------------
def join_graph(mode_builder, n=2000, keys=5, lower=-10, upper=10, batch=64,
               mode=wf.ExecutionMode.DEFAULT):
    g = wf.PipeGraph("j", mode)
    srcA = (Source_Builder(native.seq_source(n, keys, batch))
            .withParallelism(1).withOutputSchema([0]).withOutputBatchSize(batch)
            .build())
    srcB = (Source_Builder(native.seq_source(n, keys, batch, value_offset=100000))
            .withParallelism(1).withOutputSchema([0]).withOutputBatchSize(batch)
            .build())
    mpA = g.add_source(srcA)
    mpB = g.add_source(srcB)
    mp = mpA.merge(mpB)
    jb = Interval_Join_Builder().withBoundaries(lower, upper).withValueCols(0)
    jb = mode_builder(jb)
    mp.add(jb.withOutputSchema([0, 0]).build())
    col = Collect()
    mp.add_sink(Sink_Builder(col).withParallelism(1).build())
    g.run()
    return col.rows


def oracle_join(n, keys, lower, upper):
    """(a from A, b from B) with same key and b.ts - a.ts in [lower, upper].
    A: ts=v, val=v, key=v%keys; B: ts=v, val=v+100000,
    key=(v+100000)%keys (seq_source keys by the OFFSET value)."""
    out = Counter()
    for v in range(1, n + 1):         # a
        k = v % keys
        for bts in range(v + lower, v + upper + 1):
            if 1 <= bts <= n and (bts + 100000) % keys == k:
                out[(k, v, bts + 100000)] += 1
    return out


def test_interval_join_kp():
    rows = join_graph(lambda b: b.withKPMode(), n=1500)
    got = Counter((k, a, b) for _, k, a, b in rows)
    assert got == oracle_join(1500, 5, -10, 10)


def test_interval_join_kp_parallel():
    rows = join_graph(lambda b: b.withKPMode().withParallelism(3), n=1500)
    got = Counter((k, a, b) for _, k, a, b in rows)
    assert got == oracle_join(1500, 5, -10, 10)


def test_interval_join_dp():
    rows = join_graph(lambda b: b.withDPMode().withParallelism(3), n=1200)
    got = Counter((k, a, b) for _, k, a, b in rows)
    assert got == oracle_join(1200, 5, -10, 10)


# ---------------- lateness / out-of-order ----------------
def run_ooo_graph(win_op, stream_len=3000, disorder=30, batch=100, seed=5):
    """Source emitting bounded-disorder event times: ts values are a
    permutation of 1..N where |perm[i] - i| <= disorder; the watermark
    lags max-emitted-ts by `disorder` (an honest source bound)."""
    import numpy as np
    rng = np.random.default_rng(seed)
    ts = np.arange(1, stream_len + 1, dtype=np.int64)
    for s in range(0, stream_len, disorder):
        rng.shuffle(ts[s:s + disorder])
    state = dict(pos=0)

    def src(replica, par):
        p = state['pos']
        if p >= stream_len:
            return None
        n = min(batch, stream_len - p)
        t = ts[p:p + n]
        state['pos'] += n
        return dict(ts=t, key=(t % 7).astype(np.uint64), c0=t,
                    watermark=int(ts[:p + n].max()) - disorder)

    g = wf.PipeGraph("ooo")
    mp = g.add_source(Source_Builder(src).withParallelism(1)
                      .withOutputSchema([0]).withOutputBatchSize(batch).build())
    mp.add(win_op)
    col = Collect()
    mp.add_sink(Sink_Builder(col).withParallelism(1).build())
    g.run()
    return col.rows


def test_tb_lateness_absorbs_disorder():
    """With lateness >= source disorder, out-of-order arrival must not
    change any window result (firing is gated on watermark - lateness)."""
    win, slide, disorder = 120, 30, 30
    rows = run_ooo_graph(Keyed_Windows_Builder(func=("sum", 0))
                         .withTBWindows(win, slide).withLateness(disorder)
                         .withOutputSchema([0]).build())
    per = defaultdict(list)
    for v in range(1, 3001):
        per[v % 7].append((v, v))
    assert got_counter(rows) == oracle_tb(per, win, slide, "sum")


def test_tb_ffat_lateness_absorbs_disorder():
    win, slide, disorder = 120, 30, 30
    rows = run_ooo_graph(Ffat_Windows_Builder(comb=("sum", 0))
                         .withTBWindows(win, slide).withLateness(disorder)
                         .withOutputSchema([0]).build())
    per = defaultdict(list)
    for v in range(1, 3001):
        per[v % 7].append((v, v))
    assert got_counter(rows) == oracle_tb(per, win, slide, "sum")


# ---------------- deterministic mode ----------------
def test_windows_deterministic_mode():
    win, slide = 40, 10
    rows = run_graph(Keyed_Windows_Builder(func=("sum", 0))
                     .withCBWindows(win, slide).withParallelism(2)
                     .withOutputSchema([0]).build(),
                     mode=wf.ExecutionMode.DETERMINISTIC)
    assert got_counter(rows) == oracle_cb(seq_stream(3000, 7), win, slide)


def test_py_window_parallel_replicas():
    """Non-incremental python windows across 3 keyed replicas."""
    win, slide = 30, 10

    def spread(w):
        return float(w['c0'].max() - w['c0'].min()) if len(w['c0']) else 0.0

    rows = run_graph(Keyed_Windows_Builder(func=spread).withCBWindows(win, slide)
                     .withParallelism(3).withOutputSchema([1]).build(),
                     stream_len=2100)
    per = seq_stream(2100, 7)
    exp = Counter()
    for k, r in per.items():
        vals = [v for _, v in r]
        w = 0
        while w * slide < len(vals):
            seg = vals[w * slide: w * slide + win]
            exp[(k, float(max(seg) - min(seg)))] += 1
            w += 1
    assert Counter((k, v) for _, k, v in rows) == exp


def test_py_window_paned_and_mapreduce_fallback():
    """Python (non-incremental) window functions on the paned and mapreduce
    forms: executed on the keyed python engine (same windows, same results —
    the parallel decomposition differs, which is invisible to the user)."""
    win, slide = 30, 10

    def total(w):
        return float(w['c0'].sum())

    for B in (Paned_Windows_Builder, MapReduce_Windows_Builder):
        kw = ("plq_func" if B is Paned_Windows_Builder else "map_func")
        rows = run_graph(B(**{kw: total}).withCBWindows(win, slide)
                         .withParallelism(3).withOutputSchema([0]).build(),
                         stream_len=1800)
        assert got_counter(rows) == oracle_cb(seq_stream(1800, 7), win, slide,
                                              "sum"), B.__name__


def test_avg_through_all_forms():
    """AVG with real slice weights on every decomposed form: the partials
    wire carries (value, slice count), so pane/subset recombination weights
    partial averages correctly (sum-of-sums / sum-of-counts)."""
    win, slide, n, keys = 40, 10, 3000, 7
    per = seq_stream(n, keys)
    exp = {}
    for k, r in per.items():
        vals = [v for _, v in r]
        outs = []
        w = 0
        while w * slide < len(vals):
            seg = vals[w * slide: w * slide + win]
            outs.append(round(sum(seg) / len(seg), 6))
            w += 1
        exp[k] = sorted(outs)
    for B in (Keyed_Windows_Builder, Parallel_Windows_Builder,
              Paned_Windows_Builder, MapReduce_Windows_Builder):
        kw = {"plq_func": ("avg", 0)} if B is Paned_Windows_Builder else (
             {"map_func": ("avg", 0)} if B is MapReduce_Windows_Builder else
             {"func": ("avg", 0)})
        rows = run_graph(B(**kw).withCBWindows(win, slide)
                         .withParallelism(3).withOutputSchema([1]).build(),
                         stream_len=n, n_keys=keys)
        got = {}
        for _, k, v in rows:
            got.setdefault(k, []).append(round(float(v), 6))
        got = {k: sorted(v) for k, v in got.items()}
        assert got == exp, B.__name__


def test_interval_join_user_function():
    """Arbitrary user predicate/result on the interval join (reference
    interval_join.hpp:279-307): keep pairs where (a+b) is even, result
    a*1000+b, vectorized over matched pairs."""
    import numpy as np
    n, n_keys, batch = 4000, 7, 128

    OFF = 1_000_000_001  # divisible by n_keys=7: A and B keys stay aligned

    def joinfn(p):
        a = p['a'].astype(np.int64)
        b = p['b'].astype(np.int64) - OFF
        keep = ((a + b) % 2) == 0
        return keep, (a * 1000 + b).astype(np.float64)

    g = wf.PipeGraph("ijfn")
    sa = (wf.Source_Builder(native.seq_source(n, n_keys, batch))
          .withParallelism(1).withOutputSchema([0])
          .withOutputBatchSize(batch).build())
    sb = (wf.Source_Builder(native.seq_source(n, n_keys, batch,
                                              value_offset=OFF))
          .withParallelism(1).withOutputSchema([0])
          .withOutputBatchSize(batch).build())
    mpA = g.add_source(sa)
    mpB = g.add_source(sb)
    mp = mpA.merge(mpB)
    from windflow_amd.builders import Interval_Join_Builder
    mp.add(Interval_Join_Builder(joinfn).withBoundaries(-1, 1).withKPMode()
           .withValueCols(0).withParallelism(2)
           .withOutputSchema([1]).withOutputBatchSize(batch).build())
    got = dict(s=0.0, n=0)

    def sink(cols):
        got['s'] += float(np.asarray(cols['c0']).sum())
        got['n'] += len(cols['c0'])

    snk = wf.Sink_Builder(sink).withParallelism(1).build()
    snk.out_schema = [1]
    mp.add_sink(snk)
    g.run()
    # oracle: seq_source emits v = 1..n, ts = v, key = v % n_keys (and
    # value_offset folds into v AND key for stream B)
    i = np.arange(1, n + 1, dtype=np.int64)        # A rows: ts=i, a=i
    exp_s = 0.0
    exp_n = 0
    for d in (-1, 0, 1):                            # b.ts - a.ts in [-1, 1]
        j = i + d                                   # B rows: ts=j, b=j+OFF
        ok = (j >= 1) & (j <= n)
        ia, jb = i[ok], j[ok]
        m = (ia % n_keys) == ((jb + OFF) % n_keys)
        ia, jb = ia[m], jb[m]
        keep = ((ia + jb) % 2) == 0
        exp_n += int(keep.sum())
        exp_s += float((ia[keep] * 1000 + jb[keep]).sum())
    assert exp_n > 0, "vacuous oracle (no joinable pairs)"
    assert got['n'] == exp_n
    assert abs(got['s'] - exp_s) <= 1e-6 * max(1.0, exp_s)
