"""Randomized differential sweep over the window operators: random
win/slide/key-count/batch-size/parallelism drawn per case, every operator
form compared against the same brute-force oracle (the reference randomizes
degrees but has no oracle)."""
import random
from collections import defaultdict

import pytest

import windflow_amd as wf
from windflow_amd import native
from windflow_amd.builders import (Keyed_Windows_Builder, Parallel_Windows_Builder,
                                   Paned_Windows_Builder, MapReduce_Windows_Builder,
                                   Ffat_Windows_Builder)

import sys, os
sys.path.insert(0, os.path.dirname(__file__))
from test_windows import (run_graph, seq_stream, oracle_cb, oracle_tb,
                          got_counter)  # noqa: E402

BUILDERS = {
    "keyed": lambda f: Keyed_Windows_Builder(func=f),
    "parallel": lambda f: Parallel_Windows_Builder(func=f),
    "paned": lambda f: Paned_Windows_Builder(plq_func=f),
    "mapreduce": lambda f: MapReduce_Windows_Builder(map_func=f),
    "ffat": lambda f: Ffat_Windows_Builder(comb=f),
}


@pytest.mark.parametrize("case", range(18))
def test_window_fuzz(case):
    # widened space (min/count aggs, slide=1, 64 keys, batch > stream): the
    # original narrow draw missed three real bugs — MIN/MAX identity on
    # empty partials, empty-TB-window skip at EOS, and intermediate-batch
    # watermark stamping (all found by a 300-config campaign of this loop)
    rng = random.Random(50000 + case * 17)
    kind = rng.choice(list(BUILDERS))
    agg = rng.choice(["sum", "max", "min", "count"])
    wt = rng.choice(["cb", "tb"])
    slide = rng.choice([1, 3, 5, 10, 25, 60])
    win = slide * rng.randint(1, 8)
    n_keys = rng.choice([1, 2, 3, 11, 64])
    batch = rng.choice([16, 32, 128, 1000, 4096])
    par = rng.randint(1, 4)
    stream = rng.choice([700, 1500, 3100, 7000])
    mode = rng.choice([wf.ExecutionMode.DEFAULT, wf.ExecutionMode.DETERMINISTIC])
    b = BUILDERS[kind]((agg, 0))
    b = (b.withCBWindows(win, slide) if wt == "cb" else b.withTBWindows(win, slide))
    op = b.withParallelism(par).withOutputSchema([0]).build()
    rows = run_graph(op, stream_len=stream, n_keys=n_keys, batch=batch, mode=mode)
    per = seq_stream(stream, n_keys)
    exp = (oracle_cb if wt == "cb" else oracle_tb)(per, win, slide, agg)
    assert got_counter(rows) == exp, (kind, agg, wt, win, slide, n_keys, batch, par)


@pytest.mark.parametrize("case", range(8))
def test_join_fuzz(case):
    """Randomized interval joins: bounds/keys/batch/parallelism/mode drawn
    per case, vs the brute-force pair oracle."""
    from collections import Counter
    from test_windows import join_graph, oracle_join
    rng = random.Random(90000 + case * 13)
    lower = -rng.randint(0, 30)
    upper = rng.randint(0, 30)
    keys = rng.choice([1, 2, 5, 9, 33])
    n = rng.choice([300, 800, 1500, 4000])
    batch = rng.choice([8, 16, 64, 300, 2048])
    par = rng.randint(1, 4)
    kp = rng.random() < 0.5
    rows = join_graph(lambda b: (b.withKPMode() if kp else b.withDPMode())
                      .withParallelism(par),
                      n=n, keys=keys, lower=lower, upper=upper, batch=batch)
    got = Counter((k, a, bb) for _, k, a, bb in rows)
    assert got == oracle_join(n, keys, lower, upper), (lower, upper, keys, n,
                                                       batch, par, kp)


@pytest.mark.parametrize("case", range(10))
def test_py_window_fn_fuzz(case):
    """Non-incremental python window functions (median/span/sumsq) on every
    window form vs brute-force oracle (caught the parallel form silently
    running the native sum combiner instead of the user's function)."""
    from collections import Counter
    import numpy as np
    from test_windows import run_graph, seq_stream
    from windflow_amd.builders import (Keyed_Windows_Builder,
                                       Parallel_Windows_Builder,
                                       Paned_Windows_Builder,
                                       MapReduce_Windows_Builder)
    FNS = {"median": lambda a: float(np.median(a)),
           "span": lambda a: float(max(a) - min(a)),
           "sumsq": lambda a: float(np.sum(np.asarray(a, np.float64) ** 2))}
    BUILD = {"keyed": lambda f: Keyed_Windows_Builder(func=f),
             "parallel": lambda f: Parallel_Windows_Builder(func=f),
             "paned": lambda f: Paned_Windows_Builder(plq_func=f),
             "mapreduce": lambda f: MapReduce_Windows_Builder(map_func=f)}
    rng = random.Random(777 + case * 7)
    if case >= 8:
        # the regime that exposed the PyWindowLogic TB EOS bug: key stride
        # wider than the window -> trailing/empty windows flushed at EOS
        rng = random.Random(26_000_055 + (case - 8) * 26)
    form = rng.choice(list(BUILD))
    fn = FNS[rng.choice(list(FNS))]
    wt = "tb" if case >= 8 else rng.choice(["cb", "tb"])
    slide = rng.choice([3, 5, 10, 20] if case >= 8 else [5, 10, 20])
    win = slide * rng.randint(1, 5)
    keys = rng.choice([1, 3, 7])
    par = rng.randint(1, 3)
    stream = rng.choice([600, 1500])
    pyfn = lambda w, _f=fn: _f(w['c0']) if len(w['c0']) else 0.0
    b = BUILD[form](pyfn)
    b = b.withCBWindows(win, slide) if wt == "cb" else b.withTBWindows(win, slide)
    rows = run_graph(b.withParallelism(par).withOutputSchema([1]).build(),
                     stream_len=stream, n_keys=keys)
    per = seq_stream(stream, keys)
    exp = Counter()
    for k, r in per.items():
        vals = [v for _, v in r]
        tss = [t for t, _ in r]
        if wt == "cb":
            w = 0
            while w * slide < len(vals):
                exp[(k, round(fn(vals[w*slide:w*slide+win]), 6))] += 1
                w += 1
        else:
            t0, tmax = tss[0], tss[-1]
            w0 = max(0, -(-(t0 - win + 1) // slide))
            for w in range(w0, tmax // slide + 1):
                seg = [v for t, v in zip(tss, vals) if w*slide <= t < w*slide+win]
                exp[(k, round(fn(seg) if seg else 0.0, 6))] += 1
    got = Counter((k, round(v, 6)) for _, k, v in rows)
    assert got == exp, (form, wt, win, slide, keys, par)


@pytest.mark.parametrize("case", range(6))
def test_asymmetric_stage_parallelism_fuzz(case):
    """Paned/MapReduce with different per-stage parallelism degrees
    (withPLQ/WLQ/MAP/REDUCEParallelism) vs oracle (80-config campaign ran
    clean)."""
    from test_windows import run_graph, seq_stream, oracle_cb, oracle_tb, got_counter
    rng = random.Random(8800 + case * 13)
    form = rng.choice(["paned", "mapreduce"])
    agg = rng.choice(["sum", "max", "min", "count"])
    wt = rng.choice(["cb", "tb"])
    slide = rng.choice([5, 10, 25])
    win = slide * rng.randint(1, 6)
    keys = rng.choice([1, 3, 11])
    p1, p2 = rng.randint(1, 4), rng.randint(1, 4)
    stream = rng.choice([1000, 3000])
    if form == "paned":
        b = (Paned_Windows_Builder(plq_func=(agg, 0))
             .withPLQParallelism(p1).withWLQParallelism(p2))
    else:
        b = (MapReduce_Windows_Builder(map_func=(agg, 0))
             .withMAPParallelism(p1).withREDUCEParallelism(p2))
    b = b.withCBWindows(win, slide) if wt == "cb" else b.withTBWindows(win, slide)
    rows = run_graph(b.withOutputSchema([0]).build(), stream_len=stream,
                     n_keys=keys)
    per = seq_stream(stream, keys)
    exp = (oracle_cb if wt == "cb" else oracle_tb)(per, win, slide, agg)
    assert got_counter(rows) == exp, (form, agg, wt, win, slide, keys, p1, p2)


@pytest.mark.parametrize("case", range(4))
def test_irregular_ts_tb_fuzz(case):
    """TB windows over bursty streams: duplicate timestamps, gaps, negative
    values, python source (80-config campaign ran clean)."""
    from collections import Counter
    import numpy as np
    import windflow_amd as wf
    from windflow_amd.builders import Keyed_Windows_Builder, Ffat_Windows_Builder
    rng = random.Random(940_000 + case * 19)
    n = rng.choice([500, 2000])
    keys = rng.choice([1, 3, 9])
    max_gap = rng.choice([1, 4, 30])
    slide = rng.choice([5, 10, 50])
    win = slide * rng.randint(1, 5)
    agg = rng.choice(["sum", "max", "min", "count"])
    par = rng.randint(1, 3)
    batch = rng.choice([32, 256])
    form = rng.choice(["keyed", "ffat"])
    ts, t = [], 1
    for _ in range(n):
        if rng.random() >= 0.3:
            t += rng.randint(1, max_gap)
        ts.append(t)
    key = [rng.randrange(keys) for _ in range(n)]
    val = [rng.randint(-50, 50) for _ in range(n)]
    state = dict(pos=0)

    def src(replica, parallelism):
        p = state['pos']
        if p >= len(ts):
            return None
        q = min(p + batch, len(ts))
        state['pos'] = q
        return dict(ts=np.array(ts[p:q], np.int64),
                    key=np.array(key[p:q], np.uint64),
                    c0=np.array(val[p:q], np.int64),
                    watermark=int(ts[q - 1]))

    rows = []

    def snk(cols):
        for i in range(len(cols['key'])):
            rows.append((int(cols['key'][i]), int(cols['c0'][i])))

    B = Keyed_Windows_Builder if form == "keyed" else Ffat_Windows_Builder
    kw = {"func": (agg, 0)} if form == "keyed" else {"comb": (agg, 0)}
    g = wf.PipeGraph("irr")
    mp = g.add_source(wf.Source_Builder(src).withParallelism(1)
                      .withOutputSchema([0]).withOutputBatchSize(batch).build())
    mp.add(B(**kw).withTBWindows(win, slide).withParallelism(par)
           .withOutputSchema([0]).build())
    mp.add_sink(wf.Sink_Builder(snk).withParallelism(1).build())
    g.run()
    per = {}
    for t2, k, v in zip(ts, key, val):
        per.setdefault(k, []).append((t2, v))
    F = {"sum": sum, "max": max, "min": min, "count": len}[agg]
    exp = Counter()
    for k, r in per.items():
        tss = sorted(t2 for t2, _ in r)
        t0, tmax = tss[0], tss[-1]
        w0 = max(0, -(-(t0 - win + 1) // slide))
        for w in range(w0, tmax // slide + 1):
            seg = [v for t2, v in r if w * slide <= t2 < w * slide + win]
            exp[(k, F(seg) if seg else 0)] += 1
    assert Counter(rows) == exp, (form, agg, win, slide, keys, max_gap, par)


@pytest.mark.parametrize("case", range(3))
def test_lateness_disorder_fuzz(case):
    """Out-of-order streams with lateness >= jitter: every tuple lands in
    its windows (no drops).  The first window of a key is anchored at its
    first ARRIVING tuple (reference window_replica initialization); the
    oracle reflects that (60-config campaign ran clean)."""
    from collections import Counter
    import numpy as np
    import windflow_amd as wf
    from windflow_amd.builders import Keyed_Windows_Builder
    rng = random.Random(950_000 + case * 7)
    n = rng.choice([800, 2500])
    keys = rng.choice([1, 3, 9])
    slide = rng.choice([5, 20])
    win = slide * rng.randint(1, 5)
    agg = rng.choice(["sum", "max", "count"])
    par = rng.randint(1, 3)
    batch = rng.choice([64, 256])
    J = rng.choice([3, 10, 40])
    L = J + rng.randint(0, 20)
    ts = [max(1, i + rng.randint(-J, J)) for i in range(1, n + 1)]
    key = [rng.randrange(keys) for _ in range(n)]
    val = [rng.randint(-30, 30) for _ in range(n)]
    state = dict(pos=0, wm=0)

    def src(replica, parallelism):
        p = state['pos']
        if p >= n:
            return None
        q = min(p + batch, n)
        state['pos'] = q
        state['wm'] = max(state['wm'], max(ts[p:q]) - J)
        return dict(ts=np.array(ts[p:q], np.int64),
                    key=np.array(key[p:q], np.uint64),
                    c0=np.array(val[p:q], np.int64),
                    watermark=int(state['wm']))

    rows = []

    def snk(cols):
        for i in range(len(cols['key'])):
            rows.append((int(cols['key'][i]), int(cols['c0'][i])))

    g = wf.PipeGraph("late")
    mp = g.add_source(wf.Source_Builder(src).withParallelism(1)
                      .withOutputSchema([0]).withOutputBatchSize(batch).build())
    mp.add(Keyed_Windows_Builder(func=(agg, 0)).withTBWindows(win, slide)
           .withLateness(L).withParallelism(par).withOutputSchema([0]).build())
    mp.add_sink(wf.Sink_Builder(snk).withParallelism(1).build())
    g.run()
    per, first_arrival = {}, {}
    for t, k, v in zip(ts, key, val):
        per.setdefault(k, []).append((t, v))
        first_arrival.setdefault(k, t)
    F = {"sum": sum, "max": max, "count": len}[agg]
    exp = Counter()
    for k, r in per.items():
        tmax = max(t for t, _ in r)
        ta = first_arrival[k]
        w0 = max(0, -(-(ta - win + 1) // slide))
        for w in range(w0, tmax // slide + 1):
            seg = [v for t, v in r if w * slide <= t < w * slide + win]
            exp[(k, F(seg) if seg else 0)] += 1
    assert Counter(rows) == exp, (agg, win, slide, keys, J, L, par)


@pytest.mark.parametrize("case", range(2))
def test_tiny_batch_oversubscribed_fuzz(case):
    """Per-tuple-sized batches (1-4) with replica oversubscription (up to
    12 threads): punctuation cadence and watermark folding at the extreme
    (60-config campaign ran clean)."""
    from test_windows import run_graph, seq_stream, oracle_cb, oracle_tb, got_counter
    rng = random.Random(8_000_000 + case * 3)
    kind = rng.choice(list(BUILDERS))
    agg = rng.choice(["sum", "max", "min", "count"])
    wt = rng.choice(["cb", "tb"])
    slide = rng.choice([2, 5, 10])
    win = slide * rng.randint(1, 5)
    n_keys = rng.choice([1, 3, 7])
    batch = rng.choice([1, 2, 4])
    par = rng.choice([1, 4, 8, 12])
    stream = rng.choice([150, 400])
    b = BUILDERS[kind]((agg, 0))
    b = (b.withCBWindows(win, slide) if wt == "cb" else b.withTBWindows(win, slide))
    op = b.withParallelism(par).withOutputSchema([0]).build()
    rows = run_graph(op, stream_len=stream, n_keys=n_keys, batch=batch,
                     mode=wf.ExecutionMode.DEFAULT)
    exp = (oracle_cb if wt == "cb" else oracle_tb)(seq_stream(stream, n_keys),
                                                   win, slide, agg)
    assert got_counter(rows) == exp, (kind, agg, wt, win, slide, n_keys, batch, par)
