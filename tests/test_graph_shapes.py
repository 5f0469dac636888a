"""Merge/split DAG topologies (reference: tests/graph_tests, merge_tests,
split_tests) with the same randomized differential invariant."""
import random

import numpy as np

import windflow_amd as wf
from windflow_amd import native


def test_merge_two_pipes():
    # Source1->Map \
    #               +--> Filter -> Sink    (test_graph_1 shape, single sink)
    # Source2->Map /
    rng = random.Random(7)
    stream_len = 15000
    exp_one = [3 * v + 1 for v in range(1, stream_len + 1)]
    exp = sum(x for x in exp_one if x % 5 != 0) * 2  # two sources par=1
    for _ in range(3):
        g = wf.PipeGraph("merge")
        s1 = (wf.Source_Builder(native.seq_source(stream_len, 11, 128))
              .withParallelism(1).withOutputSchema([0]).build())
        s2 = (wf.Source_Builder(native.seq_source(stream_len, 11, 128))
              .withParallelism(1).withOutputSchema([0]).build())
        mp1 = g.add_source(s1)
        mp2 = g.add_source(s2)
        mp1.add(wf.Map_Builder(native.affine_map(0, 3, 1))
                .withParallelism(rng.randint(1, 3)).withOutputSchema([0]).build())
        mp2.add(wf.Map_Builder(native.affine_map(0, 3, 1))
                .withParallelism(rng.randint(1, 3)).withOutputSchema([0]).build())
        mp = mp1.merge(mp2)
        mp.add(wf.Filter_Builder(native.mod_filter(0, 5, 0))
               .withParallelism(rng.randint(1, 3)).withOutputSchema([0]).build())
        snk = wf.Sink_Builder(native.sum_sink(0)).withParallelism(rng.randint(1, 2)).build()
        mp.add_sink(snk)
        g.run()
        assert g.sink_sum(snk) == exp


def test_split_two_branches():
    # Source -> split(v%2) -> [branch0 -> SinkA, branch1 -> Map -> SinkB]
    stream_len = 10000
    g = wf.PipeGraph("split")
    src = (wf.Source_Builder(native.seq_source(stream_len, 5, 64))
           .withParallelism(1).withOutputSchema([0]).build())
    mp = g.add_source(src)
    mp.split(native.split_mod(0), 2)
    b0 = mp.select(0)
    b1 = mp.select(1)
    snk_a = wf.Sink_Builder(native.sum_sink(0)).withParallelism(1).build()
    b0.add_sink(snk_a)
    b1.add(wf.Map_Builder(native.affine_map(0, 1, 100)).withParallelism(2)
           .withOutputSchema([0]).build())
    snk_b = wf.Sink_Builder(native.sum_sink(0)).withParallelism(1).build()
    b1.add_sink(snk_b)
    g.run()
    evens = sum(v for v in range(1, stream_len + 1) if v % 2 == 0)
    odds = sum(v + 100 for v in range(1, stream_len + 1) if v % 2 == 1)
    assert g.sink_sum(snk_a) == evens
    assert g.sink_sum(snk_b) == odds


def test_python_callbacks_pipeline():
    """Per-batch numpy callbacks: source, map, filter, flatmap, sink."""
    stream_len = 5000
    state = dict(pos=0, total=0, n=0)

    def pysource(replica, par):
        if state['pos'] >= stream_len:
            return None
        n = min(512, stream_len - state['pos'])
        v = np.arange(state['pos'] + 1, state['pos'] + n + 1, dtype=np.int64)
        state['pos'] += n
        return dict(c0=v, ts=v, key=(v % 9).astype(np.uint64),
                    watermark=int(v[-1]))

    def pymap(cols):
        cols['c0'][:] = cols['c0'] * 2 + 1

    def pyfilter(cols):
        return (cols['c0'] % 3 != 0)

    def pyflat(cols):
        v = np.repeat(cols['c0'], 2)
        return dict(c0=v, ts=np.repeat(cols['ts'], 2), key=np.repeat(cols['key'], 2))

    def pysink(cols):
        state['total'] += int(cols['c0'].sum())
        state['n'] += len(cols['c0'])

    g = wf.PipeGraph("py")
    mp = g.add_source(wf.Source_Builder(pysource).withParallelism(1)
                      .withOutputSchema([0]).build())
    mp.add(wf.Map_Builder(pymap).withParallelism(1).withOutputSchema([0]).build())
    mp.add(wf.Filter_Builder(pyfilter).withParallelism(1).withOutputSchema([0]).build())
    mp.add(wf.FlatMap_Builder(pyflat).withParallelism(1).withOutputSchema([0]).build())
    mp.add_sink(wf.Sink_Builder(pysink).withParallelism(1).build())
    g.run()
    vals = [2 * v + 1 for v in range(1, stream_len + 1)]
    keep = [x for x in vals if x % 3 != 0]
    assert state['total'] == sum(keep) * 2
    assert state['n'] == len(keep) * 2


def test_broadcast_routing():
    """BROADCAST: every replica of the consumer sees every tuple."""
    from windflow_amd.operators import Operator
    stream_len = 3000
    g = wf.PipeGraph("bc")
    src = (wf.Source_Builder(native.seq_source(stream_len, 3, 128))
           .withParallelism(1).withOutputSchema([0]).build())
    mp = g.add_source(src)
    snk_op = wf.Sink_Builder(native.sum_sink(0)).withParallelism(3).build()
    snk_op.broadcast_input = True
    mp.add_sink(snk_op)
    g.run()
    exp = sum(range(1, stream_len + 1)) * 3
    assert g.sink_sum(snk_op) == exp


def test_split_then_merge():
    """Source -> split(v%2) -> [b0 -> Map(+100), b1 -> Map(+200)] -> merge
    -> Sink (reference pipegraph merge of split children)."""
    stream_len = 8000
    g = wf.PipeGraph("sm")
    src = (wf.Source_Builder(native.seq_source(stream_len, 5, 64))
           .withParallelism(1).withOutputSchema([0]).build())
    mp = g.add_source(src)
    mp.split(native.split_mod(0), 2)
    b0 = mp.select(0)
    b1 = mp.select(1)
    b0.add(wf.Map_Builder(native.affine_map(0, 1, 100)).withParallelism(2)
           .withOutputSchema([0]).build())
    b1.add(wf.Map_Builder(native.affine_map(0, 1, 200)).withParallelism(1)
           .withOutputSchema([0]).build())
    merged = b0.merge(b1)
    snk = wf.Sink_Builder(native.sum_sink(0)).withParallelism(2).build()
    merged.add_sink(snk)
    g.run()
    exp = sum(v + 100 for v in range(1, stream_len + 1) if v % 2 == 0) + \
          sum(v + 200 for v in range(1, stream_len + 1) if v % 2 == 1)
    assert g.sink_sum(snk) == exp


def test_merge_three_pipes():
    stream_len = 5000
    g = wf.PipeGraph("m3")
    pipes = []
    for i in range(3):
        s = (wf.Source_Builder(native.seq_source(stream_len, 3, 128,
                                                 value_offset=i * 100000))
             .withParallelism(1).withOutputSchema([0]).build())
        pipes.append(g.add_source(s))
    mp = pipes[0].merge(pipes[1], pipes[2])
    snk = wf.Sink_Builder(native.sum_sink(0)).withParallelism(1).build()
    mp.add_sink(snk)
    g.run()
    exp = sum(sum(range(1 + i * 100000, stream_len + 1 + i * 100000))
              for i in range(3))
    assert g.sink_sum(snk) == exp


def test_chained_windows():
    """window -> window: keyed tumbling sums then keyed tumbling sums of
    sums — exercises watermark propagation through a firing operator."""
    from windflow_amd.builders import Keyed_Windows_Builder
    from collections import defaultdict, Counter
    stream_len, keys = 4000, 5
    g = wf.PipeGraph("ww")
    src = (wf.Source_Builder(native.seq_source(stream_len, keys, 128))
           .withParallelism(1).withOutputSchema([0]).build())
    mp = g.add_source(src)
    mp.add(Keyed_Windows_Builder(func=("sum", 0)).withCBWindows(10, 10)
           .withOutputSchema([0]).build())
    mp.add(Keyed_Windows_Builder(func=("sum", 0)).withCBWindows(4, 4)
           .withOutputSchema([0]).build())
    rows = []

    def sink(cols):
        for i in range(len(cols['ts'])):
            rows.append((int(cols['key'][i]), cols['c0'][i].item()))

    mp.add_sink(wf.Sink_Builder(sink).withParallelism(1).build())
    g.run()
    # oracle: per key tumbling-10 sums, then tumbling-4 sums over those
    per = defaultdict(list)
    for v in range(1, stream_len + 1):
        per[v % keys].append(v)
    exp = Counter()
    for k, vals in per.items():
        l1 = [sum(vals[i:i + 10]) for i in range(0, len(vals), 10)]
        for i in range(0, len(l1), 4):
            exp[(k, sum(l1[i:i + 4]))] += 1
    assert Counter(rows) == exp


def test_rebalancing_routing():
    """REBALANCING: round-robin batches across replicas regardless of key."""
    stream_len = 6000
    g = wf.PipeGraph("rb")
    src = (wf.Source_Builder(native.seq_source(stream_len, 1, 64))
           .withParallelism(1).withOutputSchema([0]).build())
    mp = g.add_source(src)
    mp.add(wf.Map_Builder(native.affine_map(0, 1, 0)).withParallelism(3)
           .withRebalancing().withOutputSchema([0]).build())
    snk = wf.Sink_Builder(native.sum_sink(0)).withParallelism(1).build()
    mp.add_sink(snk)
    g.run()
    assert g.sink_sum(snk) == sum(range(1, stream_len + 1))


def test_ingress_time_policy():
    """INGRESS_TIME stamps arrival-clock timestamps (monotone, recent)."""
    import time
    seen = dict(lo=None, hi=None)

    def sink(cols):
        ts = cols['ts']
        if len(ts):
            seen['lo'] = int(ts.min()) if seen['lo'] is None else min(seen['lo'], int(ts.min()))
            seen['hi'] = int(ts.max()) if seen['hi'] is None else max(seen['hi'], int(ts.max()))

    t0 = time.monotonic_ns() // 1000
    g = wf.PipeGraph("it", wf.ExecutionMode.DEFAULT, wf.TimePolicy.INGRESS_TIME)
    src = (wf.Source_Builder(native.seq_source(5000, 3, 256))
           .withParallelism(1).withOutputSchema([0]).build())
    mp = g.add_source(src)
    mp.add_sink(wf.Sink_Builder(sink).withParallelism(1).build())
    g.run()
    t1 = time.monotonic_ns() // 1000
    assert seen['lo'] is not None
    assert t0 <= seen['lo'] <= seen['hi'] <= t1


def test_stats_records():
    g = wf.PipeGraph("st")
    src = (wf.Source_Builder(native.seq_source(1000, 3, 128))
           .withParallelism(2).withOutputSchema([0]).build())
    mp = g.add_source(src)
    snk = wf.Sink_Builder(native.count_sink()).withParallelism(1).build()
    mp.add_sink(snk)
    g.run()
    st = g.stats()
    assert len(st) == 3
    tuples_in = sum(r['tuples_in'] for r in st if r['name'] == 'sink')
    assert tuples_in == 2000
    assert g.sink_count(snk) == 2000
    js = g.stats_json()
    assert 'operators' in js


def test_replica_error_propagates():
    """A failing user logic aborts the graph and surfaces the error via
    run() instead of killing the process (improvement over the reference,
    which exits(EXIT_FAILURE))."""
    import pytest

    def bad_source(replica, par):
        raise ValueError("boom")

    g = wf.PipeGraph("err")
    mp = g.add_source(wf.Source_Builder(bad_source).withParallelism(1)
                      .withOutputSchema([0]).build())
    mp.add_sink(wf.Sink_Builder(native.count_sink()).withParallelism(1).build())
    with pytest.raises(RuntimeError, match="boom"):
        g.run()


def test_partial_merge_of_split_branches():
    """Split 3 ways; merge branches 0+2, keep branch 1 separate (the
    reference's merge-partial shape, pipegraph.hpp:387)."""
    n = 9000
    g = wf.PipeGraph("pm")
    src = (wf.Source_Builder(native.seq_source(n, 5, 64))
           .withParallelism(1).withOutputSchema([0]).build())
    mp = g.add_source(src)
    mp.split(native.split_mod(0), 3)
    b0, b1, b2 = mp.select(0), mp.select(1), mp.select(2)
    merged = b0.merge(b2)
    s02 = wf.Sink_Builder(native.sum_sink(0)).withParallelism(2).build()
    merged.add_sink(s02)
    s1 = wf.Sink_Builder(native.sum_sink(0)).withParallelism(1).build()
    b1.add_sink(s1)
    g.run()
    exp02 = sum(v for v in range(1, n + 1) if v % 3 in (0, 2))
    exp1 = sum(v for v in range(1, n + 1) if v % 3 == 1)
    assert g.sink_sum(s02) == exp02
    assert g.sink_sum(s1) == exp1


def test_nested_split():
    """Split a split branch again (nested DAG)."""
    n = 8000
    g = wf.PipeGraph("ns")
    src = (wf.Source_Builder(native.seq_source(n, 5, 64))
           .withParallelism(1).withOutputSchema([0]).build())
    mp = g.add_source(src)
    mp.split(native.split_mod(0), 2)          # by v % 2
    b0, b1 = mp.select(0), mp.select(1)
    s1 = wf.Sink_Builder(native.sum_sink(0)).withParallelism(1).build()
    b1.add_sink(s1)
    # halve evens again: v/2 parity
    b0.add(wf.Map_Builder(native.affine_map(0, 1, 0)).withParallelism(1)
           .withOutputSchema([0]).build())
    b0.split(native.split_mod(0), 2)          # v%2==0 -> all to branch 0...
    c0, c1 = b0.select(0), b0.select(1)
    sc0 = wf.Sink_Builder(native.sum_sink(0)).withParallelism(1).build()
    sc1 = wf.Sink_Builder(native.count_sink()).withParallelism(1).build()
    c0.add_sink(sc0)
    c1.add_sink(sc1)
    g.run()
    evens = [v for v in range(1, n + 1) if v % 2 == 0]
    odds = sum(v for v in range(1, n + 1) if v % 2 == 1)
    assert g.sink_sum(s1) == odds
    assert g.sink_sum(sc0) == sum(evens)   # evens % 2 == 0 -> branch 0
    assert g.sink_count(sc1) == 0


def test_python_split_branch_ids():
    """Python split fn(cols) -> int32 branch ids (reference
    splitting_emitter.hpp integral_t form)."""
    import numpy as np
    n = 5000
    g = wf.PipeGraph("pysplit")
    src = (wf.Source_Builder(native.seq_source(n, 1, 256))
           .withParallelism(1).withOutputSchema([0]).build())
    mp = g.add_source(src)
    br = mp.split(lambda cols: (cols['c0'] % 3).astype(np.int32), 3)
    sinks = []
    for i in range(3):
        s = wf.Sink_Builder(native.sum_sink(0)).withParallelism(1).build()
        br.select(i).add_sink(s)
        sinks.append(s)
    g.run()
    exp = [sum(v for v in range(1, n + 1) if v % 3 == i) for i in range(3)]
    assert [g.sink_sum(s) for s in sinks] == exp


def test_python_split_multi_branch_masks():
    """Python split fn(cols) -> list of per-branch masks: one tuple may go
    to SEVERAL branches (reference vector<integral_t> form)."""
    import numpy as np
    n = 3000
    g = wf.PipeGraph("pysplit2")
    src = (wf.Source_Builder(native.seq_source(n, 1, 128))
           .withParallelism(1).withOutputSchema([0]).build())
    mp = g.add_source(src)

    def route(cols):
        v = cols['c0']
        return [v % 2 == 0, v % 3 == 0]   # overlap: multiples of 6 hit both

    br = mp.split(route, 2)
    sinks = []
    for i in range(2):
        s = wf.Sink_Builder(native.sum_sink(0)).withParallelism(1).build()
        br.select(i).add_sink(s)
        sinks.append(s)
    g.run()
    exp0 = sum(v for v in range(1, n + 1) if v % 2 == 0)
    exp1 = sum(v for v in range(1, n + 1) if v % 3 == 0)
    assert [g.sink_sum(s) for s in sinks] == [exp0, exp1]


def test_python_pipeline_under_backpressure(monkeypatch):
    """Queue capacity 2: a chain of python operators must not deadlock.
    Guards the GIL discipline in the python logic wrappers — holding the
    GIL across a blocking queue push wedges the pipeline once queues fill
    (the downstream python stage needs the GIL to drain that queue)."""
    monkeypatch.setenv("WFA_QUEUE_CAP", "2")
    stream_len = 4000
    state = dict(pos=0, total=0)

    def pysource(replica, par):
        if state['pos'] >= stream_len:
            return None
        n = min(256, stream_len - state['pos'])
        v = np.arange(state['pos'] + 1, state['pos'] + n + 1, dtype=np.int64)
        state['pos'] += n
        return dict(c0=v, ts=v, key=(v % 5).astype(np.uint64),
                    watermark=int(v[-1]))

    def pyflat(cols):
        v = np.repeat(cols['c0'], 3)
        return dict(c0=v, ts=np.repeat(cols['ts'], 3),
                    key=np.repeat(cols['key'], 3))

    def pysink(cols):
        state['total'] += int(cols['c0'].sum())

    g = wf.PipeGraph("bp")
    mp = g.add_source(wf.Source_Builder(pysource).withParallelism(1)
                      .withOutputSchema([0]).withOutputBatchSize(64).build())
    mp.add(wf.FlatMap_Builder(pyflat).withParallelism(1)
           .withOutputSchema([0]).withOutputBatchSize(64).build())
    def ident_map(cols):
        cols['c0'][:] = cols['c0']
    mp.add(wf.Map_Builder(ident_map).withParallelism(1)
           .withOutputSchema([0]).withOutputBatchSize(64).build())
    mp.add_sink(wf.Sink_Builder(pysink).withParallelism(1).build())
    g.run()
    assert state['total'] == 3 * stream_len * (stream_len + 1) // 2


def test_three_way_merge():
    """Merge of three pipes (reference MultiPipe::merge accepts N pipes)."""
    g = wf.PipeGraph("m3")
    mps = []
    for off in (0, 1000000, 2000000):
        s = (wf.Source_Builder(native.seq_source(3000, 4, 256, value_offset=off))
             .withParallelism(1).withOutputSchema([0]).build())
        mps.append(g.add_source(s))
    mp = mps[0].merge(mps[1], mps[2])
    snk = wf.Sink_Builder(native.sum_sink(0)).withParallelism(1).build()
    mp.add_sink(snk)
    g.run()
    exp = sum(sum(range(off + 1, off + 3001)) for off in (0, 1000000, 2000000))
    assert g.sink_sum(snk) == exp


def test_windows_inside_split_branches():
    """split -> keyed windows per branch -> merge -> sink (reference
    graph_tests compose windows inside MultiPipe branches)."""
    from collections import Counter
    from windflow_amd.builders import Keyed_Windows_Builder
    import sys as _s, os as _o
    _s.path.insert(0, _o.path.dirname(__file__))
    from test_windows import seq_stream, oracle_cb
    n, keys, win, slide = 4000, 6, 30, 10
    g = wf.PipeGraph("ws")
    mp = g.add_source(wf.Source_Builder(native.seq_source(n, keys, 256))
                      .withParallelism(1).withOutputSchema([0]).build())
    br = mp.split(native.split_mod(0), 2)   # by value parity
    rows = []

    def snk(cols):
        for i in range(len(cols['key'])):
            rows.append((int(cols['key'][i]), int(cols['c0'][i])))

    outs = []
    for i in range(2):
        b = br.select(i)
        b.add(Keyed_Windows_Builder(func=("sum", 0)).withCBWindows(win, slide)
              .withParallelism(2).withOutputSchema([0]).build())
        outs.append(b)
    merged = outs[0].merge(outs[1])
    merged.add_sink(wf.Sink_Builder(snk).withParallelism(1).build())
    g.run()
    # oracle: each branch windows its parity sub-stream independently
    exp = Counter()
    for parity in (0, 1):
        per = {}
        for v in range(1, n + 1):
            if v % 2 == parity:
                per.setdefault(v % keys, []).append((v, v))
        for (k, s), c in oracle_cb(per, win, slide, "sum").items():
            exp[(k, s)] += c
    assert Counter(rows) == exp


def test_gpu_graph_lowering_on_cpu():
    """All four GPU bench graphs construct and lower (build_engine) on a
    GPU-less box: validation, window lowering and spec generation are
    python/host-side and must not regress where there is no device."""
    import bench
    from windflow_amd import _core
    for name, build in (
            ("ffat", lambda: bench.build_ffat_graph(1_000_000, 250_000, 8192,
                                                    1000, 100, 0, 1, 0)),
            ("mapfilter", lambda: bench.build_mapfilter_graph(
                1_000_000, 250_000, 8192, 0, 1, 0)),
            ("a2a", lambda: bench.build_a2a_graph(
                1_000_000, 250_000, 8192, 0, 1, 0,
                (0, 1, _core.rccl_unique_id()))),
            ("ffat_x", lambda: bench.build_ffat_x_graph(
                1_000_000, 250_000, 8192, 1000, 100, 0, 1, 0,
                (0, 1, _core.rccl_unique_id()))),
    ):
        g, snk = build()
        e = g.build_engine()
        assert e is not None, name


def test_composite_split_windows_merge_fuzz():
    """Randomized composite DAGs: split -> per-branch windows (independent
    configs, keyed and FFAT forms) -> merge (40-config campaign clean)."""
    import random
    from collections import Counter
    from windflow_amd.builders import Keyed_Windows_Builder, Ffat_Windows_Builder
    import sys as _s, os as _o
    _s.path.insert(0, _o.path.dirname(__file__))
    from test_windows import oracle_cb
    for case in (3, 17):
        rng = random.Random(12_000_000 + case)
        n = rng.choice([2000, 6000])
        keys = rng.choice([4, 9])
        nb = rng.choice([2, 3])
        cfgs = []
        for i in range(nb):
            slide = rng.choice([5, 10])
            win = slide * rng.randint(1, 4)
            agg = rng.choice(["sum", "max", "count"])
            par = rng.randint(1, 3)
            form = rng.choice([Keyed_Windows_Builder, Ffat_Windows_Builder])
            cfgs.append((win, slide, agg, par, form))
        g = wf.PipeGraph("comp")
        mp = g.add_source(wf.Source_Builder(native.seq_source(n, keys, 256))
                          .withParallelism(1).withOutputSchema([0]).build())
        br = mp.split(native.split_mod(0), nb)
        rows = []

        def snk(cols):
            for i in range(len(cols['key'])):
                rows.append((int(cols['key'][i]), int(cols['c0'][i])))

        outs = []
        for i, (win, slide, agg, par, B) in enumerate(cfgs):
            bsel = br.select(i)
            kw = ({"func": (agg, 0)} if B is Keyed_Windows_Builder
                  else {"comb": (agg, 0)})
            bsel.add(B(**kw).withCBWindows(win, slide).withParallelism(par)
                     .withOutputSchema([0]).build())
            outs.append(bsel)
        merged = outs[0].merge(*outs[1:])
        merged.add_sink(wf.Sink_Builder(snk).withParallelism(1).build())
        g.run()
        exp = Counter()
        for i, (win, slide, agg, par, B) in enumerate(cfgs):
            per = {}
            for v in range(1, n + 1):
                if v % nb == i:
                    per.setdefault(v % keys, []).append((v, v))
            for (k, s), c in oracle_cb(per, win, slide, agg).items():
                exp[(k, s)] += c
        assert Counter(rows) == exp, case


def test_cross_device_chain_lowering():
    """withDevice(0) -> withDevice(1) chains lower to SEPARATE nodes with
    the right device per op (cross-GPU forward path): chain()
    must not fuse across devices, and the lowered graph keeps the edge."""
    from windflow_amd import native_gpu
    from windflow_amd.builders_gpu import (Source_GPU_Builder, Map_GPU_Builder,
                                           Sink_GPU_Builder)
    src = (Source_GPU_Builder(native_gpu.gpu_source(1000, 7, 100, vdt=0))
           .withOutputSchema([0]).withOutputBatchSize(100)
           .withDevice(0).build())
    m0 = (Map_GPU_Builder(native_gpu.gpu_affine_map(0, 2, 0, dtype=0))
          .withOutputSchema([0]).withOutputBatchSize(100)
          .withDevice(0).build())
    m1 = (Map_GPU_Builder(native_gpu.gpu_affine_map(0, 3, 0, dtype=0))
          .withOutputSchema([0]).withOutputBatchSize(100)
          .withDevice(1).build())
    snk = Sink_GPU_Builder(native_gpu.gpu_count_sink()).withDevice(1).build()
    g = wf.PipeGraph("xdev")
    mp = g.add_source(src)
    mp.chain(m0)     # same device: fuses
    mp.chain(m1)     # device 0 -> 1: must NOT fuse
    mp.chain_sink(snk)
    assert len(g.nodes) == 2, "cross-device chain must break the fusion"
    assert g.nodes[0].device == 0 and g.nodes[1].device == 1
    assert len(g.nodes[0].ops) == 2 and len(g.nodes[1].ops) == 2
    assert g.edges[0]['src'] == 0 and g.edges[0]['dst'] == 1
