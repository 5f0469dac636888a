"""Edge conditions: batch size 1 (per-tuple 'singles'), empty streams,
branches that receive nothing, zero-filter pipelines, watermark-only flow."""
import numpy as np

import windflow_amd as wf
from windflow_amd import native
from windflow_amd.builders import Keyed_Windows_Builder


def test_batch_size_one_singles():
    """out_batch=1 == the reference's unbatched Single_t mode."""
    n = 500
    g = wf.PipeGraph("s1")
    mp = g.add_source(wf.Source_Builder(native.seq_source(n, 3, 1))
                      .withParallelism(1).withOutputSchema([0])
                      .withOutputBatchSize(1).build())
    mp.add(wf.Map_Builder(native.affine_map(0, 2, 1)).withParallelism(2)
           .withOutputSchema([0]).withOutputBatchSize(1).withKeyBy(0).build())
    snk = wf.Sink_Builder(native.sum_sink(0)).withParallelism(1).build()
    mp.add_sink(snk)
    g.run()
    assert g.sink_sum(snk) == sum(2 * v + 1 for v in range(1, n + 1))


def test_empty_stream():
    g = wf.PipeGraph("e0")
    mp = g.add_source(wf.Source_Builder(native.seq_source(0, 1, 64))
                      .withParallelism(1).withOutputSchema([0]).build())
    mp.add(Keyed_Windows_Builder(func=("sum", 0)).withCBWindows(10, 5)
           .withOutputSchema([0]).build())
    snk = wf.Sink_Builder(native.count_sink()).withParallelism(1).build()
    mp.add_sink(snk)
    g.run()
    assert g.sink_count(snk) == 0


def test_filter_all_dropped_keeps_watermarks():
    """Everything filtered out: downstream windows still terminate (the
    punctuation keep-alive path, reference filter.hpp:151)."""
    n = 2000
    g = wf.PipeGraph("fa")
    mp = g.add_source(wf.Source_Builder(native.seq_source(n, 3, 128))
                      .withParallelism(1).withOutputSchema([0]).build())
    # keep x % 1 == 1  -> never true: drops everything
    mp.add(wf.Filter_Builder(native.mod_filter(0, 1, 1, keep_eq=True))
           .withParallelism(1).withOutputSchema([0]).build())
    mp.add(Keyed_Windows_Builder(func=("sum", 0)).withTBWindows(100, 50)
           .withOutputSchema([0]).build())
    snk = wf.Sink_Builder(native.count_sink()).withParallelism(1).build()
    mp.add_sink(snk)
    g.run()   # must terminate; no windows fire
    assert g.sink_count(snk) == 0


def test_split_branch_receives_nothing():
    """A split branch whose predicate never matches still EOSes cleanly."""
    n = 1000
    g = wf.PipeGraph("sb")
    mp = g.add_source(wf.Source_Builder(native.seq_source(n, 1, 64,
                                                          value_offset=0))
                      .withParallelism(1).withOutputSchema([0]).build())
    # values 1..n; mod-3 split: branch picks x%3; use 3 branches — all hit.
    # Force an empty branch by mapping values to even first.
    mp.add(wf.Map_Builder(native.affine_map(0, 2, 0)).withParallelism(1)
           .withOutputSchema([0]).build())
    mp.split(native.split_mod(0), 2)   # x even -> branch 0 always
    b0, b1 = mp.select(0), mp.select(1)
    s0 = wf.Sink_Builder(native.count_sink()).withParallelism(1).build()
    s1 = wf.Sink_Builder(native.count_sink()).withParallelism(1).build()
    b0.add_sink(s0)
    b1.add_sink(s1)
    g.run()
    assert g.sink_count(s0) == n
    assert g.sink_count(s1) == 0


def test_flatmap_zero_and_many():
    """FlatMap emitting 0 rows for some batches and many for others."""
    state = dict(pos=0)

    def src(replica, par):
        if state['pos'] >= 10:
            return None
        i = state['pos']
        state['pos'] += 1
        v = np.arange(i * 10, i * 10 + 10, dtype=np.int64)
        return dict(c0=v, ts=v + 1, key=np.zeros(10, dtype=np.uint64),
                    watermark=int(v[-1] + 1))

    def fl(cols):
        v = cols['c0']
        keep = v[v % 20 == 0]          # most batches -> empty output
        out = np.repeat(keep, 3)
        return dict(c0=out, ts=np.repeat(cols['ts'][v % 20 == 0], 3),
                    key=np.zeros(len(out), dtype=np.uint64))

    g = wf.PipeGraph("fm")
    mp = g.add_source(wf.Source_Builder(src).withParallelism(1)
                      .withOutputSchema([0]).withOutputBatchSize(16).build())
    mp.add(wf.FlatMap_Builder(fl).withParallelism(1).withOutputSchema([0]).build())
    snk = wf.Sink_Builder(native.sum_sink(0)).withParallelism(1).build()
    mp.add_sink(snk)
    g.run()
    exp = 3 * sum(v for v in range(0, 100) if v % 20 == 0)
    assert g.sink_sum(snk) == exp


def test_multi_column_mixed_dtypes_through_ops():
    """3 payload columns (i64, f64, f32) through python source -> map ->
    filter -> split -> sinks: per-column values stay aligned per row."""
    import numpy as np
    n_batches, bsz = 20, 500
    state = dict(i=0)

    def src(replica, parallelism):
        if state['i'] >= n_batches:
            return None
        base = state['i'] * bsz
        state['i'] += 1
        idx = np.arange(base, base + bsz)
        return {"ts": (idx + 1).astype(np.int64),
                "key": (idx % 7).astype(np.uint64),
                "c0": idx.astype(np.int64),
                "c1": (idx * 0.5).astype(np.float64),
                "c2": (idx * 2.0).astype(np.float32),
                "watermark": int(base + bsz)}

    def mapper(cols):
        cols['c1'][:] = cols['c1'] * 2.0          # f64: now == c0
        cols['c2'][:] = cols['c2'] + 1.0          # f32: now == 2*c0+1

    def fltr(cols):
        return (cols['c0'] % 3 == 0)

    rows = []

    def snk(cols):
        for i in range(len(cols['c0'])):
            rows.append((int(cols['c0'][i]), float(cols['c1'][i]),
                         float(cols['c2'][i]), int(cols['key'][i])))

    g = wf.PipeGraph("mc")
    mp = g.add_source(wf.Source_Builder(src).withParallelism(1)
                      .withOutputSchema([0, 1, 2]).build())
    mp.add(wf.Map_Builder(mapper).withParallelism(2)
           .withOutputSchema([0, 1, 2]).build())
    mp.add(wf.Filter_Builder(fltr).withParallelism(2)
           .withOutputSchema([0, 1, 2]).build())
    br = mp.split(lambda cols: (cols['c0'] % 2).astype(np.int32), 2)
    s0 = wf.Sink_Builder(snk).withParallelism(1).build()
    s1 = wf.Sink_Builder(snk).withParallelism(1).build()
    br.select(0).add_sink(s0)
    br.select(1).add_sink(s1)
    g.run()

    total = n_batches * bsz
    exp = [(v, float(v), float(np.float32(2 * v + 1)), v % 7)
           for v in range(total) if v % 3 == 0]
    assert sorted(rows) == sorted(exp)


def test_reduce_initial_state_and_broadcast():
    """withInitialState seeds every key's accumulator; withBroadcast routes
    all inputs to every replica (reference builders.hpp:252/:627)."""
    n, keys, init = 3000, 5, 1000
    g = wf.PipeGraph("ribc")
    mp = g.add_source(wf.Source_Builder(native.seq_source(n, keys, 256))
                      .withParallelism(1).withOutputSchema([0]).build())
    red = (wf.Reduce_Builder(native.keyed_sum_reduce(0)).withInitialState(init)
           .withParallelism(2).withOutputSchema([0]).build())
    mp.add(red)
    snk = (wf.Sink_Builder(native.last_per_key_sink(0))
           .withParallelism(1).build())
    mp.add_sink(snk)
    g.run()
    per = {k: init for k in range(keys)}
    for v in range(1, n + 1):
        per[v % keys] += v
    assert g.sink_sum(snk) == sum(per.values())

    # broadcast: every map replica sees every tuple -> count multiplies
    g2 = wf.PipeGraph("bc")
    mp2 = g2.add_source(wf.Source_Builder(native.seq_source(n, keys, 256))
                        .withParallelism(1).withOutputSchema([0]).build())
    mp2.add(wf.Map_Builder(native.affine_map(0, 1, 0)).withBroadcast()
            .withParallelism(3).withOutputSchema([0]).build())
    snk2 = wf.Sink_Builder(native.count_sink()).withParallelism(1).build()
    mp2.add_sink(snk2)
    g2.run()
    assert g2.sink_count(snk2) == 3 * n


def test_invalid_window_extents_rejected():
    """win=0/slide=0 used to HANG the engine (infinite window-open loop);
    all window builders now reject non-positive extents up front."""
    import pytest
    from windflow_amd.builders import Keyed_Windows_Builder, Ffat_Windows_Builder
    from windflow_amd.persistent import P_Keyed_Windows_Builder
    from windflow_amd import native_gpu
    for bad in ((0, 0), (10, 0), (0, 10), (-5, 2), (5, -2)):
        with pytest.raises(ValueError):
            Keyed_Windows_Builder(func=("sum", 0)).withCBWindows(*bad)
        with pytest.raises(ValueError):
            Ffat_Windows_Builder(comb=("sum", 0)).withTBWindows(*bad)
        with pytest.raises(ValueError):
            P_Keyed_Windows_Builder(sum).withCBWindows(*bad)
        with pytest.raises(ValueError):
            native_gpu.gpu_ffat_windows(0, 0, bad[0], bad[1])


def test_invalid_configs_rejected_cleanly():
    """parallelism < 1 is rejected at build; an out-of-range column index
    is a clean engine error, not a segfault."""
    import pytest
    with pytest.raises(ValueError):
        wf.Source_Builder(native.seq_source(10, 1, 8)).withParallelism(0)
    g = wf.PipeGraph("bad")
    mp = g.add_source(wf.Source_Builder(native.seq_source(100, 3, 32))
                      .withParallelism(1).withOutputSchema([0]).build())
    mp.add(wf.Map_Builder(native.affine_map(5, 1, 0))
           .withOutputSchema([0]).build())
    mp.add_sink(wf.Sink_Builder(native.count_sink()).build())
    with pytest.raises(RuntimeError, match="out of range"):
        g.run()


def test_merge_schema_and_select_bounds_validated():
    import pytest
    g = wf.PipeGraph("v")
    a = g.add_source(wf.Source_Builder(native.seq_source(100, 3, 32))
                     .withParallelism(1).withOutputSchema([0]).build())
    b = g.add_source(wf.Source_Builder(native.seq_source(100, 3, 32))
                     .withParallelism(1).withOutputSchema([1, 1]).build())
    with pytest.raises(TypeError, match="identical output schemas"):
        a.merge(b)
    g2 = wf.PipeGraph("v2")
    mp = g2.add_source(wf.Source_Builder(native.seq_source(100, 3, 32))
                       .withParallelism(1).withOutputSchema([0]).build())
    br = mp.split(native.split_mod(0), 2)
    with pytest.raises(IndexError):
        br.select(5)


def test_unterminated_pipe_rejected():
    import pytest
    g = wf.PipeGraph("nt")
    mp = g.add_source(wf.Source_Builder(native.seq_source(100, 3, 32))
                      .withParallelism(1).withOutputSchema([0]).build())
    mp.add(wf.Map_Builder(native.affine_map(0, 1, 0)).withOutputSchema([0]).build())
    with pytest.raises(RuntimeError, match="no consumers"):
        g.run()


def test_python_user_reduce():
    """Arbitrary user reduce fn(acc, v) -> acc' with per-key state and
    KEYBY routing (reference reduce.hpp arbitrary functor)."""
    n, n_keys, batch = 8000, 5, 256
    g = wf.PipeGraph("pyred")
    src = (wf.Source_Builder(native.seq_source(n, n_keys, batch))
           .withParallelism(1).withOutputSchema([0])
           .withOutputBatchSize(batch).build())
    mp = g.add_source(src)
    mp.add(wf.Reduce_Builder(lambda acc, v: max(acc, int(v)))
           .withInitialState(-1)
           .withParallelism(2).withOutputSchema([0])
           .withOutputBatchSize(batch).withKeyBy('carried').build())
    rows = dict(last={})

    def sink(cols):
        for k, v in zip(cols['key'].tolist(), cols['c0'].tolist()):
            rows['last'][k] = v

    snk = wf.Sink_Builder(sink).withParallelism(1).build()
    snk.out_schema = [0]
    mp.add_sink(snk)
    g.run()
    # final per-key running max == max v per key (seq: v=1..n, key=v%5)
    exp = {}
    for v in range(1, n + 1):
        exp[v % n_keys] = max(exp.get(v % n_keys, -1), v)
    assert rows['last'] == exp
