"""GPU operator numerics vs CPU (numpy fp32/fp64) oracles.

Every HIP kernel path is compared against a plain host reference of the
same op, per the project test policy.  All tests need an MI355X.
"""
import numpy as np
import pytest

import windflow_amd as wf
from windflow_amd import native, native_gpu
from windflow_amd.builders_gpu import (Source_GPU_Builder, Map_GPU_Builder,
                                       Filter_GPU_Builder, Reduce_GPU_Builder,
                                       Ffat_Windows_GPU_Builder, Sink_GPU_Builder)
from windflow_amd.synth import gen_batch, bf16_to_f32_np

pytestmark = pytest.mark.gpu

N_STREAM = 200_000
BATCH = 30_000  # deliberately not a divisor of the stream length


def gpu_graph(*gpu_ops, sink_cpu=True):
    g = wf.PipeGraph("gpu")
    mp = g.add_source(gpu_ops[0])
    for op in gpu_ops[1:]:
        mp.chain(op)
    if sink_cpu:
        snk = wf.Sink_Builder(native.sum_sink(0)).withParallelism(1).build()
        snk.out_schema = gpu_ops[-1].out_schema
        mp.add_sink(snk)
    else:
        snk = Sink_GPU_Builder(native_gpu.gpu_count_sink()).build()
        mp.chain_sink(snk)
    return g, snk


def test_gpu_generator_matches_oracle():
    """gpu_source(i64) -> CPU sum sink == numpy generator sum."""
    src = (Source_GPU_Builder(native_gpu.gpu_source(N_STREAM, 97, BATCH, vdt=0))
           .withOutputSchema([0]).withOutputBatchSize(BATCH).build())
    g, snk = gpu_graph(src)
    g.run()
    ts, key, val = gen_batch(N_STREAM, 0, 42, 97, 0)
    assert g.sink_sum(snk) == int(val.sum())
    assert g.sink_count(snk) == N_STREAM


def test_gpu_map_filter_i64():
    src = (Source_GPU_Builder(native_gpu.gpu_source(N_STREAM, 97, BATCH, vdt=0))
           .withOutputSchema([0]).withOutputBatchSize(BATCH).build())
    mp_ = (Map_GPU_Builder(native_gpu.gpu_affine_map(0, 3, 1, dtype=0))
           .withOutputSchema([0]).withOutputBatchSize(BATCH).build())
    fl = (Filter_GPU_Builder(native_gpu.gpu_mod_filter(0, 5, 0))
          .withOutputSchema([0]).withOutputBatchSize(BATCH).build())
    g, snk = gpu_graph(src, mp_, fl)
    g.run()
    _, _, val = gen_batch(N_STREAM, 0, 42, 97, 0)
    v = val * 3 + 1
    keep = v[v % 5 != 0]
    assert g.sink_sum(snk) == int(keep.sum())
    assert g.sink_count(snk) == len(keep)


def test_gpu_map_bf16_vectorized():
    src = (Source_GPU_Builder(native_gpu.gpu_source(N_STREAM, 97, BATCH, vdt=5))
           .withOutputSchema([5]).withOutputBatchSize(BATCH).build())
    mp_ = (Map_GPU_Builder(native_gpu.gpu_affine_map(0, 2.0, 0.25, dtype=5))
           .withOutputSchema([5]).withOutputBatchSize(BATCH).build())

    got = dict(s=0.0, n=0)

    def pysink(cols):
        got['s'] += float(bf16_to_f32_np(cols['c0']).sum())
        got['n'] += len(cols['c0'])

    g = wf.PipeGraph("bf16")
    p = g.add_source(src)
    p.chain(mp_)
    snk = wf.Sink_Builder(pysink).withParallelism(1).build()
    snk.out_schema = [5]
    p.add_sink(snk)
    g.run()
    _, _, val = gen_batch(N_STREAM, 0, 42, 97, 5)
    from windflow_amd.synth import f32_to_bf16_np
    ref = bf16_to_f32_np(f32_to_bf16_np(bf16_to_f32_np(val) * np.float32(2.0)
                                        + np.float32(0.25))).astype(np.float64).sum()
    assert got['n'] == N_STREAM
    assert abs(got['s'] - ref) <= 1e-6 * max(1.0, abs(ref))


def _seg_comb(seg, win, slide, comb):
    # fp32 pane-partials then pane combine — mirrors the GPU fold order
    pane = int(np.gcd(win, slide))
    segs = [seg[i:i + pane] for i in range(0, len(seg), pane)]
    if comb == "sum":
        partials = [np.sum(np.array(s, dtype=np.float32), dtype=np.float32)
                    for s in segs]
        return float(np.sum(np.array(partials, dtype=np.float64)))
    if comb == "min":
        return float(min(min(s) for s in segs))
    return float(max(max(s) for s in segs))


def _ffat_oracle(n, n_keys, win, slide, vdt=2, comb="sum", batch=BATCH):
    """Per-key sliding CB windows in generator order; windows at
    [w*slide, w*slide+win).  Returns {key: [result,...]}: full windows in
    firing order, then the EOS partial-window flush (every open window
    start w*slide < #tuples fires with the tail — engine EOS semantics,
    csrc windows.cpp FfatCpu on_eos / wfa_ffat_cb_flush)."""
    ts, key, val = gen_batch(n, 0, 42, n_keys, vdt)
    if vdt == 5:
        val = bf16_to_f32_np(val)
    elif vdt == 0:
        val = val.astype(np.float32)
    out = {}
    from collections import defaultdict
    per = defaultdict(list)
    for k, v in zip(key.tolist(), val.astype(np.float32).tolist()):
        per[k].append(v)
    for k, vs in per.items():
        res = []
        w = 0
        while w * slide + win <= len(vs):
            res.append(_seg_comb(vs[w * slide: w * slide + win], win, slide,
                                 comb))
            w += 1
        while w * slide < len(vs):  # EOS flush: partial tails
            res.append(_seg_comb(vs[w * slide:], win, slide, comb))
            w += 1
        out[k] = res
    return out


@pytest.mark.parametrize("use_tree", [False, True])
def test_gpu_ffat_cb_sum_vs_oracle(use_tree):
    n, n_keys, win, slide = 120_000, 101, 40, 10
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, 17_000, vdt=2))
           .withOutputSchema([2]).withOutputBatchSize(17_000).build())
    ff = (Ffat_Windows_GPU_Builder(
        native_gpu.gpu_ffat_windows(native_gpu.COMB_SUM, 0, win, slide,
                                    max_keys=1024, use_tree=use_tree))
          .withOutputSchema([2]).withOutputBatchSize(17_000).build())
    res = dict(rows=[])

    def pysink(cols):
        res['rows'].append((cols['key'].copy(), cols['c0'].copy(), cols['ts'].copy()))

    g = wf.PipeGraph("ffat")
    p = g.add_source(src)
    p.chain(ff)
    snk = wf.Sink_Builder(pysink).withParallelism(1).build()
    snk.out_schema = [2]
    p.add_sink(snk)
    g.run()
    oracle = _ffat_oracle(n, n_keys, win, slide, vdt=2, comb="sum")
    # EOS-flush partials share the key's last ts, so compare per-key value
    # multisets (sorted) + exact window counts
    from collections import defaultdict
    got = defaultdict(list)
    for k_arr, v_arr, t_arr in res['rows']:
        for k, v in zip(k_arr.tolist(), v_arr.tolist()):
            got[k].append(v)
    n_windows_oracle = sum(len(v) for v in oracle.values())
    n_windows_got = sum(len(v) for v in got.values())
    assert n_windows_got == n_windows_oracle
    for k, vals in got.items():
        ref = sorted(oracle[k])
        vals = sorted(vals)
        assert len(vals) == len(ref), f"key {k}"
        for a, b in zip(vals, ref):
            assert abs(a - b) <= 1e-3 * max(1.0, abs(b)), f"key {k}: {a} vs {b}"


def test_gpu_ffat_cb_min_tree_vs_oracle():
    n, n_keys, win, slide = 80_000, 53, 60, 12
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, 11_000, vdt=2))
           .withOutputSchema([2]).withOutputBatchSize(11_000).build())
    ff = (Ffat_Windows_GPU_Builder(
        native_gpu.gpu_ffat_windows(native_gpu.COMB_MIN, 0, win, slide,
                                    max_keys=512, use_tree=True))
          .withOutputSchema([2]).withOutputBatchSize(11_000).build())
    res = dict(rows=[])

    def pysink(cols):
        res['rows'].append((cols['key'].copy(), cols['c0'].copy(), cols['ts'].copy()))

    g = wf.PipeGraph("ffatmin")
    p = g.add_source(src)
    p.chain(ff)
    snk = wf.Sink_Builder(pysink).withParallelism(1).build()
    snk.out_schema = [2]
    p.add_sink(snk)
    g.run()
    oracle = _ffat_oracle(n, n_keys, win, slide, vdt=2, comb="min")
    from collections import defaultdict
    got = defaultdict(list)
    for k_arr, v_arr, t_arr in res['rows']:
        for k, v in zip(k_arr.tolist(), v_arr.tolist()):
            got[k].append(v)
    assert sum(len(v) for v in got.values()) == sum(len(v) for v in oracle.values())
    for k, vals in got.items():
        for a, b in zip(sorted(vals), sorted(oracle[k])):
            assert abs(a - b) <= 1e-5 * max(1.0, abs(b))


def test_gpu_merge_two_sources():
    """Two device sources merged into one GPU map (reference
    merge_tests_gpu): fan-in of device batches across streams."""
    n, b = 60_000, 15_000
    s1 = (Source_GPU_Builder(native_gpu.gpu_source(n, 31, b, vdt=0, seed=42))
          .withOutputSchema([0]).withOutputBatchSize(b).build())
    s2 = (Source_GPU_Builder(native_gpu.gpu_source(n, 31, b, vdt=0, seed=43))
          .withOutputSchema([0]).withOutputBatchSize(b).build())
    g = wf.PipeGraph("gmerge")
    p1 = g.add_source(s1)
    p2 = g.add_source(s2)
    mp = p1.merge(p2)
    mp.add(Map_GPU_Builder(native_gpu.gpu_affine_map(0, 2, 1, dtype=0))
           .withOutputSchema([0]).withOutputBatchSize(b).build())
    snk = wf.Sink_Builder(native.sum_sink(0)).withParallelism(1).build()
    snk.out_schema = [0]
    mp.add_sink(snk)
    g.run()
    exp = 0
    for seed in (42, 43):
        _, _, val = gen_batch(n, 0, seed, 31, 0)
        exp += int((val * 2 + 1).sum())
    assert g.sink_sum(snk) == exp
    assert g.sink_count(snk) == 2 * n


def test_gpu_split_round_robin():
    """split_gpu: device batches distributed across two GPU branches
    (reference split_tests_gpu; rr distribution instead of replication)."""
    from windflow_amd.operators import Operator
    from windflow_amd.builders_gpu import Sink_GPU_Builder
    n, b = 80_000, 10_000
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, 31, b, vdt=0))
           .withOutputSchema([0]).withOutputBatchSize(b).build())
    g = wf.PipeGraph("gsplit")
    mp = g.add_source(src)
    mp.split_gpu(2)
    snks = []
    for br in range(2):
        bmp = mp.select(br)
        bmp.add(Map_GPU_Builder(native_gpu.gpu_affine_map(0, 1, 100 * (br + 1),
                                                          dtype=0))
                .withOutputSchema([0]).withOutputBatchSize(b).build())
        snk = Sink_GPU_Builder(native_gpu.gpu_count_sink()).build()
        bmp.chain_sink(snk)
        snks.append(snk)
    g.run()
    assert g.sink_count(snks[0]) + g.sink_count(snks[1]) == n
    assert g.sink_count(snks[0]) == n // 2  # rr over 8 batches


def test_gpu_jit_map_filter():
    """hiprtc-compiled user device logic (reference __device__ lambda
    parity): custom expression map + predicate filter."""
    n, n_keys, b = 100_000, 53, 25_000
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=0))
           .withOutputSchema([0]).withOutputBatchSize(b).build())
    jm = (Map_GPU_Builder(native_gpu.gpu_jit_map("v * 3 + (i64)(key & 1)", 0))
          .withOutputSchema([0]).withOutputBatchSize(b).build())
    jf = (Filter_GPU_Builder(native_gpu.gpu_jit_filter("v % 7 != 0 && ts % 2 == 0", 0))
          .withOutputSchema([0]).withOutputBatchSize(b).build())
    g, snk = gpu_graph(src, jm, jf)
    g.run()
    ts, key, val = gen_batch(n, 0, 42, n_keys, 0)
    v = val * 3 + (key & 1).astype(np.int64)
    keep = (v % 7 != 0) & (ts % 2 == 0)
    assert g.sink_sum(snk) == int(v[keep].sum())
    assert g.sink_count(snk) == int(keep.sum())


def test_gpu_ffat_high_key_count():
    """1M distinct keys (config #5 scale direction): the batched multi-key
    fold must stay correct when segments are tiny (avg ~4 tuples/key)."""
    n, n_keys, win, slide = 4_000_000, 1_000_000, 8, 2
    b = 1_000_000
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=2))
           .withOutputSchema([2]).withOutputBatchSize(b).build())
    ff = (Ffat_Windows_GPU_Builder(
        native_gpu.gpu_ffat_windows(native_gpu.COMB_SUM, 0, win, slide,
                                    max_keys=2_000_000))
          .withOutputSchema([2]).withOutputBatchSize(2 * b).build())
    g = wf.PipeGraph("bigkeys")
    p = g.add_source(src)
    p.chain(ff)
    from windflow_amd.builders_gpu import Sink_GPU_Builder
    snk = Sink_GPU_Builder(native_gpu.gpu_count_sink()).build()
    p.chain_sink(snk)
    g.run()
    # oracle on fired-window COUNT only (sum oracle would be slow in python);
    # every window start w*slide < c fires (EOS flush covers the partials)
    ts, key, val = gen_batch(n, 0, 42, n_keys, 2)
    from collections import Counter
    per = Counter(key.tolist())
    exp = sum((c - 1) // slide + 1 for c in per.values())
    assert g.sink_count(snk) == exp


def test_gpu_stateful_map_running_sum():
    """Keyed device state advanced in key order (reference
    Stateful_MAPGPU_Kernel): per-key running sum across batches."""
    n, n_keys, b = 100_000, 64, 25_000
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=0))
           .withOutputSchema([0]).withOutputBatchSize(b).build())
    sm = (Map_GPU_Builder(native_gpu.gpu_keyed_running_sum(0, max_keys=256))
          .withOutputSchema([0]).withOutputBatchSize(b).build())
    g, snk = gpu_graph(src, sm)
    g.run()
    _, key, val = gen_batch(n, 0, 42, n_keys, 0)
    acc = {}
    exp = 0
    for k, v in zip(key.tolist(), val.tolist()):
        acc[k] = acc.get(k, 0) + v
        exp += acc[k]
    assert g.sink_sum(snk) == exp
    assert g.sink_count(snk) == n


def test_gpu_stateful_filter_dedup():
    """Stateful filter: drop consecutive per-key duplicate values."""
    n, n_keys, b = 100_000, 8, 25_000  # few keys -> many consecutive dups
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=0))
           .withOutputSchema([0]).withOutputBatchSize(b).build())
    fl = (Filter_GPU_Builder(native_gpu.gpu_keyed_dedup(0, max_keys=64))
          .withOutputSchema([0]).withOutputBatchSize(b).build())
    g, snk = gpu_graph(src, fl)
    g.run()
    _, key, val = gen_batch(n, 0, 42, n_keys, 0)
    last = {}
    s = cnt = 0
    for k, v in zip(key.tolist(), val.tolist()):
        if last.get(k) != v:
            s += v
            cnt += 1
        last[k] = v
    assert g.sink_sum(snk) == s
    assert g.sink_count(snk) == cnt


def test_gpu_ffat_tb_vs_oracle():
    """Event-time (TB) GPU windows vs brute-force oracle: windows
    [w*slide, w*slide+win) on ts, aligned at the first window containing
    each key's first tuple, fired once their last pane holds data <= wm."""
    n, n_keys, win, slide = 120_000, 101, 400, 100  # pane = 100
    b = 17_000
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=2))
           .withOutputSchema([2]).withOutputBatchSize(b).build())
    ff = (Ffat_Windows_GPU_Builder(
        native_gpu.gpu_ffat_windows(native_gpu.COMB_SUM, 0, win, slide,
                                    max_keys=1024, tb=True, pend_ring_log2=10))
          .withOutputSchema([2]).withOutputBatchSize(2 * b).build())
    res = dict(rows=[])

    def pysink(cols):
        res['rows'].append((cols['key'].copy(), cols['c0'].copy(),
                            cols['ts'].copy()))

    g = wf.PipeGraph("ffattb")
    p = g.add_source(src)
    p.chain(ff)
    snk = wf.Sink_Builder(pysink).withParallelism(1).build()
    snk.out_schema = [2]
    p.add_sink(snk)
    g.run()
    # oracle
    ts, key, val = gen_batch(n, 0, 42, n_keys, 2)
    from collections import defaultdict
    per_t = defaultdict(list)
    for t, k, v in zip(ts.tolist(), key.tolist(), val.tolist()):
        per_t[k].append((t, np.float32(v)))
    pane = int(np.gcd(win, slide))
    exp = defaultdict(list)
    for k, rows in per_t.items():
        tss = [t for t, _ in rows]
        t0, tmax = min(tss), max(tss)
        w0 = max(0, -(-(t0 - win + 1) // slide))
        w = w0
        while (w * slide + win - 1) // pane <= tmax // pane:
            s = float(np.sum(np.array(
                [v for t, v in rows if w * slide <= t < w * slide + win],
                dtype=np.float32), dtype=np.float64))
            exp[k].append((w * slide + win - 1, s))
            w += 1
        # EOS partial flush (windows with data panes but incomplete span)
        # fire with ts = the key's last tuple ts
        while (w * slide) // pane <= tmax // pane:
            s = float(np.sum(np.array(
                [v for t, v in rows if w * slide <= t < w * slide + win],
                dtype=np.float32), dtype=np.float64))
            exp[k].append((tmax, s))
            w += 1
    got = defaultdict(list)
    for k_arr, v_arr, t_arr in res['rows']:
        for k, v, t in zip(k_arr.tolist(), v_arr.tolist(), t_arr.tolist()):
            got[k].append((t, v))
    assert sum(len(v) for v in got.values()) == sum(len(v) for v in exp.values())
    for k, pairs in got.items():
        pairs.sort()
        ref = sorted(exp[k])
        for (t_g, v_g), (t_r, v_r) in zip(pairs, ref):
            assert t_g == t_r, f"key {k}: ts {t_g} vs {t_r}"
            assert abs(v_g - v_r) <= 1e-3 * max(1.0, abs(v_r)), (k, v_g, v_r)


def test_gpu_keyby_exchange_world1(monkeypatch):
    """RCCL self-exchange (world=1): every row routes back to rank 0, so the
    pipeline is value-preserving; WFA_XCHG_NO_SELFPASS forces the full
    bucket->sort->gather->allgather->transfer machinery (the production
    world=1 path self-forwards without copying)."""
    import os
    monkeypatch.setenv("WFA_XCHG_NO_SELFPASS", "1")
    from windflow_amd import _core
    from windflow_amd.builders_gpu import KeyBy_Exchange_GPU_Builder
    n, n_keys, b = 100_000, 97, 25_000
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=0))
           .withOutputSchema([0]).withOutputBatchSize(b).build())
    ex = (KeyBy_Exchange_GPU_Builder(native_gpu.gpu_keyby_exchange())
          .withOutputSchema([0]).withOutputBatchSize(2 * b).build())
    g = wf.PipeGraph("a2a")
    g.set_dist(0, 1, _core.rccl_unique_id())
    p = g.add_source(src)
    p.chain(ex)
    snk = wf.Sink_Builder(native.sum_sink(0)).withParallelism(1).build()
    snk.out_schema = [0]
    p.add_sink(snk)
    g.run()
    _, _, val = gen_batch(n, 0, 42, n_keys, 0)
    assert g.sink_sum(snk) == int(val.sum())
    assert g.sink_count(snk) == n


def test_gpu_exchange_then_reduce():
    """config #4 shape on one rank: map_gpu -> keyby exchange -> reduce_gpu."""
    from windflow_amd import _core
    from windflow_amd.builders_gpu import KeyBy_Exchange_GPU_Builder
    n, n_keys, b = 100_000, 64, 25_000
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=2))
           .withOutputSchema([2]).withOutputBatchSize(b).build())
    mp_ = (Map_GPU_Builder(native_gpu.gpu_affine_map(0, 2.0, 0.0, dtype=2))
           .withOutputSchema([2]).withOutputBatchSize(b).build())
    ex = (KeyBy_Exchange_GPU_Builder(native_gpu.gpu_keyby_exchange())
          .withOutputSchema([2]).withOutputBatchSize(2 * b).build())
    rd = (Reduce_GPU_Builder(native_gpu.gpu_keyed_reduce(native_gpu.COMB_SUM, 0, 256))
          .withOutputSchema([2]).withOutputBatchSize(2 * b).build())
    res = dict(s=0.0)

    def pysink(cols):
        res['s'] += float(cols['c0'].astype(np.float64).sum())

    g = wf.PipeGraph("a2a_red")
    g.set_dist(0, 1, _core.rccl_unique_id())
    p = g.add_source(src)
    p.chain(mp_)
    p.chain(ex)
    p.chain(rd)
    snk = wf.Sink_Builder(pysink).withParallelism(1).build()
    snk.out_schema = [2]
    p.add_sink(snk)
    g.run()
    _, _, val = gen_batch(n, 0, 42, n_keys, 2)
    ref = float((val.astype(np.float64) * 2.0).sum())
    assert abs(res['s'] - ref) <= 2e-3 * max(1.0, abs(ref))


def test_gpu_reduce_keyed_sum():
    n, n_keys, b = 100_000, 64, 25_000
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=2))
           .withOutputSchema([2]).withOutputBatchSize(b).build())
    rd = (Reduce_GPU_Builder(native_gpu.gpu_keyed_reduce(native_gpu.COMB_SUM, 0, 256))
          .withOutputSchema([2]).withOutputBatchSize(b).build())
    res = dict(s=0.0, n=0, rows=[])

    def pysink(cols):
        res['s'] += float(cols['c0'].astype(np.float64).sum())
        res['n'] += len(cols['c0'])
        res['rows'].append((cols['key'].copy(), cols['ts'].copy()))

    g = wf.PipeGraph("red")
    p = g.add_source(src)
    p.chain(rd)
    snk = wf.Sink_Builder(pysink).withParallelism(1).build()
    snk.out_schema = [2]
    p.add_sink(snk)
    g.run()
    # per-batch keyed sums sum to the global sum; count = sum of per-batch
    # distinct key counts
    ts, key, val = gen_batch(n, 0, 42, n_keys, 2)
    exp_n = 0
    for s in range(0, n, b):
        exp_n += len(set(key[s:s + b].tolist()))
    assert res['n'] == exp_n
    ref = float(val.astype(np.float64).sum())
    assert abs(res['s'] - ref) <= 2e-3 * max(1.0, abs(ref))
    # output ts = max ts of that key within its batch (reference Reduce_GPU
    # ts = max; exercises the monotonic-ts fast path: out_ts from the
    # segment's LAST row must equal the true max)
    assert len(res['rows']) == n // b
    for bi, (ks_, ts_) in enumerate(res['rows']):
        kb = key[bi * b:(bi + 1) * b]
        tb = ts[bi * b:(bi + 1) * b]
        for k, t in zip(ks_.tolist(), ts_.tolist()):
            assert t == int(tb[kb == k].max()), (bi, k)


def test_gpu_ffat_count_comb():
    """COUNT combiner through the FFAT folds (ring and tree): every full CB
    window counts exactly `win` tuples; EOS-flushed partial windows count
    their tail (idx - w*slide); window multiset per key is exact."""
    n, n_keys, b, win, slide = 400_000, 101, 100_000, 300, 100
    for tree in (False, True):
        src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=2))
               .withOutputSchema([2]).withOutputBatchSize(b).build())
        ff = (Ffat_Windows_GPU_Builder(
            native_gpu.gpu_ffat_windows(native_gpu.COMB_COUNT, 0, win, slide,
                                        max_keys=256, use_tree=tree))
            .withOutputSchema([2]).withOutputBatchSize(b).build())
        got = dict(per={})

        def pysink(cols):
            for k, v in zip(cols['key'].tolist(), cols['c0'].tolist()):
                got['per'].setdefault(k, []).append(int(round(v)))

        g = wf.PipeGraph("cnt")
        p = g.add_source(src)
        p.chain(ff)
        snk = wf.Sink_Builder(pysink).withParallelism(1).build()
        snk.out_schema = [2]
        p.add_sink(snk)
        g.run()
        ts, key, val = gen_batch(n, 0, 42, n_keys, 2)
        import collections
        per = collections.Counter(key.tolist())
        exp = {}
        for k, c in per.items():
            counts = [win] * (max(0, (c - win) // slide + 1))
            w = len(counts)
            while w * slide < c:
                counts.append(c - w * slide)
                w += 1
            exp[k] = sorted(counts)
        assert {k: sorted(v) for k, v in got['per'].items()} == exp


def test_gpu_jit_expr_fuzz():
    """Random hiprtc-compiled map expressions vs numpy float32 evaluation."""
    import random
    EXPRS = [
        ("v * 2.5f + 1.0f",            lambda v, t, k: v * np.float32(2.5) + 1),
        ("fmaxf(v, 100.0f)",           lambda v, t, k: np.maximum(v, 100)),
        ("fminf(v * v, 9000.0f)",      lambda v, t, k: np.minimum(v * v, 9000)),
        ("fabsf(v - 500.0f)",          lambda v, t, k: np.abs(v - 500)),
        ("v + (float)(key & 7)",       lambda v, t, k: v + (k & 7).astype(np.float32)),
        ("v * 0.5f + (float)(ts % 11)", lambda v, t, k: v * np.float32(0.5)
                                        + (t % 11).astype(np.float32)),
    ]
    n, n_keys, b = 200_000, 64, 50_000
    rng = random.Random(99)
    for expr, ref in rng.sample(EXPRS, 4):
        src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=2))
               .withOutputSchema([2]).withOutputBatchSize(b).build())
        jm = (Map_GPU_Builder(native_gpu.gpu_jit_map(expr, 0))
              .withOutputSchema([2]).withOutputBatchSize(b).build())
        acc = dict(s=0.0, n=0)

        def pysink(cols):
            acc['s'] += float(cols['c0'].astype(np.float64).sum())
            acc['n'] += len(cols['c0'])

        g = wf.PipeGraph("jitf")
        p = g.add_source(src)
        p.chain(jm)
        snk = wf.Sink_Builder(pysink).withParallelism(1).build()
        snk.out_schema = [2]
        p.add_sink(snk)
        g.run()
        ts, key, val = gen_batch(n, 0, 42, n_keys, 2)
        exp = float(ref(val.astype(np.float32), ts, key)
                    .astype(np.float64).sum())
        assert acc['n'] == n, expr
        assert abs(acc['s'] - exp) <= 2e-3 * max(1.0, abs(exp)), \
            (expr, acc['s'], exp)


def test_gpu_reduce_all():
    """Unkeyed full-batch reduce vs numpy: one tuple per batch (sum and max)."""
    n, b = 1_000_000, 250_000
    for comb, oracle in ((native_gpu.COMB_SUM, lambda v: v.astype(np.float64).sum()),
                         (native_gpu.COMB_MAX, lambda v: float(v.max()))):
        src = (Source_GPU_Builder(native_gpu.gpu_source(n, 64, b, vdt=2))
               .withOutputSchema([2]).withOutputBatchSize(b).build())
        rd = (Reduce_GPU_Builder(native_gpu.gpu_reduce_all(comb, 0))
              .withOutputSchema([2]).withOutputBatchSize(1024).build())
        got = []

        def pysink(cols):
            got.extend(float(x) for x in cols['c0'])

        g = wf.PipeGraph("redall")
        p = g.add_source(src)
        p.chain(rd)
        snk = wf.Sink_Builder(pysink).withParallelism(1).build()
        snk.out_schema = [2]
        p.add_sink(snk)
        g.run()
        assert len(got) == n // b  # exactly one result per batch, in order
        ts, key, val = gen_batch(n, 0, 42, 64, 2)
        for i in range(n // b):
            ref = oracle(val[i * b:(i + 1) * b])
            tol = 2e-3 * max(1.0, abs(ref)) if comb == native_gpu.COMB_SUM else 1e-6
            assert abs(got[i] - ref) <= tol


def test_gpu_mfma_gram_windows():
    """MFMA Gram windows vs numpy einsum: per-key tumbling windows of 32
    16-dim vectors, window aggregate sum(v v^T) on the matrix cores."""
    from windflow_amd.builders_gpu import _GpuBuilder

    class Gram_Windows_GPU_Builder(_GpuBuilder):
        _kind = "gpu_gram"

    n, n_keys, win, b = 20_000, 13, 32, 5_000
    rng = np.random.default_rng(3)
    data = rng.standard_normal((n, 16)).astype(np.float32)
    keys = rng.integers(0, n_keys, size=n).astype(np.uint64)
    state = dict(pos=0)

    def src(replica, par):
        p = state['pos']
        if p >= n:
            return None
        m = min(b, n - p)
        state['pos'] += m
        d = dict(ts=np.arange(p, p + m, dtype=np.int64), key=keys[p:p + m],
                 watermark=p + m)
        for c in range(16):
            d[f'c{c}'] = np.ascontiguousarray(data[p:p + m, c])
        return d

    rows = []

    def sink(cols):
        for i in range(len(cols['ts'])):
            rows.append((int(cols['key'][i]), int(cols['c0'][i]),
                         [float(cols[f'c{c}'][i]) for c in range(1, 17)]))

    g = wf.PipeGraph("gram")
    mp = g.add_source(wf.Source_Builder(src).withParallelism(1)
                      .withOutputSchema([2] * 16).withOutputBatchSize(b).build())
    gr = (Gram_Windows_GPU_Builder(native_gpu.gpu_gram_windows(win, max_keys=64))
          .withOutputSchema([0] + [2] * 16).withOutputBatchSize(4 * b).build())
    mp.chain(gr)
    snk = wf.Sink_Builder(sink).withParallelism(1).build()
    snk.out_schema = [0] + [2] * 16
    mp.add_sink(snk)
    g.run()

    # oracle: per key, windows of `win` vectors in arrival order
    from collections import defaultdict
    per = defaultdict(list)
    for i in range(n):
        per[int(keys[i])].append(data[i])
    exp = {}
    n_windows = 0
    for k, vs in per.items():
        for w in range(len(vs) // win):
            seg = np.stack(vs[w * win:(w + 1) * win])
            exp[(k, w)] = seg.T @ seg
            n_windows += 1
    assert len(rows) == 16 * n_windows
    got = defaultdict(dict)
    for k, gwid, vals in rows:
        ref = exp[(k, gwid)]
        # identify the row by best match (rows arrive in C/D lane order)
        # simpler: accumulate rows into a matrix keyed by arrival order
        got[(k, gwid)].setdefault('rows', []).append(vals)
    for (k, w), d in got.items():
        m = np.array(sorted(d['rows']))
        ref = np.array(sorted(exp[(k, w)].tolist()))
        assert np.allclose(m, ref, rtol=1e-4, atol=1e-4), (k, w)


def test_gpu_jit_avg_reduce_vs_oracle():
    """Fused per-batch keyed AVG via the generalized JIT fold (user
    lift/comb/finalize — reference arbitrary-combine Reduce_GPU)."""
    n, n_keys, b = 100_000, 64, 25_000
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=2))
           .withOutputSchema([2]).withOutputBatchSize(b).build())
    rd = (Reduce_GPU_Builder(native_gpu.gpu_avg_reduce(0, max_keys=256))
          .withOutputSchema([2]).withOutputBatchSize(b).build())
    rows = []

    def pysink(cols):
        for k, v in zip(cols['key'].tolist(), cols['c0'].tolist()):
            rows.append((k, v))

    g = wf.PipeGraph("javg")
    p = g.add_source(src)
    p.chain(rd)
    snk = wf.Sink_Builder(pysink).withParallelism(1).build()
    snk.out_schema = [2]
    p.add_sink(snk)
    g.run()
    ts, key, val = gen_batch(n, 0, 42, n_keys, 2)
    # one avg per (batch, key); compare per-key sorted multisets
    from collections import defaultdict
    exp = defaultdict(list)
    for s in range(0, n, b):
        kb, vb = key[s:s + b], val[s:s + b].astype(np.float64)
        for k in set(kb.tolist()):
            exp[k].append(float(vb[kb == k].mean()))
    got = defaultdict(list)
    for k, v in rows:
        got[k].append(v)
    assert sum(map(len, got.values())) == sum(map(len, exp.values()))
    for k in exp:
        for a, bb in zip(sorted(got[k]), sorted(exp[k])):
            assert abs(a - bb) <= 2e-3 * max(1.0, abs(bb)), (k, a, bb)


@pytest.mark.parametrize("win,slide", [(40, 10), (400, 100)])
def test_gpu_jit_ffat_avg_vs_oracle(win, slide):
    """Sliding-window AVG via the JIT fold — (sum,count) pane pairs with the
    invertible running-total machine; pane 10 exercises the thread kernel,
    pane 100 the wave kernel.  Includes EOS partial flush."""
    n, n_keys, b = 120_000, 101, 17_000
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=2))
           .withOutputSchema([2]).withOutputBatchSize(b).build())
    ff = (Ffat_Windows_GPU_Builder(
        native_gpu.gpu_avg_ffat_windows(win, slide, col=0, max_keys=1024))
          .withOutputSchema([2]).withOutputBatchSize(2 * b).build())
    res = dict(rows=[])

    def pysink(cols):
        res['rows'].append((cols['key'].copy(), cols['c0'].copy()))

    g = wf.PipeGraph("jffavg")
    p = g.add_source(src)
    p.chain(ff)
    snk = wf.Sink_Builder(pysink).withParallelism(1).build()
    snk.out_schema = [2]
    p.add_sink(snk)
    g.run()
    ts, key, val = gen_batch(n, 0, 42, n_keys, 2)
    from collections import defaultdict
    per = defaultdict(list)
    for k, v in zip(key.tolist(), val.astype(np.float64).tolist()):
        per[k].append(v)
    exp = defaultdict(list)
    for k, vs in per.items():
        w = 0
        while w * slide < len(vs):
            seg = vs[w * slide: w * slide + win]
            exp[k].append(float(np.mean(seg)))
            w += 1
    got = defaultdict(list)
    for k_arr, v_arr in res['rows']:
        for k, v in zip(k_arr.tolist(), v_arr.tolist()):
            got[k].append(v)
    assert sum(map(len, got.values())) == sum(map(len, exp.values()))
    for k in exp:
        for a, bb in zip(sorted(got[k]), sorted(exp[k])):
            assert abs(a - bb) <= 2e-3 * max(1.0, abs(bb)), (k, a, bb)


def test_gpu_jit_ffat_minmax_multicol():
    """Multi-column JIT window fold: 2 value columns in, 2 result columns
    out (min of c0, max of c1) through the non-invertible P-pane recombine
    path (reference arbitrary result structs)."""
    n, n_keys, win, slide, b = 80_000, 53, 60, 12, 11_000
    rng = np.random.default_rng(7)
    vals0 = rng.standard_normal(n).astype(np.float32)
    vals1 = rng.standard_normal(n).astype(np.float32)
    keys = rng.integers(0, n_keys, size=n).astype(np.uint64)
    state = dict(pos=0)

    def src(replica, par):
        p = state['pos']
        if p >= n:
            return None
        m = min(b, n - p)
        state['pos'] += m
        return dict(ts=np.arange(p, p + m, dtype=np.int64), key=keys[p:p + m],
                    c0=vals0[p:p + m], c1=vals1[p:p + m], watermark=p + m)

    ff = (Ffat_Windows_GPU_Builder(
        lift="v0;v1",
        comb="fminf(a0, b0);fmaxf(a1, b1)",
        finalize="f0;f1",
        identity=(float("inf"), float("-inf")), cols=(0, 1), max_keys=256)
        .withCBWindows(win, slide)
        .withOutputSchema([2, 2]).withOutputBatchSize(4 * b).build())
    rows = []

    def sink(cols):
        for i in range(len(cols['ts'])):
            rows.append((int(cols['key'][i]), float(cols['c0'][i]),
                         float(cols['c1'][i])))

    g = wf.PipeGraph("jmm")
    mp = g.add_source(wf.Source_Builder(src).withParallelism(1)
                      .withOutputSchema([2, 2]).withOutputBatchSize(b).build())
    mp.chain(ff)
    snk = wf.Sink_Builder(sink).withParallelism(1).build()
    snk.out_schema = [2, 2]
    mp.add_sink(snk)
    g.run()
    from collections import defaultdict
    per = defaultdict(list)
    for i in range(n):
        per[int(keys[i])].append((float(vals0[i]), float(vals1[i])))
    exp = defaultdict(list)
    for k, vs in per.items():
        w = 0
        while w * slide < len(vs):
            seg = vs[w * slide: w * slide + win]
            exp[k].append((min(a for a, _ in seg), max(c for _, c in seg)))
            w += 1
    got = defaultdict(list)
    for k, v0, v1 in rows:
        got[k].append((v0, v1))
    assert sum(map(len, got.values())) == sum(map(len, exp.values()))
    for k in exp:
        for (a0, a1), (b0, b1) in zip(sorted(got[k]), sorted(exp[k])):
            assert abs(a0 - b0) <= 1e-5 and abs(a1 - b1) <= 1e-5, (k,)


@pytest.mark.parametrize("acc", ["f32", "f64"])
def test_gpu_jit_ffat_tb_avg_vs_oracle(acc):
    """Event-time JIT windows: AVG over TB panes + watermark advance +
    EOS flush, vs brute-force oracle — in both accumulator precisions
    (f64 exercises the double pend arena through the TB wave lift)."""
    n, n_keys, win, slide, b = 120_000, 101, 400, 100, 17_000
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=2))
           .withOutputSchema([2]).withOutputBatchSize(b).build())
    ff = (Ffat_Windows_GPU_Builder(
        native_gpu.gpu_avg_ffat_windows(win, slide, col=0, max_keys=1024,
                                        tb=True, pend_ring_log2=10, acc=acc))
          .withOutputSchema([2]).withOutputBatchSize(2 * b).build())
    res = dict(rows=[])

    def pysink(cols):
        res['rows'].append((cols['key'].copy(), cols['c0'].copy()))

    g = wf.PipeGraph("jtbavg")
    p = g.add_source(src)
    p.chain(ff)
    snk = wf.Sink_Builder(pysink).withParallelism(1).build()
    snk.out_schema = [2]
    p.add_sink(snk)
    g.run()
    ts, key, val = gen_batch(n, 0, 42, n_keys, 2)
    pane = int(np.gcd(win, slide))
    from collections import defaultdict
    per = defaultdict(list)
    for t, k, v in zip(ts.tolist(), key.tolist(), val.tolist()):
        per[k].append((t, float(v)))
    exp = defaultdict(list)
    for k, rowsk in per.items():
        tss = [t for t, _ in rowsk]
        t0, tmax = min(tss), max(tss)
        w = max(0, -(-(t0 - win + 1) // slide))
        while (w * slide) // pane <= tmax // pane:
            seg = [v for t, v in rowsk if w * slide <= t < w * slide + win]
            # empty (gap) windows fire the default result 0 (reference
            # semantics; the avg finalize gates the division on the count)
            exp[k].append(float(np.mean(seg)) if seg else 0.0)
            w += 1
    got = defaultdict(list)
    for k_arr, v_arr in res['rows']:
        for k, v in zip(k_arr.tolist(), v_arr.tolist()):
            got[k].append(v)
    assert sum(map(len, got.values())) == sum(map(len, exp.values()))
    for k in exp:
        ga, ex = sorted(got[k]), sorted(exp[k])
        assert len(ga) == len(ex), f"key {k}"
        for a, bb in zip(ga, ex):
            assert abs(a - bb) <= 2e-3 * max(1.0, abs(bb)), (k, a, bb)


def test_gpu_jit_stateful_map_filter():
    """Arbitrary stateful device bodies (reference stateful MAP/FILTER_GPU
    functors): EMA map + every-3rd filter, JIT-compiled."""
    n, n_keys, b = 100_000, 64, 25_000
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=2))
           .withOutputSchema([2]).withOutputBatchSize(b).build())
    sm = (Map_GPU_Builder(native_gpu.gpu_jit_stateful_map(
        "s0 = 0.5 * s0 + 0.5 * (double)v0; v0 = (float)s0", max_keys=256))
          .withOutputSchema([2]).withOutputBatchSize(b).build())
    fl = (Filter_GPU_Builder(native_gpu.gpu_jit_stateful_filter(
        "s0 = s0 + 1.0; keep = ((i64)s0 % 3) == 0", max_keys=256))
          .withOutputSchema([2]).withOutputBatchSize(b).build())
    acc = dict(s=0.0, n=0)

    def pysink(cols):
        acc['s'] += float(cols['c0'].astype(np.float64).sum())
        acc['n'] += len(cols['c0'])

    g = wf.PipeGraph("jst")
    p = g.add_source(src)
    p.chain(sm)
    p.chain(fl)
    snk = wf.Sink_Builder(pysink).withParallelism(1).build()
    snk.out_schema = [2]
    p.add_sink(snk)
    g.run()
    ts, key, val = gen_batch(n, 0, 42, n_keys, 2)
    ema = {}
    cnt = {}
    s = 0.0
    m = 0
    for k, v in zip(key.tolist(), val.astype(np.float32).tolist()):
        e = 0.5 * ema.get(k, 0.0) + 0.5 * float(np.float32(v))
        ema[k] = e
        c = cnt.get(k, 0) + 1
        cnt[k] = c
        if c % 3 == 0:
            s += float(np.float32(e))
            m += 1
    assert acc['n'] == m
    assert abs(acc['s'] - s) <= 2e-3 * max(1.0, abs(s))


def test_gpu_ffat_vik_fallback_key_overflow():
    """Automatic VIK fallback: bf16 values but max_keys > 65535 forces the
    indirect-gather path; results must match the VIK-eligible config."""
    n, n_keys, win, slide, b = 200_000, 101, 500, 100, 50_000
    sums = []
    for mk in (1024, 70_000):  # VIK on / off
        src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=5))
               .withOutputSchema([5]).withOutputBatchSize(b).build())
        ff = (Ffat_Windows_GPU_Builder(
            native_gpu.gpu_ffat_windows(native_gpu.COMB_SUM, 0, win, slide,
                                        max_keys=mk))
              .withOutputSchema([2]).withOutputBatchSize(2 * b).build())
        acc = dict(s=0.0, n=0)

        def pysink(cols):
            acc['s'] += float(cols['c0'].astype(np.float64).sum())
            acc['n'] += len(cols['c0'])

        g = wf.PipeGraph(f"vik{mk}")
        p = g.add_source(src)
        p.chain(ff)
        snk = wf.Sink_Builder(pysink).withParallelism(1).build()
        snk.out_schema = [2]
        p.add_sink(snk)
        g.run()
        sums.append((acc['n'], acc['s']))
    assert sums[0][0] == sums[1][0]
    assert abs(sums[0][1] - sums[1][1]) <= 1e-6 * max(1.0, abs(sums[0][1]))


def test_gpu_split_per_tuple():
    """Per-tuple device split (reference splitting_emitter_gpu per-branch
    batches, upgraded from replication to true routing): a JIT branch
    expression routes each row; per-branch compaction on device."""
    n, n_keys, b = 120_000, 97, 20_000
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=0))
           .withOutputSchema([0]).withOutputBatchSize(b).build())
    g = wf.PipeGraph("gsplit2")
    mp = g.add_source(src)
    mp.split_gpu(3, expr="key % 3")
    snks = []
    for br in range(3):
        bmp = mp.select(br)
        bmp.add(Map_GPU_Builder(native_gpu.gpu_affine_map(0, 1, 0, dtype=0))
                .withOutputSchema([0]).withOutputBatchSize(b).build())
        snk = Sink_GPU_Builder(native_gpu.gpu_count_sink()).build()
        bmp.chain_sink(snk)
        snks.append(snk)
    g.run()
    ts, key, val = gen_batch(n, 0, 42, n_keys, 0)
    for br in range(3):
        assert g.sink_count(snks[br]) == int((key % 3 == br).sum()), br


def test_gpu_split_per_tuple_value_expr():
    """Branch expression over the VALUE column with drop semantics: rows
    whose expression falls outside [0, n_branches) disappear."""
    n, b = 80_000, 20_000
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, 31, b, vdt=2))
           .withOutputSchema([2]).withOutputBatchSize(b).build())
    g = wf.PipeGraph("gsplitv")
    mp = g.add_source(src)
    # v0 in [0,1): branch 0 for v<0.25, 1 for v<0.5, else 7 (dropped)
    mp.split_gpu(2, expr="v0 < 0.25f ? 0 : (v0 < 0.5f ? 1 : 7)")
    snks = []
    for br in range(2):
        bmp = mp.select(br)
        snk = Sink_GPU_Builder(native_gpu.gpu_count_sink()).build()
        bmp.chain_sink(snk)
        snks.append(snk)
    g.run()
    ts, key, val = gen_batch(n, 0, 42, 31, 2)
    assert g.sink_count(snks[0]) == int((val < 0.25).sum())
    assert g.sink_count(snks[1]) == int(((val >= 0.25) & (val < 0.5)).sum())


def test_gpu_device_broadcast():
    """Device-batch broadcast (round 1 raised an error): every replica of a
    broadcast-input GPU sink sees ALL batches via D2D clones."""
    n, b = 60_000, 15_000
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, 31, b, vdt=0))
           .withOutputSchema([0]).withOutputBatchSize(b).build())
    g = wf.PipeGraph("gbcast")
    mp = g.add_source(src)
    snk = (Sink_GPU_Builder(native_gpu.gpu_count_sink())
           .withParallelism(3).withBroadcast().build())
    mp.add_sink(snk)
    g.run()
    assert g.sink_count(snk) == 3 * n  # each of the 3 replicas saw all rows


def test_gpu_peer_copy_path_forced():
    """Cross-device forward path (hipMemcpyPeerAsync): a 1-GPU box can't
    place ops on two devices, so WFA_FORCE_PEER_COPY=1 routes every
    device->device hand-off through the peer-copy machinery (device 0->0)
    — same code path, same events/sync protocol.  Results must be
    identical to the pointer-passing run."""
    import os
    import subprocess
    import sys
    script = r"""
import sys
sys.path.insert(0, ".")
import windflow_amd as wf
from windflow_amd import native, native_gpu
from windflow_amd.builders_gpu import (Source_GPU_Builder, Map_GPU_Builder,
                                       Filter_GPU_Builder)
from windflow_amd.synth import gen_batch
n, n_keys, b = 100_000, 97, 25_000
src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=0))
       .withOutputSchema([0]).withOutputBatchSize(b).build())
mp_ = (Map_GPU_Builder(native_gpu.gpu_affine_map(0, 3, 1, dtype=0))
       .withOutputSchema([0]).withOutputBatchSize(b)
       .withParallelism(1).build())
fl = (Filter_GPU_Builder(native_gpu.gpu_mod_filter(0, 5, 0))
      .withOutputSchema([0]).withOutputBatchSize(b)
      .withParallelism(1).build())
g = wf.PipeGraph("peer")
p = g.add_source(src)
p.add(mp_)   # add (not chain): a real queue hop -> peer copy on receive
p.add(fl)
snk = wf.Sink_Builder(native.sum_sink(0)).withParallelism(1).build()
snk.out_schema = [0]
p.add_sink(snk)
g.run()
_, _, val = gen_batch(n, 0, 42, n_keys, 0)
v = val * 3 + 1
keep = v[v % 5 != 0]
assert g.sink_sum(snk) == int(keep.sum()), (g.sink_sum(snk), int(keep.sum()))
assert g.sink_count(snk) == len(keep)
print("PEER_OK")
"""
    env = dict(os.environ, WFA_FORCE_PEER_COPY="1")
    r = subprocess.run([sys.executable, "-c", script], env=env,
                       capture_output=True, text=True, timeout=300,
                       cwd=os.path.dirname(os.path.dirname(
                           os.path.abspath(__file__))))
    assert r.returncode == 0 and "PEER_OK" in r.stdout, r.stderr[-2000:]


def test_gpu_split_merge_diamond():
    """Device diamond (reference split_tests_gpu + merge_tests_gpu shapes):
    source -> per-tuple split(2) -> branch maps (x+100 / x+200) -> merge ->
    GPU sink; per-branch transforms verified through the merged sum."""
    n, b = 100_000, 20_000
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, 31, b, vdt=0))
           .withOutputSchema([0]).withOutputBatchSize(b).build())
    g = wf.PipeGraph("gdia")
    mp = g.add_source(src)
    mp.split_gpu(2, expr="key & 1")
    branches = []
    for br in range(2):
        bmp = mp.select(br)
        bmp.add(Map_GPU_Builder(
            native_gpu.gpu_affine_map(0, 1, 100 * (br + 1), dtype=0))
            .withOutputSchema([0]).withOutputBatchSize(b).build())
        branches.append(bmp)
    merged = branches[0].merge(branches[1])
    snk = wf.Sink_Builder(native.sum_sink(0)).withParallelism(1).build()
    snk.out_schema = [0]
    merged.add_sink(snk)
    g.run()
    ts, key, val = gen_batch(n, 0, 42, 31, 0)
    exp = int(val[key % 2 == 0].sum() + 100 * (key % 2 == 0).sum()
              + val[key % 2 == 1].sum() + 200 * (key % 2 == 1).sum())
    assert g.sink_sum(snk) == exp
    assert g.sink_count(snk) == n


def test_gpu_broadcast_into_parallel_windows_shape():
    """Broadcast device batches into 2 replicas of a keyed GPU window op:
    each replica sees the full stream (D2D clones), so each fires the full
    window set — 2x the single-replica count (reference
    Parallel_Windows-on-GPU broadcast shape)."""
    n, n_keys, b, win, slide = 200_000, 101, 50_000, 500, 100
    counts = []
    for par in (1, 2):
        src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=5))
               .withOutputSchema([5]).withOutputBatchSize(b).build())
        ff = (Ffat_Windows_GPU_Builder(
            native_gpu.gpu_ffat_windows(native_gpu.COMB_SUM, 0, win, slide,
                                        max_keys=1024))
            .withOutputSchema([2]).withOutputBatchSize(2 * b)
            .withParallelism(par).build())
        if par > 1:
            ff.broadcast_input = True
        snk = (Sink_GPU_Builder(native_gpu.gpu_count_sink())
               .withParallelism(par).build())
        g = wf.PipeGraph(f"gbw{par}")
        mp = g.add_source(src)
        mp.add(ff)
        mp.add(snk)
        g.run()
        counts.append(g.sink_count(snk))
    assert counts[1] == 2 * counts[0]


def test_gpu_stateful_map_bf16_column():
    """Regression: keyed stateful ops on a bf16 column (round 2 found the
    kernel doing 4-byte f32 accesses on the 2-byte column — OOB writes
    corrupting neighboring device allocations).  EMA on bf16 vs oracle."""
    n, n_keys, b = 100_000, 64, 25_000
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=5))
           .withOutputSchema([5]).withOutputBatchSize(b).build())
    sm = (Map_GPU_Builder(native_gpu.gpu_keyed_ema(0, alpha=0.5, max_keys=256))
          .withOutputSchema([5]).withOutputBatchSize(b).build())
    got = dict(s=0.0, n=0)

    def pysink(cols):
        got['s'] += float(bf16_to_f32_np(cols['c0']).astype(np.float64).sum())
        got['n'] += len(cols['c0'])

    g = wf.PipeGraph("bf16ema")
    p = g.add_source(src)
    p.chain(sm)
    snk = wf.Sink_Builder(pysink).withParallelism(1).build()
    snk.out_schema = [5]
    p.add_sink(snk)
    g.run()
    _, key, val = gen_batch(n, 0, 42, n_keys, 5)
    from windflow_amd.synth import f32_to_bf16_np
    vals = bf16_to_f32_np(val).astype(np.float64)
    ema = {}
    exp = 0.0
    for k, v in zip(key.tolist(), vals.tolist()):
        e = 0.5 * ema.get(k, 0.0) + 0.5 * v
        ema[k] = e
        # output column stores the bf16-rounded state
        exp += float(bf16_to_f32_np(f32_to_bf16_np(
            np.array([e], dtype=np.float32)))[0])
    assert got['n'] == n
    assert abs(got['s'] - exp) <= 5e-3 * max(1.0, abs(exp))


def test_gpu_ffat_dense_keys_matches_hashed():
    """withDenseKeys fast path (slot = key, no hash probe): identical
    results to the hashed mode on the same stream (gpu_source keys are
    already dense integers in [0, n_keys))."""
    n, n_keys, win, slide, b = 200_000, 1024, 500, 100, 50_000
    sums = []
    for dense in (False, True):
        src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=5))
               .withOutputSchema([5]).withOutputBatchSize(b).build())
        ff = (Ffat_Windows_GPU_Builder(
            native_gpu.gpu_ffat_windows(native_gpu.COMB_SUM, 0, win, slide,
                                        max_keys=n_keys, dense_keys=dense))
              .withOutputSchema([2]).withOutputBatchSize(2 * b).build())
        acc = dict(s=0.0, n=0, k=0)

        def pysink(cols):
            acc['s'] += float(cols['c0'].astype(np.float64).sum())
            acc['n'] += len(cols['c0'])
            acc['k'] += int(cols['key'].astype(np.int64).sum())

        g = wf.PipeGraph(f"dense{dense}")
        p = g.add_source(src)
        p.chain(ff)
        snk = wf.Sink_Builder(pysink).withParallelism(1).build()
        snk.out_schema = [2]
        p.add_sink(snk)
        g.run()
        sums.append((acc['n'], acc['k'], acc['s']))
    assert sums[0][0] == sums[1][0]
    assert sums[0][1] == sums[1][1]  # same keys emitted
    assert abs(sums[0][2] - sums[1][2]) <= 1e-6 * max(1.0, abs(sums[0][2]))


def test_gpu_dense_keys_overflow_raises():
    """A key >= max_keys in dense mode must fail loudly, not corrupt."""
    n, b = 200_000, 25_000
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, 1024, b, vdt=5))
           .withOutputSchema([5]).withOutputBatchSize(b).build())
    ff = (Ffat_Windows_GPU_Builder(
        native_gpu.gpu_ffat_windows(native_gpu.COMB_SUM, 0, 500, 100,
                                    max_keys=128, dense_keys=True))
          .withOutputSchema([2]).withOutputBatchSize(2 * b).build())
    snk = Sink_GPU_Builder(native_gpu.gpu_count_sink()).build()
    g = wf.PipeGraph("denseovf")
    p = g.add_source(src)
    p.chain(ff)
    p.chain_sink(snk)
    with pytest.raises(RuntimeError, match="withDenseKeys|dense"):
        g.run()


def test_gpu_backpressure_tiny_queues():
    """Un-chained GPU pipeline under 2-deep queues (WFA_QUEUE_CAP=2):
    source -> map -> ffat -> sink in four threads with full backpressure.
    Deadlock-freedom + exact counts (the regime where the round-2
    standalone-op bugs lived)."""
    import os
    import subprocess
    import sys
    script = r"""
import sys
sys.path.insert(0, ".")
import windflow_amd as wf
from windflow_amd import native_gpu
from windflow_amd.builders_gpu import (Source_GPU_Builder, Map_GPU_Builder,
                                       Ffat_Windows_GPU_Builder,
                                       Sink_GPU_Builder)
n, n_keys, b = 400_000, 101, 10_000   # 40 batches through 2-deep queues
src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=5))
       .withOutputSchema([5]).withOutputBatchSize(b).build())
mp_ = (Map_GPU_Builder(native_gpu.gpu_affine_map(0, 1.0, 0.0, dtype=5))
       .withOutputSchema([5]).withOutputBatchSize(b).build())
ff = (Ffat_Windows_GPU_Builder(
    native_gpu.gpu_ffat_windows(native_gpu.COMB_COUNT, 0, 300, 100,
                                max_keys=256))
    .withOutputSchema([2]).withOutputBatchSize(2 * b).build())
snk = Sink_GPU_Builder(native_gpu.gpu_count_sink()).build()
g = wf.PipeGraph("bp")
p = g.add_source(src)
p.add(mp_)   # add: separate threads + queues
p.add(ff)
p.add(snk)
g.run()
from windflow_amd.synth import gen_batch
from collections import Counter
ts, key, val = gen_batch(n, 0, 42, n_keys, 5)
per = Counter(key.tolist())
exp = sum((c - 1) // 100 + 1 for c in per.values())  # full + EOS partials
assert g.sink_count(snk) == exp, (g.sink_count(snk), exp)
print("BP_OK")
"""
    env = dict(os.environ, WFA_QUEUE_CAP="2")
    r = subprocess.run([sys.executable, "-c", script], env=env,
                       capture_output=True, text=True, timeout=300,
                       cwd=os.path.dirname(os.path.dirname(
                           os.path.abspath(__file__))))
    assert r.returncode == 0 and "BP_OK" in r.stdout, r.stderr[-2000:]


def test_gpu_h2d_overlap_recycling_exact():
    """CPU source -> GPU map -> CPU sink under 2-deep queues: the round-2
    H2D overlap holds each host batch on its copy event instead of a
    per-batch stream sync (reference forward_emitter_gpu.hpp one-batch
    overlap).  If a host batch were recycled and refilled before its
    copies landed, the exact i64 sum below would be corrupted."""
    import os
    import subprocess
    import sys
    script = r"""
import sys
sys.path.insert(0, ".")
import numpy as np
import windflow_amd as wf
from windflow_amd import native_gpu
from windflow_amd.builders_gpu import Map_GPU_Builder

n, b = 2_000_000, 8192
state = dict(pos=0)

def src(replica, par):
    p = state['pos']
    if p >= n:
        return None
    m = min(b, n - p)
    state['pos'] += m
    idx = np.arange(p, p + m, dtype=np.int64)
    return dict(ts=idx, key=(idx % 257).astype(np.uint64),
                c0=(idx * 3) % 1001, watermark=p + m)

got = dict(s=0, n=0)

def sink(cols):
    got['s'] += int(cols['c0'].sum())
    got['n'] += len(cols['c0'])

g = wf.PipeGraph("h2d")
p = g.add_source(wf.Source_Builder(src).withParallelism(1)
                 .withOutputSchema([0]).withOutputBatchSize(b).build())
mp_ = (Map_GPU_Builder(native_gpu.gpu_affine_map(0, 2, 7, dtype=0))
       .withOutputSchema([0]).withOutputBatchSize(b).build())
p.add(mp_)   # separate threads + 2-deep queues: fast host-batch recycling
snk = wf.Sink_Builder(sink).withParallelism(1).build()
snk.out_schema = [0]
p.add_sink(snk)
g.run()

idx = np.arange(n, dtype=np.int64)
exp = int(((idx * 3) % 1001 * 2 + 7).sum())
assert got['n'] == n, (got['n'], n)
assert got['s'] == exp, (got['s'], exp)
print("H2D_OK")
"""
    env = dict(os.environ, WFA_QUEUE_CAP="2")
    r = subprocess.run([sys.executable, "-c", script], env=env,
                       capture_output=True, text=True, timeout=300,
                       cwd=os.path.dirname(os.path.dirname(
                           os.path.abspath(__file__))))
    assert r.returncode == 0 and "H2D_OK" in r.stdout, r.stderr[-2000:]


def test_gpu_jit_ffat_avg_f64_accumulator():
    """acc="f64": double-precision accumulator fields/arenas.  Values are
    2^24 + (i % 7) — an f32 accumulator loses the small addends against
    the 2^24 offset, an f64 one keeps them exactly (output is still an F32
    column, quantum 2 at this magnitude)."""
    n, n_keys, b, win, slide = 120_000, 31, 15_000, 300, 100
    state = dict(pos=0)

    def src(replica, par):
        p = state['pos']
        if p >= n:
            return None
        m = min(b, n - p)
        state['pos'] += m
        idx = np.arange(p, p + m, dtype=np.int64)
        return dict(ts=idx, key=(idx % n_keys).astype(np.uint64),
                    c0=(2.0**24 + (idx % 7)).astype(np.float64),
                    watermark=p + m)

    ff = (Ffat_Windows_GPU_Builder(
        native_gpu.gpu_avg_ffat_windows(win, slide, col=0, max_keys=64,
                                        acc="f64"))
          .withOutputSchema([2]).withOutputBatchSize(2 * b).build())
    res = dict(rows=[])

    def pysink(cols):
        res['rows'].append((cols['key'].copy(), cols['c0'].copy()))

    g = wf.PipeGraph("jf64")
    p = g.add_source(wf.Source_Builder(src).withParallelism(1)
                     .withOutputSchema([1]).withOutputBatchSize(b).build())
    p.add(ff)
    snk = wf.Sink_Builder(pysink).withParallelism(1).build()
    snk.out_schema = [2]
    p.add_sink(snk)
    g.run()

    from collections import defaultdict
    idx = np.arange(n, dtype=np.int64)
    key = idx % n_keys
    val = (2.0**24 + (idx % 7)).astype(np.float64)
    per = defaultdict(list)
    for k, v in zip(key.tolist(), val.tolist()):
        per[k].append(v)
    exp = defaultdict(list)
    for k, vs in per.items():
        w = 0
        while w * slide < len(vs):
            exp[k].append(float(np.mean(vs[w * slide: w * slide + win])))
            w += 1
    got = defaultdict(list)
    for k_arr, v_arr in res['rows']:
        for k, v in zip(k_arr.tolist(), v_arr.tolist()):
            got[k].append(v)
    assert sum(map(len, got.values())) == sum(map(len, exp.values()))
    for k in exp:
        for a, bb in zip(sorted(got[k]), sorted(exp[k])):
            # f64-exact value rounded once to the F32 output column
            assert abs(a - bb) <= 1.5, (k, a, bb)


def test_gpu_jit_ffat_avg_dense_keys_matches_oracle():
    """JIT window fold with dense_keys=True (slot = key, no hash probe)
    must match the same oracle as the hashed path."""
    n, n_keys, b, win, slide = 120_000, 101, 17_000, 400, 100
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=2))
           .withOutputSchema([2]).withOutputBatchSize(b).build())
    ff = (Ffat_Windows_GPU_Builder(
        native_gpu.gpu_avg_ffat_windows(win, slide, col=0, max_keys=256,
                                        dense_keys=True))
          .withOutputSchema([2]).withOutputBatchSize(2 * b).build())
    res = dict(rows=[])

    def pysink(cols):
        res['rows'].append((cols['key'].copy(), cols['c0'].copy()))

    g = wf.PipeGraph("jdense")
    p = g.add_source(src)
    p.chain(ff)
    snk = wf.Sink_Builder(pysink).withParallelism(1).build()
    snk.out_schema = [2]
    p.add_sink(snk)
    g.run()
    ts, key, val = gen_batch(n, 0, 42, n_keys, 2)
    from collections import defaultdict
    per = defaultdict(list)
    for k, v in zip(key.tolist(), val.astype(np.float64).tolist()):
        per[k].append(v)
    exp = defaultdict(list)
    for k, vs in per.items():
        w = 0
        while w * slide < len(vs):
            exp[k].append(float(np.mean(vs[w * slide: w * slide + win])))
            w += 1
    got = defaultdict(list)
    for k_arr, v_arr in res['rows']:
        for k, v in zip(k_arr.tolist(), v_arr.tolist()):
            got[k].append(v)
    assert sum(map(len, got.values())) == sum(map(len, exp.values()))
    for k in exp:
        for a, bb in zip(sorted(got[k]), sorted(exp[k])):
            assert abs(a - bb) <= 2e-3 * max(1.0, abs(bb)), (k, a, bb)


def test_gpu_ffat_single_key_giant_segment():
    """n_keys=1: the whole batch is ONE segment (single boundary through
    the dense segment-compaction path; longest possible per-segment fold
    chain)."""
    n, b, win, slide = 300_000, 60_000, 1000, 100
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, 1, b, vdt=5))
           .withOutputSchema([5]).withOutputBatchSize(b).build())
    ff = (Ffat_Windows_GPU_Builder(
        native_gpu.gpu_ffat_windows(native_gpu.COMB_COUNT, 0, win, slide,
                                    max_keys=16))
          .withOutputSchema([2]).withOutputBatchSize(2 * b).build())
    g = wf.PipeGraph("onekey")
    p = g.add_source(src)
    p.chain(ff)
    snk = Sink_GPU_Builder(native_gpu.gpu_count_sink()).build()
    p.chain_sink(snk)
    g.run()
    # one key, n tuples: a window fires every `slide` tuples (+ EOS tails)
    exp = (n - 1) // slide + 1
    assert g.sink_count(snk) == exp, (g.sink_count(snk), exp)


def test_gpu_ffat_at_64k_boundary():
    """max_keys exactly 65536: the bounded-slot segments table still
    engages (its boundary), VIK does not (needs <= 65535) — fallback
    gather fold + dense segments together."""
    n, n_keys, b, win, slide = 600_000, 60_000, 120_000, 40, 10
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=5))
           .withOutputSchema([5]).withOutputBatchSize(b).build())
    ff = (Ffat_Windows_GPU_Builder(
        native_gpu.gpu_ffat_windows(native_gpu.COMB_COUNT, 0, win, slide,
                                    max_keys=65536))
          .withOutputSchema([2]).withOutputBatchSize(2 * b).build())
    g = wf.PipeGraph("b64k")
    p = g.add_source(src)
    p.chain(ff)
    snk = Sink_GPU_Builder(native_gpu.gpu_count_sink()).build()
    p.chain_sink(snk)
    g.run()
    ts, key, val = gen_batch(n, 0, 42, n_keys, 5)
    from collections import Counter
    per = Counter(key.tolist())
    exp = sum((c - 1) // slide + 1 for c in per.values())
    assert g.sink_count(snk) == exp, (g.sink_count(snk), exp)
