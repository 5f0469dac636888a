"""Persistent state tier tests (reference rocksdb_tests)."""
import os
import struct

import numpy as np

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

import windflow_amd as wf
from windflow_amd import native, _core
from windflow_amd.persistent import P_Reduce_Builder, P_Map_Builder


def test_statestore_roundtrip(tmp_path):
    s = _core.StateStore(str(tmp_path / "kv.log"), cache_capacity=4)
    for k in range(100):
        s.put(k, struct.pack("<q", k * 7))
    s.flush()
    assert len(s) >= 96  # all but the cached-dirty few are on disk pre-flush
    for k in range(100):
        assert struct.unpack("<q", s.get(k))[0] == k * 7
    assert s.get(12345) is None


def test_statestore_eviction_writeback(tmp_path):
    # cache capacity 2 forces eviction write-back on every put
    s = _core.StateStore(str(tmp_path / "kv.log"), cache_capacity=2)
    for k in range(1000):
        s.put(k % 10, struct.pack("<q", k))
    s.flush()
    for k in range(10):
        # last write per key: largest k' with k' % 10 == k
        assert struct.unpack("<q", s.get(k))[0] == 990 + k


def test_p_reduce_matches_in_memory(tmp_path):
    n = 20000
    for builder in ("mem", "disk"):
        g = wf.PipeGraph("p")
        src = (wf.Source_Builder(native.seq_source(n, 13, 256))
               .withParallelism(1).withOutputSchema([0]).build())
        mp = g.add_source(src)
        if builder == "mem":
            red = (wf.Reduce_Builder(native.keyed_sum_reduce(0))
                   .withParallelism(2).withOutputSchema([0]).build())
        else:
            red = (P_Reduce_Builder(col=0)
                   .withStatePath(str(tmp_path / "st"))
                   .withCacheCapacity(8)  # tiny: forces disk traffic
                   .withParallelism(2).withOutputSchema([0]).build())
        mp.add(red)
        snk = wf.Sink_Builder(native.sum_sink(0)).withParallelism(1).build()
        mp.add_sink(snk)
        g.run()
        if builder == "mem":
            exp = g.sink_sum(snk)
        else:
            assert g.sink_sum(snk) == exp


def test_p_map_python_state(tmp_path):
    """Persistent python map: per-key running count stored as bytes."""
    n = 5000
    counts = {}

    def fn(cols, store):
        keys = cols['key']
        for i in range(len(keys)):
            k = int(keys[i])
            prev = store.get(k)
            c = (struct.unpack("<q", prev)[0] if prev else 0) + 1
            store.put(k, struct.pack("<q", c))
            counts[k] = c
            cols['c0'][i] = c

    g = wf.PipeGraph("pm")
    src = (wf.Source_Builder(native.seq_source(n, 7, 128))
           .withParallelism(1).withOutputSchema([0]).build())
    mp = g.add_source(src)
    mp.add(P_Map_Builder(fn).withStatePath(str(tmp_path / "pm"))
           .withParallelism(1).withOutputSchema([0]).build())
    snk = wf.Sink_Builder(native.sum_sink(0)).withParallelism(1).build()
    mp.add_sink(snk)
    g.run()
    # sum over all tuples of their per-key running count: sum_k (1+..+n_k)
    per = {k: 0 for k in range(7)}
    for v in range(1, n + 1):
        per[v % 7] += 1
    exp = sum(m * (m + 1) // 2 for m in per.values())
    assert g.sink_sum(snk) == exp


def test_statestore_resume(tmp_path):
    """Self-describing log records: a store reopened with fresh=False
    rebuilds its index from disk (checkpoint/resume — beyond the reference,
    whose DBs are destroyed on teardown)."""
    p = str(tmp_path / "kv.log")
    s = _core.StateStore(p, cache_capacity=4)
    for k in range(50):
        s.put(k, struct.pack("<q", k * 3))
    s.put(7, struct.pack("<q", 777))  # overwrite: replay must keep the LAST record
    s.flush()
    del s
    r = _core.StateStore(p, cache_capacity=4, fresh=False)
    assert len(r) == 50
    assert struct.unpack("<q", r.get(7))[0] == 777
    for k in range(50):
        if k != 7:
            assert struct.unpack("<q", r.get(k))[0] == k * 3
    # and fresh=True truncates
    t = _core.StateStore(p, cache_capacity=4)
    assert len(t) == 0 and t.get(7) is None


def test_p_reduce_keep_state_resumes(tmp_path):
    """withKeepState(): a second graph run continues the accumulators of the
    first (per-replica deterministic log paths + log replay)."""
    n = 8000
    sums = []
    for _ in range(2):
        g = wf.PipeGraph("pk")
        src = (wf.Source_Builder(native.seq_source(n, 13, 256))
               .withParallelism(1).withOutputSchema([0]).build())
        mp = g.add_source(src)
        red = (P_Reduce_Builder(col=0)
               .withStatePath(str(tmp_path / "st"))
               .withKeepState()
               .withParallelism(2).withOutputSchema([0]).build())
        mp.add(red)
        snk = wf.Sink_Builder(native.last_per_key_sink(0)).withParallelism(1).build()
        mp.add_sink(snk)
        g.run()
        sums.append(g.sink_sum(snk))
    # run 2 ends with every key's accumulator at exactly 2x run 1's
    assert sums[1] == 2 * sums[0] and sums[0] > 0


def test_p_flatmap_store(tmp_path):
    """Persistent flatmap: emit one row per NEW key only (store as seen-set)."""
    n = 4000
    def fn(cols, store):
        keys = cols['key']
        new = []
        for i in range(len(keys)):
            k = int(keys[i])
            if store.get(k) is None:
                store.put(k, b"\x01")
                new.append(i)
        if not new:
            return None
        idx = np.array(new)
        return {"ts": cols['ts'][idx], "key": keys[idx], "c0": cols['c0'][idx]}

    from windflow_amd.persistent import P_FlatMap_Builder
    g = wf.PipeGraph("pfm")
    src = (wf.Source_Builder(native.seq_source(n, 97, 256))
           .withParallelism(1).withOutputSchema([0]).build())
    mp = g.add_source(src)
    mp.add(P_FlatMap_Builder(fn).withStatePath(str(tmp_path / "fm"))
           .withParallelism(1).withOutputSchema([0]).build())
    snk = wf.Sink_Builder(native.count_sink()).withParallelism(1).build()
    mp.add_sink(snk)
    g.run()
    assert g.sink_count(snk) == 97  # one row per distinct key


def _run_pkw(tmp_path, tb, win, slide, n, keys, par):
    from windflow_amd.persistent import P_Keyed_Windows_Builder
    res = []

    def sink_fn(cols):
        for i in range(len(cols['key'])):
            res.append((int(cols['key'][i]), float(cols['c0'][i])))

    g = wf.PipeGraph("pkw")
    src = (wf.Source_Builder(native.seq_source(n, keys, 256))
           .withParallelism(1).withOutputSchema([0]).build())
    mp = g.add_source(src)
    b = (P_Keyed_Windows_Builder(np.sum).withStatePath(str(tmp_path / "kw"))
         .withParallelism(par).withOutputSchema([1]))  # f64 out
    b = b.withTBWindows(win, slide) if tb else b.withCBWindows(win, slide)
    mp.add(b.build())
    mp.add_sink(wf.Sink_Builder(sink_fn).withParallelism(1).build())
    g.run()
    return res


import pytest


@pytest.mark.parametrize("case", range(6))
def test_p_keyed_windows_fuzz(case, tmp_path):
    """Random CB/TB shapes on the store-backed keyed windows vs the window
    oracles (caught the TB cursor skipping data-free gaps that must fire
    as 0)."""
    import random
    from collections import Counter
    from tests.test_windows import oracle_cb, oracle_tb
    rng = random.Random(6000 + case * 11)
    tb = rng.random() < 0.5
    slide = rng.choice([3, 7, 12, 50])
    win = slide * rng.randint(1, 6)
    keys = rng.choice([1, 5, 17, 40])
    n = rng.choice([800, 3000, 6000])
    par = rng.randint(1, 3)
    res = _run_pkw(tmp_path, tb, win, slide, n, keys, par=par)
    per = {}
    for v in range(1, n + 1):
        per.setdefault(v % keys, []).append((v, v))
    exp = (oracle_tb if tb else oracle_cb)(per, win, slide, "sum")
    got = Counter((k, float(v)) for k, v in res)
    assert got == Counter({(k, float(v)): c for (k, v), c in exp.items()}), \
        (tb, win, slide, keys, n, par)


def test_p_keyed_windows_cb(tmp_path):
    from collections import Counter
    from tests.test_windows import oracle_cb
    n, keys, win, slide = 5000, 17, 40, 12
    res = _run_pkw(tmp_path, False, win, slide, n, keys, par=2)
    per = {}
    for v in range(1, n + 1):
        per.setdefault(v % keys, []).append((v, v))
    exp = oracle_cb(per, win, slide, "sum")
    got = Counter((k, float(v)) for k, v in res)
    assert got == Counter({(k, float(v)): c for (k, v), c in exp.items()})


def test_p_keyed_windows_tb(tmp_path):
    from collections import Counter
    from tests.test_windows import oracle_tb
    n, keys, win, slide = 4000, 11, 500, 150
    res = _run_pkw(tmp_path, True, win, slide, n, keys, par=2)
    per = {}
    for v in range(1, n + 1):
        per.setdefault(v % keys, []).append((v, v))
    exp = oracle_tb(per, win, slide, "sum")
    got = Counter((k, float(v)) for k, v in res)
    assert got == Counter({(k, float(v)): c for (k, v), c in exp.items()})


def test_varkv_compaction_preserves_state(tmp_path, monkeypatch):
    """Force log compaction (low threshold via env) under heavy overwrite,
    then verify values survive both in-process and across a resume (the
    compacted log keeps the self-describing record format)."""
    import os
    import subprocess
    import sys
    # compaction threshold is latched per process: run in a child
    code = f"""
import struct, sys
sys.path.insert(0, {str(ROOT)!r})
from windflow_amd import _core
p = {str(tmp_path / 'c.log')!r}
s = _core.StateStore(p, cache_capacity=2)
for rep in range(200):
    for k in range(50):
        s.put(k, struct.pack('<q', rep * 1000 + k) * 8)  # 64B values
s.flush()
for k in range(50):
    assert struct.unpack('<q', s.get(k)[:8])[0] == 199000 + k
del s
r = _core.StateStore(p, cache_capacity=2, fresh=False)
assert len(r) == 50
for k in range(50):
    assert struct.unpack('<q', r.get(k)[:8])[0] == 199000 + k
print('COMPACT_OK')
"""
    env = dict(os.environ, WFA_KV_COMPACT_BYTES="100000")
    out = subprocess.run([sys.executable, "-c", code], env=env,
                         capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr[-800:]
    assert "COMPACT_OK" in out.stdout
    # the log actually compacted: far smaller than the ~650KB written
    assert (tmp_path / "c.log").stat().st_size < 300_000


def test_statestore_resume_torn_tail(tmp_path):
    """A crash mid-append leaves a torn final record; replay must ignore it
    and keep every complete record."""
    p = str(tmp_path / "torn.log")
    s = _core.StateStore(p, cache_capacity=2)
    for k in range(20):
        s.put(k, struct.pack("<q", k * 11))
    s.flush()
    del s
    with open(p, "ab") as f:      # simulate a crash: header + half a value
        f.write(struct.pack("<QI", 99, 64) + b"\x01" * 10)
    r = _core.StateStore(p, cache_capacity=2, fresh=False)
    assert len(r) == 20 and r.get(99) is None
    for k in range(20):
        assert struct.unpack("<q", r.get(k))[0] == k * 11
    r.put(99, struct.pack("<q", 7))   # tail reclaimed cleanly
    r.flush()
    del r
    r2 = _core.StateStore(p, cache_capacity=2, fresh=False)
    assert struct.unpack("<q", r2.get(99))[0] == 7


def test_statestore_erase_and_resume(tmp_path):
    """erase() writes a tombstone: the key stays gone after a reopen
    (without it, replay would resurrect the old put)."""
    p = str(tmp_path / "er.log")
    s = _core.StateStore(p, cache_capacity=4)
    for k in range(10):
        s.put(k, struct.pack("<q", k))
    s.flush()
    assert s.erase(3) and not s.erase(3)
    assert s.get(3) is None and len(s) == 9
    s.flush()
    del s
    r = _core.StateStore(p, cache_capacity=4, fresh=False)
    assert len(r) == 9 and r.get(3) is None
    for k in range(10):
        if k != 3:
            assert struct.unpack("<q", r.get(k))[0] == k
    # re-put after erase works and survives another resume
    r.put(3, struct.pack("<q", 333))
    r.flush()
    del r
    r2 = _core.StateStore(p, cache_capacity=4, fresh=False)
    assert struct.unpack("<q", r2.get(3))[0] == 333


@pytest.mark.parametrize("seed", range(3))
def test_statestore_model_fuzz(seed, tmp_path):
    """Model-based store check vs a python dict: random put/get/erase/flush
    with mid-stream reopens and forced compaction (12-seed campaign ran
    clean; caught erase misreporting for dirty-cache-only keys)."""
    import os as _os
    import subprocess
    import sys as _sys
    code = f"""
import random, sys
sys.path.insert(0, {str(ROOT)!r})
from windflow_amd import _core
rng = random.Random({seed})
model = {{}}
path = {str(tmp_path / 'm.log')!r}
s = _core.StateStore(path, cache_capacity=rng.choice([1, 4, 64]))
for step in range(2000):
    op = rng.random()
    k = rng.randint(0, 100)
    if op < 0.45:
        v = bytes(rng.getrandbits(8) for _ in range(rng.randint(1, 60)))
        s.put(k, v); model[k] = v
    elif op < 0.6:
        assert s.erase(k) == (k in model), (step, k)
        model.pop(k, None)
    elif op < 0.85:
        assert s.get(k) == model.get(k), (step, k)
    elif op < 0.93:
        s.flush()
    else:
        s.flush(); del s
        s = _core.StateStore(path, cache_capacity=rng.choice([1, 4, 64]),
                             fresh=False)
        assert len(s) == len(model), (step, len(s), len(model))
s.flush(); del s
r = _core.StateStore(path, cache_capacity=4, fresh=False)
assert len(r) == len(model)
for k, v in model.items():
    assert r.get(k) == v
print("MODEL_OK")
"""
    env = dict(_os.environ, WFA_KV_COMPACT_BYTES="15000")
    out = subprocess.run([_sys.executable, "-c", code], env=env,
                         capture_output=True, text=True, timeout=120)
    assert out.returncode == 0 and "MODEL_OK" in out.stdout, \
        (out.stderr or out.stdout)[-500:]


def test_p_map_keep_state_resumes_and_cleanup(tmp_path):
    """Python persistent ops honor withKeepState(): a NAMED P_Map's state
    path is deterministic (user path + op name), keep=True reopens the same
    log with fresh=False (state survives runs), and without keep the log
    file is deleted when the graph closes the operator (reference
    db_handle.hpp deleteDb)."""
    import glob

    def counting(cols, store):
        for i in range(len(cols['key'])):
            k = int(cols['key'][i])
            prev = store.get(k)
            c = (struct.unpack("<q", prev)[0] if prev else 0) + 1
            store.put(k, struct.pack("<q", c))
            cols['c0'][i] = c

    def run_once(keep):
        g = wf.PipeGraph("pmk")
        src = (wf.Source_Builder(native.seq_source(100, 4, 32))
               .withParallelism(1).withOutputSchema([0]).build())
        mp = g.add_source(src)
        b = (P_Map_Builder(counting).withName("counter")
             .withStatePath(str(tmp_path / "pm")))
        if keep:
            b = b.withKeepState()
        mp.add(b.withParallelism(1).withOutputSchema([0]).build())
        snk = wf.Sink_Builder(native.sum_sink(0)).withParallelism(1).build()
        mp.add_sink(snk)
        g.run()
        return g.sink_sum(snk)

    # run 1 with keep: counts 1..25 per key (100 tuples over 4 keys)
    s1 = run_once(keep=True)
    assert s1 == 4 * (25 * 26 // 2)
    logs = glob.glob(str(tmp_path / "pm.pmap.counter.log"))
    assert logs, "deterministic state path (user path + op name) not found"
    # run 2 with keep: state RESUMED -> counts 26..50 per key
    s2 = run_once(keep=True)
    assert s2 == 4 * (50 * 51 // 2 - 25 * 26 // 2)
    # run 3 without keep: fresh store (counts restart), log deleted at close
    s3 = run_once(keep=False)
    assert s3 == s1
    assert not glob.glob(str(tmp_path / "pm.pmap.counter.log"))
