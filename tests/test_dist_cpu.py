"""Multi-process CPU coverage of the distributed keyby path (gloo, world 2).

The GPU path (gpu_keyby_exchange over RCCL) shares the bucketing semantics
(splitmix64 hash % world) verified here cross-process; its device plumbing
is covered by tests/test_gpu_ops.py on a real MI355X.
"""
import os
import subprocess
import sys

import pytest


def _run_world(world, port, extra_env=None):
    env = dict(os.environ)
    env.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
               WORLD_SIZE=str(world), GLOO_SOCKET_IFNAME="lo")
    if extra_env:
        env.update(extra_env)
    worker = os.path.join(os.path.dirname(__file__), "dist_worker.py")
    procs = []
    for r in range(world):
        e = dict(env, RANK=str(r))
        procs.append(subprocess.Popen([sys.executable, worker], env=e,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=110)
        outs.append(out.decode())
    for p, o in zip(procs, outs):
        assert p.returncode == 0, o
    assert "DIST_CPU_OK" in outs[0], outs[0]


@pytest.mark.timeout(300)
@pytest.mark.parametrize("case", range(4))
def test_cpu_keyby_exchange_fuzz(case):
    """Exchange correctness across worlds 2-3 and varied key/batch shapes:
    keys partitioned by the shared splitmix64 hash, nothing lost or
    duplicated (sum+count conservation across ranks)."""
    import random
    rng = random.Random(4000 + case)
    world = rng.choice([2, 3])
    batch = rng.choice([100, 500, 2048])
    n = batch * rng.randint(4, 30)
    keys = rng.choice([1, 7, 64, 1000])
    _run_world(world, 29600 + case,
               dict(WFZ_N=str(n), WFZ_KEYS=str(keys), WFZ_BATCH=str(batch)))


@pytest.mark.timeout(120)
def test_cpu_keyby_exchange_world2():
    env = dict(os.environ)
    env.update(MASTER_ADDR="127.0.0.1", MASTER_PORT="29571",
               WORLD_SIZE="2", GLOO_SOCKET_IFNAME="lo")
    worker = os.path.join(os.path.dirname(__file__), "dist_worker.py")
    procs = []
    for r in range(2):
        e = dict(env, RANK=str(r))
        procs.append(subprocess.Popen([sys.executable, worker], env=e,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=110)
        outs.append(out.decode())
    for p, o in zip(procs, outs):
        assert p.returncode == 0, o
    assert "DIST_CPU_OK" in outs[0], outs[0]


@pytest.mark.timeout(180)
def test_bench_driver_contract_world2_cpu():
    """The driver launches bench.py via torch.distributed.run with one rank
    per GPU; this runs the same launch shape on CPU (gloo, world=2,
    --config cpu) and checks the contract: exactly ONE JSON line from
    rank 0 with the required fields and whole-job aggregation."""
    import json
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29573", "bench.py", "--config", "cpu",
         "--gpus", "2", "--steps", "8", "--warmup", "2"],
        cwd=root, env=dict(os.environ, GLOO_SOCKET_IFNAME="lo"),
        capture_output=True, text=True, timeout=170)
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [l for l in r.stdout.splitlines() if l.startswith('{"metric"')]
    assert len(lines) == 1, r.stdout[-2000:]
    d = json.loads(lines[0])
    for k in ("metric", "value", "unit", "steps", "warmup", "ms_per_step",
              "higher_is_better", "scaling", "vs_baseline", "dtype", "data",
              "config"):
        assert k in d, k
    assert d["steps"] == 8 and d["warmup"] == 2
    assert d["value"] > 0 and d["data"] == "synthetic"
