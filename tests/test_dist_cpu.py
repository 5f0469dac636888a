"""Multi-process CPU coverage of the distributed keyby path (gloo, world 2).

The GPU path (gpu_keyby_exchange over RCCL) shares the bucketing semantics
(splitmix64 hash % world) verified here cross-process; its device plumbing
is covered by tests/test_gpu_ops.py on a real MI355X.
"""
import os
import subprocess
import sys

import pytest


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
