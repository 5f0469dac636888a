#!/usr/bin/env python3
"""Checkpoint/resume with the persistent state tier (CPU-only).

Run 1 sums a keyed stream into a persistent store with withKeepState();
run 2 opens the same state dir and CONTINUES the accumulators — something
the reference cannot do (its RocksDB handles are destroyed with the
graph).  See PARITY.md "Beyond the reference: checkpoint/resume".
"""
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import windflow_amd as wf
from windflow_amd import native
from windflow_amd.persistent import P_Reduce_Builder

state_dir = os.path.join(tempfile.gettempdir(), "wfa_resume_demo")
# start clean so re-running the demo is deterministic
import glob
for f in glob.glob(state_dir + "*"):
    os.unlink(f)

def run_once(label):
    g = wf.PipeGraph("resume_demo")
    src = (wf.Source_Builder(native.seq_source(100_000, 64, 4096))
           .withParallelism(1).withOutputSchema([0]).build())
    mp = g.add_source(src)
    mp.add(P_Reduce_Builder(col=0)
           .withStatePath(state_dir)
           .withKeepState()                 # <- resume across runs
           .withParallelism(2).withOutputSchema([0]).build())
    snk = (wf.Sink_Builder(native.last_per_key_sink(0))
           .withParallelism(1).build())
    mp.add_sink(snk)
    g.run()
    total = g.sink_sum(snk)
    print(f"{label}: sum of final per-key accumulators = {total}")
    return total

a = run_once("run 1 (fresh state)")
b = run_once("run 2 (resumed)    ")
assert b == 2 * a, "resume failed"
print("run 2 ended with exactly 2x run 1's accumulators -> state resumed OK")
