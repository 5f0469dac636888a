#!/usr/bin/env python3
"""Keyed running aggregation — the classic streaming hello-world.

Source (python, any iterator) -> keyby -> Reduce (running per-key sum)
-> Sink.  CPU-only; run anywhere: python examples/01_wordcount_style.py
"""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import windflow_amd as wf
from windflow_amd import native

state = dict(pos=0)
N, KEYS = 200_000, 50


def source(replica, parallelism):
    p = state["pos"]
    if p >= N:
        return None
    n = min(4096, N - p)
    v = np.arange(p + 1, p + n + 1, dtype=np.int64)
    state["pos"] += n
    return dict(ts=v, key=(v % KEYS).astype(np.uint64), c0=v, watermark=int(v[-1]))


totals = {}


def sink(cols):
    for k, v in zip(cols["key"], cols["c0"]):
        totals[int(k)] = int(v)  # running sum per key (last wins)


g = wf.PipeGraph("wordcount")
mp = g.add_source(wf.Source_Builder(source).withParallelism(1)
                  .withOutputSchema([0]).withOutputBatchSize(4096).build())
mp.add(wf.Reduce_Builder(native.keyed_sum_reduce(0)).withParallelism(4)
       .withOutputSchema([0]).build())
mp.add_sink(wf.Sink_Builder(sink).withParallelism(1).build())
g.run()
print(f"{len(totals)} keys; key 0 total = {totals[0]}")
assert totals[0] == sum(v for v in range(1, N + 1) if v % KEYS == 0)
print("OK")
