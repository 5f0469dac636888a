#!/usr/bin/env python3
"""Interval join of two streams + paned windows over the joined pairs,
with live dashboard tracing (run scripts/dashboard_server.py to watch)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import windflow_amd as wf
from windflow_amd import native
from windflow_amd.builders import Interval_Join_Builder, Paned_Windows_Builder

N, KEYS = 200_000, 100
g = wf.PipeGraph("join_demo", tracing=bool(os.environ.get("WF_TRACING")))
a = g.add_source(wf.Source_Builder(native.seq_source(N, KEYS, 1024))
                 .withParallelism(1).withOutputSchema([0]).build())
b = g.add_source(wf.Source_Builder(native.seq_source(N, KEYS, 1024,
                                                     value_offset=10_000_000))
                 .withParallelism(1).withOutputSchema([0]).build())
mp = a.merge(b)
mp.add(Interval_Join_Builder().withBoundaries(-KEYS, KEYS).withKPMode()
       .withValueCols(0).withParallelism(4)
       .withOutputSchema([0, 0]).build())
mp.add(Paned_Windows_Builder(plq_func=("sum", 0)).withTBWindows(5000, 500)
       .withParallelism(4).withOutputSchema([0]).build())
snk = wf.Sink_Builder(native.count_sink()).withParallelism(1).build()
mp.add_sink(snk)
g.run()
print(f"window results: {g.sink_count(snk)}; dropped: {g.getNumDroppedTuples()}")
print(g.stats_json()[:400], "...")
