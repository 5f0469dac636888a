#!/usr/bin/env python3
"""Sliding windows four ways: keyed, FlatFAT, paned, mapreduce — all
producing the same results (differential check included).  CPU-only."""
import os
import sys
from collections import Counter

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import windflow_amd as wf
from windflow_amd import native
from windflow_amd.builders import (Keyed_Windows_Builder, Ffat_Windows_Builder,
                                   Paned_Windows_Builder, MapReduce_Windows_Builder)

N, KEYS, WIN, SLIDE = 100_000, 32, 1000, 100


def run(op):
    g = wf.PipeGraph("win")
    mp = g.add_source(wf.Source_Builder(native.seq_source(N, KEYS, 2048))
                      .withParallelism(1).withOutputSchema([0]).build())
    mp.add(op)
    rows = Counter()

    def sink(cols):
        for k, v in zip(cols["key"], cols["c0"]):
            rows[(int(k), int(v))] += 1

    mp.add_sink(wf.Sink_Builder(sink).withParallelism(1).build())
    g.run()
    return rows


results = {
    "keyed": run(Keyed_Windows_Builder(func=("sum", 0)).withCBWindows(WIN, SLIDE)
                 .withParallelism(4).withOutputSchema([0]).build()),
    "ffat": run(Ffat_Windows_Builder(comb=("sum", 0)).withCBWindows(WIN, SLIDE)
                .withParallelism(4).withOutputSchema([0]).build()),
    "paned": run(Paned_Windows_Builder(plq_func=("sum", 0)).withCBWindows(WIN, SLIDE)
                 .withParallelism(4).withOutputSchema([0]).build()),
    "mapreduce": run(MapReduce_Windows_Builder(map_func=("sum", 0))
                     .withCBWindows(WIN, SLIDE).withParallelism(2)
                     .withOutputSchema([0]).build()),
}
base = results["keyed"]
for name, r in results.items():
    assert r == base, name
    print(f"{name:<10} {sum(r.values())} windows  OK")
