#!/usr/bin/env python3
"""User join functions + the live-socket Kafka stand-in.  CPU-only:
    python examples/08_user_join_and_kafka.py

1. Interval join with an ARBITRARY predicate/result function (vectorized
   over matched pairs — reference interval_join.hpp user functor).
2. Kafka source/sink running against the in-repo socket broker
   (windflow_amd.kafka_client) — same application code works against a
   real cluster with confluent_kafka installed.
"""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import windflow_amd as wf                                   # noqa: E402
from windflow_amd import native                             # noqa: E402
from windflow_amd.builders import Interval_Join_Builder     # noqa: E402
from windflow_amd.kafka import (Kafka_Source_Builder,       # noqa: E402
                                Kafka_Sink_Builder)
from windflow_amd.kafka_client import FakeBroker, Producer, Consumer  # noqa: E402

# ---- 1. user join function ----
N, KEYS = 50_000, 17


def joinfn(pairs):
    """Keep pairs whose values differ by less than 8; result = |a - b|."""
    diff = np.abs(pairs["a"] - (pairs["b"] - 1_000_000_000))
    return diff < 8, diff


g = wf.PipeGraph("user_join")
a = g.add_source(wf.Source_Builder(native.seq_source(N, KEYS, 1024))
                 .withParallelism(1).withOutputSchema([0]).build())
b = g.add_source(wf.Source_Builder(
    native.seq_source(N, KEYS, 1024, value_offset=1_000_000_000))
    .withParallelism(1).withOutputSchema([0]).build())
mp = a.merge(b)
mp.add(Interval_Join_Builder(joinfn).withBoundaries(-10, 10).withKPMode()
       .withValueCols(0).withParallelism(2)
       .withOutputSchema([1]).withOutputBatchSize(1024).build())
snk = wf.Sink_Builder(native.sum_sink_f(0, dtype="f64")).withParallelism(1).build()
mp.add_sink(snk)
g.run()
print(f"join: sum of kept |a-b| = {g.sink_sum_f(snk):.0f}")

# ---- 2. Kafka round trip over the socket broker ----
brk = FakeBroker(n_partitions=2)
prod = Producer({"bootstrap.servers": brk.address})
for v in range(1, 501):
    prod.produce("events", b"%d" % v, partition=v % 2)
prod.flush()

idle = dict(n=0)


def deser(payload, out):
    if payload is None:
        idle["n"] += 1
        return idle["n"] < 5
    v = int(payload)
    out["ts"] = np.array([v], np.int64)
    out["key"] = np.array([v % 7], np.uint64)
    out["c0"] = np.array([v], np.int64)
    return True


def ser(cols, i):
    return ("doubled", int(cols["key"][i]) % 2, b"%d" % int(cols["c0"][i]))


gk = wf.PipeGraph("kafka_roundtrip")
mpk = gk.add_source(Kafka_Source_Builder(deser).withBrokers(brk.address)
                    .withTopics("events").withGroupID("demo")
                    .withParallelism(1).withOutputSchema([0]).build())
mpk.add(wf.Map_Builder(native.affine_map(0, 2, 0)).withParallelism(1)
        .withOutputSchema([0]).build())
mpk.add_sink(Kafka_Sink_Builder(ser).withBrokers(brk.address)
             .withParallelism(1).build())
gk.run()

c = Consumer({"bootstrap.servers": brk.address, "group.id": "check"})
c.subscribe(["doubled"])
got, idle_polls = [], 0
while idle_polls < 5:
    m = c.poll(0.01)
    if m is None:
        idle_polls += 1
        continue
    idle_polls = 0
    got.append(int(m.value()))
c.close()
brk.close()
print(f"kafka: {len(got)} messages round-tripped, "
      f"sum={sum(got)} (expected {sum(2 * v for v in range(1, 501))})")
