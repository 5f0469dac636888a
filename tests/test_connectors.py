"""Connector layer tests (reference kafka_tests — gated: no broker/client
in this environment, so the Kafka builders' surface is checked and the
generic connector source is exercised end-to-end)."""
import numpy as np
import pytest

import windflow_amd as wf
from windflow_amd import native
from windflow_amd.kafka import (Connector_Source_Builder, Kafka_Source_Builder,
                                Kafka_Sink_Builder, _have_kafka)


def test_connector_source_end_to_end():
    chunks = [dict(ts=np.arange(i * 10, i * 10 + 10, dtype=np.int64),
                   key=np.zeros(10, dtype=np.uint64),
                   c0=np.full(10, i + 1, dtype=np.int64),
                   watermark=i * 10 + 9)
              for i in range(20)]

    state = dict(i=0)

    def feed(replica, parallelism):
        if state["i"] >= len(chunks):
            return None
        state["i"] += 1
        return chunks[state["i"] - 1]

    g = wf.PipeGraph("conn")
    src = (Connector_Source_Builder(feed).withParallelism(1)
           .withOutputSchema([0]).withOutputBatchSize(64).build())
    mp = g.add_source(src)
    snk = wf.Sink_Builder(native.sum_sink(0)).withParallelism(1).build()
    mp.add_sink(snk)
    g.run()
    assert g.sink_sum(snk) == sum(10 * (i + 1) for i in range(20))
    assert g.sink_count(snk) == 200


def test_kafka_builders_surface():
    b = (Kafka_Source_Builder(lambda v, out: False)
         .withBrokers("b:9092").withTopics("t1", "t2").withGroupID("g")
         .withIdleness(50).withOffsets([("t1", 0, 100)]))
    if _have_kafka():
        b.build()
    else:
        with pytest.raises(RuntimeError, match="confluent_kafka"):
            b.build()
    s = Kafka_Sink_Builder(lambda cols, i: ("t", 0, b"x")).withBrokers("b:9092")
    if not _have_kafka():
        with pytest.raises(RuntimeError, match="confluent_kafka"):
            s.build()
