"""Kafka integration (reference wf/kafka/: Kafka_Source :125, Kafka_Sink :71,
builders_kafka.hpp) + the generic external-connector layer it builds on.

The reference consumes via librdkafka's KafkaConsumer::consume(idleTime)
loop inside the source replica and produces via producer->produce per
serialized tuple.  Here:

 - Connector_Source_Builder: any Python iterator/callable yielding column
   dicts becomes a source replica loop — the engine-facing contract the
   Kafka source plugs into (also usable for files, sockets, ...).
 - Kafka_Source_Builder / Kafka_Sink_Builder: same fluent surface as the
   reference (withBrokers/withTopics/withGroupID/withIdleness/withOffsets);
   they require `confluent_kafka` (librdkafka) at build() time and raise a
   clear error when the client library is absent (this image has none).

The user deserializer mirrors the reference signature (kafka_source.hpp:271):
  deser(msg_value: bytes, shipper: dict-appender) -> bool  (False = stop)
"""
from .operators import Operator
from .builders import _BasicBuilder


def _have_kafka():
    try:
        import confluent_kafka  # noqa: F401
        return True
    except ImportError:
        return False


def _client():
    """confluent_kafka when installed (production), else the in-repo
    socket client (windflow_amd.kafka_client) speaking the same API —
    the connector path runs end-to-end either way."""
    try:
        import confluent_kafka
        return confluent_kafka
    except ImportError:
        from . import kafka_client
        return kafka_client


class Connector_Source_Builder(_BasicBuilder):
    """Source from an external iterator: fn(replica, parallelism) -> column
    dict or None (end of stream) — the engine's PySourceLogic contract."""
    _kind = "source"


class Kafka_Source_Builder(_BasicBuilder):
    """reference builders_kafka.hpp:191-258."""
    _kind = "source"

    def __init__(self, deser):
        super().__init__(None)
        self._deser = deser
        self._cfg = dict(brokers="localhost:9092", topics=[], group="wf",
                         idle_ms=100, offsets=None, policy="roundrobin")

    def withBrokers(self, brokers):
        self._cfg["brokers"] = brokers
        return self

    def withTopics(self, *topics):
        self._cfg["topics"] = list(topics)
        return self

    def withGroupID(self, gid):
        self._cfg["group"] = gid
        return self

    def withAssignmentPolicy(self, policy):
        self._cfg["policy"] = policy
        return self

    def withIdleness(self, ms):
        self._cfg["idle_ms"] = int(ms)
        return self

    def withOffsets(self, offsets):
        """list of (topic, partition, offset) (reference withOffsets)."""
        self._cfg["offsets"] = list(offsets)
        return self

    def build(self):
        mod = _client()
        Consumer, TopicPartition = mod.Consumer, mod.TopicPartition
        cfg, deser = self._cfg, self._deser

        # One Consumer per source REPLICA (keyed by replica index): librdkafka
        # consumers are not thread-safe, and a replica whose deser stops must
        # close only its own consumer (reference kafka_source.hpp creates one
        # consumer per replica in svc_init).
        consumers = {}

        def _consumer_for(replica):
            c = consumers.get(replica)
            if c is None:
                c = Consumer({
                    "bootstrap.servers": cfg["brokers"],
                    "group.id": cfg["group"],
                    "partition.assignment.strategy": cfg["policy"],
                    "auto.offset.reset": "earliest",
                })
                if cfg["offsets"]:
                    c.assign([TopicPartition(t, p, o)
                              for (t, p, o) in cfg["offsets"]])
                else:
                    c.subscribe(cfg["topics"])
                consumers[replica] = c
            return c

        def source_fn(replica, parallelism):
            c = _consumer_for(replica)
            msg = c.poll(cfg["idle_ms"] / 1000.0)
            out = {}
            more = deser(msg.value() if msg is not None and not msg.error()
                         else None, out)
            if not more:
                c.close()
                consumers.pop(replica, None)
                return None
            return out if out else {}

        op = self._op.clone()
        op.logic = source_fn
        return op


class Kafka_Sink_Builder(_BasicBuilder):
    """reference kafka_sink.hpp:71: user serializer returns
    (topic, partition, payload bytes) per row."""
    _kind = "sink"

    def __init__(self, ser):
        super().__init__(None)
        self._ser = ser
        self._cfg = dict(brokers="localhost:9092")

    def withBrokers(self, brokers):
        self._cfg["brokers"] = brokers
        return self

    def build(self):
        Producer = _client().Producer
        cfg, ser = self._cfg, self._ser
        prod = Producer({"bootstrap.servers": cfg["brokers"]})
        failures = []

        def _on_delivery(err, msg):
            if err is not None:
                failures.append(err)

        def sink_fn(cols):
            n = len(cols["ts"])
            for i in range(n):
                topic, part, payload = ser(cols, i)
                # BufferError = librdkafka's local queue is full: drain
                # delivery callbacks and retry instead of losing the row
                while True:
                    try:
                        prod.produce(topic, payload, partition=part,
                                     on_delivery=_on_delivery)
                        break
                    except BufferError:
                        prod.poll(0.1)
            prod.poll(0)
            if failures:
                raise RuntimeError(f"Kafka delivery failed: {failures[0]}")

        def _flush():
            prod.flush()
            if failures:
                raise RuntimeError(f"Kafka delivery failed: {failures[0]}")

        sink_fn.on_eos = _flush
        op = self._op.clone()
        op.logic = sink_fn
        return op
