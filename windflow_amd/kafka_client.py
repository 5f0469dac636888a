"""In-repo Kafka stand-in: a TCP broker + a confluent_kafka-compatible
client subset (Consumer/Producer/TopicPartition).

librdkafka is not installable in this image, so the Kafka connectors
(windflow_amd/kafka.py) would otherwise only ever run against mocks.
This module provides a REAL wire path instead: a broker process/thread
holding topic/partition logs, and clients speaking newline-delimited JSON
over sockets — subscription with group assignment (rebalanced on member
join), per-partition offsets, produce/fetch/commit.  The connector
builders fall back to this client when confluent_kafka is absent, so the
same application code runs here end-to-end and against a real cluster in
production.

This is NOT a Kafka protocol implementation — it is the API contract the
connectors need (reference kafka_source.hpp consume loop, kafka_sink.hpp
produce), live over sockets.
"""
import base64
import json
import socket
import threading


# ===== broker =====
class FakeBroker:
    """Single-threaded-per-connection topic log server."""

    def __init__(self, port=0, n_partitions=2):
        self.n_partitions = n_partitions
        self.logs = {}          # (topic, partition) -> [bytes]
        self.commits = {}       # (group, topic, partition) -> offset
        self.members = {}       # group -> [member_id]
        self.lock = threading.Lock()
        self.srv = socket.socket()
        self.srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self.srv.bind(("127.0.0.1", port))
        self.port = self.srv.getsockname()[1]
        self.srv.listen(16)
        self._stop = False
        self._th = threading.Thread(target=self._accept, daemon=True)
        self._th.start()

    @property
    def address(self):
        return f"127.0.0.1:{self.port}"

    def close(self):
        self._stop = True
        try:
            self.srv.close()
        except OSError:
            pass

    def _accept(self):
        while not self._stop:
            try:
                conn, _ = self.srv.accept()
            except OSError:
                return
            threading.Thread(target=self._serve, args=(conn,),
                             daemon=True).start()

    def _serve(self, conn):
        f = conn.makefile("rwb")
        try:
            for line in f:
                req = json.loads(line)
                resp = self._handle(req)
                f.write(json.dumps(resp).encode() + b"\n")
                f.flush()
        except (OSError, ValueError):
            pass
        finally:
            conn.close()

    def _handle(self, req):
        op = req["op"]
        with self.lock:
            if op == "produce":
                key = (req["topic"], int(req["partition"]))
                log = self.logs.setdefault(key, [])
                log.append(base64.b64decode(req["value"]))
                return dict(ok=True, offset=len(log) - 1)
            if op == "join":
                g = req["group"]
                ms = self.members.setdefault(g, [])
                if req["member"] not in ms:
                    ms.append(req["member"])  # triggers rebalance
                return dict(ok=True, generation=len(ms))
            if op == "assign":
                # round-robin partitions of the topics over group members
                g, member = req["group"], req["member"]
                ms = self.members.get(g, [])
                idx = ms.index(member) if member in ms else 0
                parts = [(t, p) for t in req["topics"]
                         for p in range(self.n_partitions)]
                mine = [tp for i, tp in enumerate(parts)
                        if i % max(1, len(ms)) == idx]
                return dict(ok=True, assignment=mine,
                            generation=len(ms))
            if op == "fetch":
                key = (req["topic"], int(req["partition"]))
                log = self.logs.get(key, [])
                off = int(req["offset"])
                if off < len(log):
                    return dict(ok=True, offset=off,
                                value=base64.b64encode(log[off]).decode())
                return dict(ok=True, value=None)
            if op == "commit":
                self.commits[(req["group"], req["topic"],
                              int(req["partition"]))] = int(req["offset"])
                return dict(ok=True)
            if op == "committed":
                off = self.commits.get((req["group"], req["topic"],
                                        int(req["partition"])), -1)
                return dict(ok=True, offset=off)
        return dict(ok=False, error=f"unknown op {op}")


# ===== client (confluent_kafka API subset) =====
class _Conn:
    def __init__(self, brokers):
        host, port = brokers.split(",")[0].split(":")
        self.sock = socket.create_connection((host, int(port)), timeout=5)
        self.f = self.sock.makefile("rwb")
        self.lock = threading.Lock()

    def rpc(self, **req):
        with self.lock:
            self.f.write(json.dumps(req).encode() + b"\n")
            self.f.flush()
            line = self.f.readline()
        if not line:
            raise OSError("broker closed")
        return json.loads(line)

    def close(self):
        try:
            self.sock.close()
        except OSError:
            pass


class KafkaError(Exception):
    pass


class TopicPartition:
    def __init__(self, topic, partition=0, offset=0):
        self.topic, self.partition, self.offset = topic, partition, offset


class Message:
    def __init__(self, topic, partition, offset, value):
        self._t, self._p, self._o, self._v = topic, partition, offset, value

    def topic(self):
        return self._t

    def partition(self):
        return self._p

    def offset(self):
        return self._o

    def value(self):
        return self._v

    def error(self):
        return None


class Consumer:
    _seq = [0]

    def __init__(self, conf):
        self.conf = conf
        self.group = conf.get("group.id", "g")
        self.member = f"m{Consumer._seq[0]}"
        Consumer._seq[0] += 1
        self.conn = _Conn(conf["bootstrap.servers"])
        self.assignment_ = []       # [(topic, partition)]
        self.offsets = {}           # (topic, partition) -> next offset
        self.topics = []
        self.generation = -1
        self.rr = 0
        self.closed = False

    def subscribe(self, topics):
        self.topics = list(topics)
        self.conn.rpc(op="join", group=self.group, member=self.member)
        self._rebalance()

    def _rebalance(self):
        r = self.conn.rpc(op="assign", group=self.group, member=self.member,
                          topics=self.topics)
        self.assignment_ = [tuple(tp) for tp in r["assignment"]]
        self.generation = r["generation"]
        for tp in self.assignment_:
            if tp not in self.offsets:
                c = self.conn.rpc(op="committed", group=self.group,
                                  topic=tp[0], partition=tp[1])
                self.offsets[tp] = max(0, c["offset"] + 1)

    def assign(self, tps):
        self.assignment_ = [(tp.topic, tp.partition) for tp in tps]
        for tp, spec in zip(self.assignment_, tps):
            self.offsets[tp] = spec.offset

    def poll(self, timeout=0.1):
        if self.closed or not self.assignment_:
            return None
        # cooperative: refresh assignment each poll (a new member joining
        # re-partitions — the reference's rebalance callback effect)
        if self.topics:
            self._rebalance()
        n = len(self.assignment_)
        for k in range(n):
            tp = self.assignment_[(self.rr + k) % n]
            off = self.offsets.get(tp, 0)
            r = self.conn.rpc(op="fetch", topic=tp[0], partition=tp[1],
                              offset=off)
            if r.get("value") is not None:
                self.rr = (self.rr + k + 1) % n
                self.offsets[tp] = off + 1
                return Message(tp[0], tp[1], off, base64.b64decode(r["value"]))
        return None

    def commit(self):
        for (t, p), off in self.offsets.items():
            self.conn.rpc(op="commit", group=self.group, topic=t, partition=p,
                          offset=off - 1)

    def close(self):
        self.commit()
        self.closed = True
        self.conn.close()


class Producer:
    def __init__(self, conf):
        self.conf = conf
        self._conn = None  # lazy: builders construct Producers before run()

    @property
    def conn(self):
        if self._conn is None:
            self._conn = _Conn(self.conf["bootstrap.servers"])
        return self._conn

    def produce(self, topic, value, partition=0, on_delivery=None):
        try:
            r = self.conn.rpc(op="produce", topic=topic, partition=partition,
                              value=base64.b64encode(value).decode())
            if on_delivery:
                on_delivery(None if r.get("ok") else KafkaError("produce"),
                            None)
        except OSError as e:
            if on_delivery:
                on_delivery(e, None)
            else:
                raise

    def poll(self, timeout=0):
        return 0

    def flush(self, timeout=None):
        return 0
