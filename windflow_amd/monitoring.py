"""Observability: per-replica stats reporting + dashboard push.

Mirrors the reference's L7:
 - Stats_Record per replica        -> Engine.stats() (csrc/engine, always on;
   the reference needs -DWF_TRACING_ENABLED at compile time)
 - PipeGraph::generateJSONStats    -> PipeGraph.stats_json()
 - generateSVGDiagram (graphviz)   -> PipeGraph DOT via graph_dot()
 - MonitoringThread (monitoring.hpp:162): 1 Hz TCP push of length-prefixed
   JSON reports to WF_DASHBOARD_MACHINE:WF_DASHBOARD_PORT (default
   localhost:20207), same wire format idea (4-byte big-endian length +
   payload); an SVG/DOT frame precedes the first JSON report.
 - RSS sampling from /proc (monitoring.hpp:51-66).
"""
import json
import os
import socket
import struct
import threading
import time


def rss_kb():
    try:
        with open("/proc/self/status") as f:
            for line in f:
                if line.startswith("VmRSS:"):
                    return int(line.split()[1])
    except OSError:
        pass
    return 0


def graph_dot(pg):
    """Graphviz DOT of the PipeGraph (reference generateSVGDiagram;
    we emit DOT text — render externally where graphviz exists)."""
    lines = ["digraph G {", "  rankdir=LR;", "  node [shape=box];"]
    for i, node in enumerate(pg.nodes):
        label = "|".join(op.name or op.kind for op in node.ops)
        dev = f"\\nGPU{node.ops[0].device}" if node.ops[0].gpu else ""
        lines.append(f'  n{i} [label="{label} (x{node.parallelism}){dev}"];')
    for e in pg.edges:
        style = {0: "", 1: ' [label="keyby"]', 2: ' [label="broadcast"]',
                 3: ' [label="rebalance"]'}.get(int(e["routing"]), "")
        lines.append(f'  n{e["src"]} -> n{e["dst"]}{style};')
    lines.append("}")
    return "\n".join(lines)


class MonitoringThread:
    """1 Hz reporter: pushes length-prefixed frames over TCP while the graph
    runs.  Frame 0: {"type":"diagram","dot":...}; then {"type":"report",...}
    every second; final report on stop."""

    def __init__(self, pg, host=None, port=None, interval=1.0):
        self.pg = pg
        self.host = host or os.environ.get("WF_DASHBOARD_MACHINE", "localhost")
        self.port = int(port or os.environ.get("WF_DASHBOARD_PORT", "20207"))
        self.interval = interval
        self._stop = threading.Event()
        self._th = None
        self.sock = None

    def _send(self, obj):
        payload = json.dumps(obj).encode()
        self.sock.sendall(struct.pack(">I", len(payload)) + payload)

    def _report(self):
        return dict(type="report", ts=time.time(), graph=self.pg.name,
                    rss_kb=rss_kb(),
                    dropped=self.pg.getNumDroppedTuples() if self.pg.engine else 0,
                    replicas=self.pg.stats())

    def _run(self):
        try:
            self.sock = socket.create_connection((self.host, self.port), timeout=2)
            self._send(dict(type="diagram", graph=self.pg.name,
                            dot=graph_dot(self.pg),
                            nodes=[dict(id=i,
                                        label="|".join(op.name or op.kind
                                                       for op in nd.ops),
                                        parallelism=nd.parallelism,
                                        gpu=bool(nd.ops[0].gpu),
                                        device=nd.ops[0].device)
                                   for i, nd in enumerate(self.pg.nodes)],
                            edges=[dict(src=e["src"], dst=e["dst"],
                                        routing=int(e["routing"]))
                                   for e in self.pg.edges]))
            while not self._stop.wait(self.interval):
                self._send(self._report())
            self._send(self._report())
            self.sock.close()
        except OSError:
            pass  # dashboard absent: tracing silently off (reference behavior)

    def start(self):
        self._th = threading.Thread(target=self._run, daemon=True)
        self._th.start()
        return self

    def stop(self):
        self._stop.set()
        if self._th:
            self._th.join(timeout=3)


def dump_stats(pg, path):
    """Log-file dump at wait_end (reference pipegraph.hpp:717-739)."""
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    with open(path, "w") as f:
        f.write(pg.stats_json())
    return path
