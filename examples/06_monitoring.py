#!/usr/bin/env python3
"""Observability: per-replica stats, graph DOT, and the 1 Hz dashboard
push (reference WF_TRACING_ENABLED + MonitoringThread)."""
import json
import os
import socket
import struct
import sys
import threading

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import windflow_amd as wf
from windflow_amd import native
from windflow_amd.monitoring import MonitoringThread, graph_dot

# a tiny dashboard: accept one connection, print frame types
frames = []
srv = socket.socket()
srv.bind(("127.0.0.1", 0))
srv.listen(1)
port = srv.getsockname()[1]

def dashboard():
    conn, _ = srv.accept()
    try:
        while True:
            hdr = conn.recv(4, socket.MSG_WAITALL)
            if len(hdr) < 4:
                return
            (ln,) = struct.unpack(">I", hdr)
            payload = b""
            while len(payload) < ln:
                chunk = conn.recv(ln - len(payload))
                if not chunk:
                    return
                payload += chunk
            frames.append(json.loads(payload))
    except OSError:
        pass

threading.Thread(target=dashboard, daemon=True).start()

g = wf.PipeGraph("monitored")
src = (wf.Source_Builder(native.seq_source(2_000_000, 64, 8192))
       .withParallelism(2).withOutputSchema([0]).build())
mp = g.add_source(src)
mp.add(wf.Map_Builder(native.affine_map(0, 3, 1)).withParallelism(2)
       .withOutputSchema([0]).build())
mp.add_sink(wf.Sink_Builder(native.sum_sink(0)).withParallelism(1).build())

print(graph_dot(g).splitlines()[0], "...", f"({len(graph_dot(g).splitlines())} DOT lines)")
mon = MonitoringThread(g, host="127.0.0.1", port=port, interval=0.2)
mon.start()
g.run()
mon.stop()

stats = json.loads(g.stats_json())
print(f"operators reported: {len(stats['operators'])}")
for op in stats['operators']:
    print(f"  {op['name']} (x{op['replicas']}): tuples_in={op['tuples_in']}"
          f" tuples_out={op['tuples_out']} svc_us={op['svc_us'][0]:.3f}")
print(f"dashboard received {len(frames)} frames "
      f"(first: {frames[0]['type'] if frames else 'none'})")
