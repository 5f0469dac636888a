#!/usr/bin/env python3
"""windflow_amd dashboard (counterpart of the reference's Java Spring
server + React client, dashboard/Server + dashboard/Client).

Two listeners:
  - TCP ingest (default :20207): MonitoringThread connections pushing
    length-prefixed JSON frames (see windflow_amd/monitoring.py) — a
    diagram frame (graph structure + DOT) then 1 Hz reports.
  - HTTP UI (default :20208): live HTML view of every connected graph —
    per-operator throughput table, RSS/dropped counters, and an inline
    SVG diagram (layered DAG layout; the reference shells out to
    graphviz, which this image lacks).

Usage: python scripts/dashboard_server.py [--port 20207] [--http 20208]
       [--save DIR] [--once]  (--once: exit after the first graph ends —
       used by the self-test)
"""
import argparse
import html
import json
import os
import socket
import struct
import threading
import time
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

STATE = dict(graphs={}, lock=threading.Lock(), ended=threading.Event())


def _ranks(nodes, edges):
    """Longest-path layer per node (simple layered DAG layout)."""
    rank = {n["id"]: 0 for n in nodes}
    for _ in range(len(nodes)):
        changed = False
        for e in edges:
            if rank[e["dst"]] < rank[e["src"]] + 1:
                rank[e["dst"]] = rank[e["src"]] + 1
                changed = True
        if not changed:
            break
    return rank


def graph_svg(diagram, stats_by_node=None):
    """Inline SVG of the graph: boxes per operator node, arrows per edge
    (reference generateSVGDiagram via graphviz)."""
    nodes, edges = diagram.get("nodes", []), diagram.get("edges", [])
    if not nodes:
        return "<svg width='10' height='10'></svg>"
    rank = _ranks(nodes, edges)
    cols = {}
    pos = {}
    W, H, XGAP, YGAP = 190, 54, 240, 80
    for n in nodes:
        r = rank[n["id"]]
        row = cols.get(r, 0)
        cols[r] = row + 1
        pos[n["id"]] = (40 + r * XGAP, 30 + row * YGAP)
    width = 80 + (max(rank.values()) + 1) * XGAP
    height = 60 + max(cols.values()) * YGAP
    out = [f"<svg xmlns='http://www.w3.org/2000/svg' width='{width}' "
           f"height='{height}' font-family='monospace' font-size='11'>",
           "<defs><marker id='arr' markerWidth='8' markerHeight='8' refX='7' "
           "refY='3' orient='auto'><path d='M0,0 L7,3 L0,6 z' fill='#555'/>"
           "</marker></defs>"]
    route = {0: "", 1: "keyby", 2: "broadcast", 3: "rebalance"}
    for e in edges:
        x1, y1 = pos[e["src"]]
        x2, y2 = pos[e["dst"]]
        out.append(f"<line x1='{x1 + W}' y1='{y1 + H // 2}' x2='{x2}' "
                   f"y2='{y2 + H // 2}' stroke='#555' marker-end='url(#arr)'/>")
        lbl = route.get(int(e.get("routing", 0)), "")
        if lbl:
            out.append(f"<text x='{(x1 + W + x2) // 2 - 24}' "
                       f"y='{(y1 + y2) // 2 + H // 2 - 4}' fill='#777'>{lbl}</text>")
    for n in nodes:
        x, y = pos[n["id"]]
        fill = "#dfe9ff" if n.get("gpu") else "#eef6e8"
        out.append(f"<rect x='{x}' y='{y}' width='{W}' height='{H}' rx='6' "
                   f"fill='{fill}' stroke='#333'/>")
        label = html.escape(str(n.get("label", n["id"]))[:26])
        dev = f" GPU{n.get('device')}" if n.get("gpu") else ""
        out.append(f"<text x='{x + 8}' y='{y + 18}'>{label}</text>")
        out.append(f"<text x='{x + 8}' y='{y + 34}' fill='#666'>"
                   f"x{n.get('parallelism', 1)}{dev}</text>")
        if stats_by_node and n["id"] in stats_by_node:
            s = stats_by_node[n["id"]]
            out.append(f"<text x='{x + 8}' y='{y + 48}' fill='#06c'>"
                       f"{s} t/s</text>")
    out.append("</svg>")
    return "".join(out)


def render_html():
    with STATE["lock"]:
        graphs = {k: dict(v) for k, v in STATE["graphs"].items()}
    parts = ["<!doctype html><html><head><meta charset='utf-8'>",
             "<meta http-equiv='refresh' content='1'>",
             "<title>windflow_amd dashboard</title>",
             "<style>body{font-family:monospace;margin:20px} "
             "table{border-collapse:collapse} td,th{border:1px solid #999;"
             "padding:3px 8px;text-align:right} th{background:#eee} "
             ".l{text-align:left}</style></head><body>",
             "<h2>windflow_amd dashboard</h2>"]
    if not graphs:
        parts.append("<p>no graphs connected yet</p>")
    for name, g in graphs.items():
        rep = g.get("report") or {}
        parts.append(f"<h3>graph '{html.escape(name)}'"
                     f"{' (ended)' if g.get('ended') else ''}</h3>")
        parts.append(f"<p>rss={rep.get('rss_kb', 0)} kB &nbsp; "
                     f"dropped={rep.get('dropped', 0)} &nbsp; "
                     f"t={time.strftime('%H:%M:%S', time.localtime(rep.get('ts', 0)))}</p>")
        if g.get("diagram"):
            parts.append(graph_svg(g["diagram"]))
        rows = rep.get("replicas", [])
        if rows:
            agg = {}
            for r in rows:
                d = agg.setdefault(r["name"], dict(reps=0, tin=0, tout=0,
                                                   svc=0.0, kern=0))
                d["reps"] += 1
                d["tin"] += r.get("tuples_in", 0)
                d["tout"] += r.get("tuples_out", 0)
                d["svc"] = max(d["svc"], r.get("svc_us_ewma", 0.0))
                d["kern"] += r.get("num_kernels", 0)
            parts.append("<table><tr><th class='l'>operator</th><th>replicas"
                         "</th><th>tuples in</th><th>tuples out</th>"
                         "<th>svc µs (ewma)</th><th>kernels</th></tr>")
            for nm, d in agg.items():
                parts.append(f"<tr><td class='l'>{html.escape(nm)}</td>"
                             f"<td>{d['reps']}</td><td>{d['tin']:,}</td>"
                             f"<td>{d['tout']:,}</td><td>{d['svc']:.1f}</td>"
                             f"<td>{d['kern']:,}</td></tr>")
            parts.append("</table>")
    parts.append("</body></html>")
    return "".join(parts)


class Ui(BaseHTTPRequestHandler):
    def do_GET(self):
        if self.path.startswith("/data"):
            body = json.dumps({k: v.get("report")
                               for k, v in STATE["graphs"].items()}).encode()
            ctype = "application/json"
        else:
            body = render_html().encode()
            ctype = "text/html; charset=utf-8"
        self.send_response(200)
        self.send_header("Content-Type", ctype)
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)

    def log_message(self, *a):
        pass


def handle(conn, addr, save_dir):
    graph = "?"
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
            f = json.loads(buf)
            graph = f.get("graph", graph)
            with STATE["lock"]:
                g = STATE["graphs"].setdefault(graph, {})
                if f["type"] == "diagram":
                    g["diagram"] = f
                elif f["type"] == "report":
                    g["report"] = f
            if f["type"] == "diagram":
                print(f"[{addr[0]}] graph '{graph}' connected")
                if save_dir:
                    with open(os.path.join(save_dir, f"{graph}.dot"), "w") as fh:
                        fh.write(f["dot"])
            elif save_dir:
                with open(os.path.join(save_dir, f"{graph}.json"), "w") as fh:
                    json.dump(f, fh, indent=1)
    finally:
        conn.close()
        with STATE["lock"]:
            STATE["graphs"].setdefault(graph, {})["ended"] = True
        STATE["ended"].set()
        print(f"[{addr[0]}] graph '{graph}' disconnected")


def serve(port=20207, http_port=20208, save=None, once=False):
    if save:
        os.makedirs(save, exist_ok=True)
    httpd = ThreadingHTTPServer(("0.0.0.0", http_port), Ui)
    threading.Thread(target=httpd.serve_forever, daemon=True).start()
    srv = socket.socket()
    srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    srv.bind(("0.0.0.0", port))
    srv.listen(8)
    srv.settimeout(0.5)
    print(f"dashboard: TCP ingest :{port}, HTTP UI http://localhost:{http_port}")
    while True:
        try:
            conn, addr = srv.accept()
        except socket.timeout:
            if once and STATE["ended"].is_set():
                return
            continue
        threading.Thread(target=handle, args=(conn, addr, save),
                         daemon=True).start()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--port", type=int,
                    default=int(os.environ.get("WF_DASHBOARD_PORT", "20207")))
    ap.add_argument("--http", type=int,
                    default=int(os.environ.get("WF_DASHBOARD_HTTP", "20208")))
    ap.add_argument("--save", default=None)
    ap.add_argument("--once", action="store_true")
    args = ap.parse_args()
    serve(args.port, args.http, args.save, args.once)


if __name__ == "__main__":
    main()
