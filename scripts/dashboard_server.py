#!/usr/bin/env python3
"""Minimal dashboard receiver (counterpart of the reference's Java Spring
server + React client, dashboard/Server): accepts MonitoringThread
connections (length-prefixed JSON frames, see windflow_amd/monitoring.py),
prints a live per-operator table and stores the last report per graph.

Usage: python scripts/dashboard_server.py [--port 20207] [--save DIR]
"""
import argparse
import json
import os
import socket
import struct
import threading


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
            if f["type"] == "diagram":
                print(f"[{addr[0]}] graph '{graph}' connected")
                if save_dir:
                    with open(os.path.join(save_dir, f"{graph}.dot"), "w") as fh:
                        fh.write(f["dot"])
            elif f["type"] == "report":
                per = {}
                for r in f.get("replicas", []):
                    d = per.setdefault(r["name"], [0, 0, 0.0])
                    d[0] += r["tuples_in"]
                    d[1] += r["tuples_out"]
                    d[2] = max(d[2], r["svc_us_ewma"])
                row = " | ".join(f"{n}: in={v[0]} out={v[1]} svc={v[2]:.1f}us"
                                 for n, v in per.items())
                print(f"[{graph}] rss={f['rss_kb']}kB dropped={f['dropped']} {row}")
                if save_dir:
                    with open(os.path.join(save_dir, f"{graph}.json"), "w") as fh:
                        json.dump(f, fh, indent=1)
    finally:
        conn.close()
        print(f"[{addr[0]}] graph '{graph}' disconnected")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--port", type=int,
                    default=int(os.environ.get("WF_DASHBOARD_PORT", "20207")))
    ap.add_argument("--save", default=None)
    args = ap.parse_args()
    if args.save:
        os.makedirs(args.save, exist_ok=True)
    srv = socket.socket()
    srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    srv.bind(("0.0.0.0", args.port))
    srv.listen(8)
    print(f"dashboard listening on :{args.port}")
    while True:
        conn, addr = srv.accept()
        threading.Thread(target=handle, args=(conn, addr, args.save),
                         daemon=True).start()


if __name__ == "__main__":
    main()
