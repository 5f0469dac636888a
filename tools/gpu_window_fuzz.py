#!/usr/bin/env python3
"""Randomized differential sweep of gpu_ffat windows (CB ring / CB tree /
TB) against the brute-force oracle.  Run on an MI355X box:
    python tools/gpu_window_fuzz.py [n_cases]
"""
import random
import sys
from collections import defaultdict

import numpy as np

sys.path.insert(0, ".")
import windflow_amd as wf                                    # noqa: E402
from windflow_amd import native_gpu                          # noqa: E402
from windflow_amd.builders_gpu import (Source_GPU_Builder,   # noqa: E402
                                       Ffat_Windows_GPU_Builder)
from windflow_amd.synth import gen_batch                     # noqa: E402

COMBS = {"sum": native_gpu.COMB_SUM, "min": native_gpu.COMB_MIN,
         "max": native_gpu.COMB_MAX, "count": native_gpu.COMB_COUNT}


def run_engine(n, n_keys, b, win, slide, comb, tree, tb, vdt):
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=vdt))
           .withOutputSchema([vdt]).withOutputBatchSize(b).build())
    ff = (Ffat_Windows_GPU_Builder(
        native_gpu.gpu_ffat_windows(COMBS[comb], 0, win, slide,
                                    max_keys=2 * n_keys, use_tree=tree, tb=tb,
                                    pend_ring_log2=12))
        .withOutputSchema([2]).withOutputBatchSize(4 * b).build())
    rows = []

    def pysink(cols):
        rows.append((cols['key'].copy(), cols['c0'].copy()))

    g = wf.PipeGraph("fz")
    p = g.add_source(src)
    p.chain(ff)
    snk = wf.Sink_Builder(pysink).withParallelism(1).build()
    snk.out_schema = [2]
    p.add_sink(snk)
    g.run()
    got = defaultdict(list)
    for k_arr, v_arr in rows:
        for k, v in zip(k_arr.tolist(), v_arr.tolist()):
            got[k].append(round(float(v), 2))
    return {k: sorted(v) for k, v in got.items()}


def oracle(n, n_keys, b, win, slide, comb, tb, vdt):
    ts, key, val = gen_batch(n, 0, 42, n_keys, vdt)
    per = defaultdict(list)
    for t, k, v in zip(ts.tolist(), key.tolist(), val.tolist()):
        per[k].append((t, np.float32(v)))
    f = {"sum": lambda a: float(np.sum(np.array(a, np.float32), dtype=np.float64)),
         "min": lambda a: float(min(a)), "max": lambda a: float(max(a)),
         "count": lambda a: float(len(a))}[comb]
    exp = defaultdict(list)
    for k, rows in per.items():
        if not tb:
            vals = [v for _, v in rows]
            w = 0
            while (w + 1) * slide + (win - slide) <= len(vals):  # full windows only
                exp[k].append(round(f(vals[w * slide: w * slide + win]), 2))
                w += 1
        else:
            pane = int(np.gcd(win, slide))
            tss = [t for t, _ in rows]
            t0, tmax = min(tss), max(tss)
            w0 = max(0, -(-(t0 - win + 1) // slide))
            w = w0
            while (w * slide + win - 1) // pane <= tmax // pane:
                seg = [v for t, v in rows if w * slide <= t < w * slide + win]
                exp[k].append(round(f(seg) if seg else 0.0, 2))
                w += 1
    return {k: sorted(v) for k, v in exp.items()}


def close(a, b):
    if set(a) != set(b):
        return False
    for k in a:
        if len(a[k]) != len(b[k]):
            return False
        for x, y in zip(a[k], b[k]):
            if abs(x - y) > 2e-3 * max(1.0, abs(y)):
                return False
    return True


def main():
    n_cases = int(sys.argv[1]) if len(sys.argv) > 1 else 20
    fails = []
    for case in range(n_cases):
        rng = random.Random(4242 + case)
        tb = rng.random() < 0.4
        comb = rng.choice(["sum", "min", "max", "count"])
        tree = (not tb) and rng.random() < 0.3
        slide = rng.choice([10, 50, 100, 250])
        win = slide * rng.randint(1, 6)
        n_keys = rng.choice([16, 101, 1024, 8192])
        n = rng.choice([200_000, 1_000_000])
        b = rng.choice([50_000, 250_000])
        vdt = rng.choice([2, 5]) if comb == "sum" else 2
        cfg = (case, comb, "tb" if tb else ("tree" if tree else "cb"),
               win, slide, n_keys, n, b, vdt)
        try:
            got = run_engine(n, n_keys, b, win, slide, comb, tree, tb, vdt)
            exp = oracle(n, n_keys, b, win, slide, comb, tb, vdt)
            if not close(got, exp):
                nd = sum(1 for k in exp
                         if got.get(k, []) != exp[k])
                fails.append(("MISMATCH", nd) + cfg)
        except Exception as e:
            fails.append(("ERROR", str(e)[:120]) + cfg)
        print(f"case {case} {'FAIL' if fails and fails[-1][2+0]==case else 'ok'} {cfg}",
              flush=True)
    print("FAILURES:", len(fails))
    for f in fails[:10]:
        print(f)
    sys.exit(1 if fails else 0)


if __name__ == "__main__":
    main()
