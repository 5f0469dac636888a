#!/usr/bin/env python3
"""Differential sweep of gpu_ffat windows vs a brute-force oracle, split so
GPU time is engine-only:
    (GPU box)  python tools/gpu_window_fuzz.py engine   # dumps gpurun_out/wfz_*.npz
    (anywhere) python tools/gpu_window_fuzz.py check    # recomputes oracles, compares
Configs are fixed (not drawn at runtime) so both halves agree.
"""
import sys
from collections import defaultdict

import numpy as np

sys.path.insert(0, ".")

# (comb, form, win, slide, n_keys, n, batch, vdt)
# combs: sum/min/max/count = native catalog; avg/jmin/jminmax = hiprtc JIT
# user folds (gpu_jit_ffat — round-2 arbitrary lift/comb surface)
CASES = [
    ("sum",   "cb",   500, 100, 101,  1_000_000, 250_000, 5),
    ("min",   "cb",   300, 300, 16,     400_000, 100_000, 2),
    ("max",   "tree", 750, 250, 101,  1_000_000, 250_000, 2),
    ("count", "cb",   400, 200, 1024, 1_000_000, 250_000, 2),
    ("count", "tb",   600, 200, 16,     400_000, 100_000, 2),
    ("sum",   "tb",  1000, 500, 101,  1_000_000, 250_000, 2),
    ("max",   "tb",   900, 300, 16,     400_000, 100_000, 2),
    ("sum",   "tree", 800, 400, 1024, 1_000_000, 250_000, 5),
    ("avg",   "cb",   500, 100, 101,  1_000_000, 250_000, 2),
    ("avg",   "tb",   600, 300, 64,     400_000, 100_000, 2),
    ("jmin",  "cb",   300, 60,  101,    400_000, 100_000, 2),
    ("avg",   "cb",    24, 8,   64,     400_000, 100_000, 2),  # thread kernel
    # round-2 dense-key variants (slot = key; same oracles)
    ("sum",   "cb",   500, 100, 101,  1_000_000, 250_000, 5, True),
    ("avg",   "tb",   600, 300, 64,     400_000, 100_000, 2, True),
    ("count", "tb",   600, 200, 128,    400_000, 100_000, 5, True),
]


def engine():
    import windflow_amd as wf
    from windflow_amd import native_gpu
    from windflow_amd.builders_gpu import (Source_GPU_Builder,
                                           Ffat_Windows_GPU_Builder)
    COMBS = {"sum": native_gpu.COMB_SUM, "min": native_gpu.COMB_MIN,
             "max": native_gpu.COMB_MAX, "count": native_gpu.COMB_COUNT}
    for ci, case in enumerate(CASES):
        comb, form, win, slide, n_keys, n, b, vdt = case[:8]
        dense = bool(case[8]) if len(case) > 8 else False
        src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=vdt))
               .withOutputSchema([vdt]).withOutputBatchSize(b).build())
        if comb == "avg":
            logic = native_gpu.gpu_avg_ffat_windows(
                win, slide, col=0, max_keys=2 * n_keys, tb=form == "tb",
                pend_ring_log2=13, dense_keys=dense)
        elif comb == "jmin":
            logic = native_gpu.gpu_jit_ffat_windows(
                win, slide, lift="v0", comb="fminf(a0, b0)", finalize="f0",
                identity=(float("inf"),), max_keys=2 * n_keys,
                tb=form == "tb", pend_ring_log2=13, dense_keys=dense)
        else:
            logic = native_gpu.gpu_ffat_windows(COMBS[comb], 0, win, slide,
                                                max_keys=2 * n_keys,
                                                use_tree=form == "tree",
                                                tb=form == "tb",
                                                pend_ring_log2=13,
                                                dense_keys=dense)
        ff = (Ffat_Windows_GPU_Builder(logic)
            .withOutputSchema([2]).withOutputBatchSize(4 * b).build())
        keys, vals = [], []

        def pysink(cols):
            keys.append(cols['key'].copy())
            vals.append(cols['c0'].copy())

        g = wf.PipeGraph(f"fz{ci}")
        p = g.add_source(src)
        p.chain(ff)
        snk = wf.Sink_Builder(pysink).withParallelism(1).build()
        snk.out_schema = [2]
        p.add_sink(snk)
        g.run()
        np.savez(f"gpurun_out/wfz_{ci}.npz",
                 key=np.concatenate(keys) if keys else np.zeros(0, np.uint64),
                 val=np.concatenate(vals) if vals else np.zeros(0, np.float32))
        print(f"case {ci} dumped {sum(len(k) for k in keys)} windows", flush=True)


def check():
    from windflow_amd.synth import gen_batch
    fails = 0
    for ci, case in enumerate(CASES):
        comb, form, win, slide, n_keys, n, b, vdt = case[:8]
        d = np.load(f"gpurun_out/wfz_{ci}.npz")
        got = defaultdict(list)
        for k, v in zip(d['key'].tolist(), d['val'].tolist()):
            got[k].append(float(v))
        ts, key, val = gen_batch(n, 0, 42, n_keys, vdt)
        if vdt == 5:  # decode raw bf16 bit patterns
            val = (val.astype(np.uint32) << 16).view(np.float32)
        per = defaultdict(list)
        for t, k, v in zip(ts.tolist(), key.tolist(), val.tolist()):
            per[k].append((t, np.float32(v)))
        f = {"sum": lambda a: float(np.sum(np.asarray(a, np.float32),
                                           dtype=np.float64)),
             "min": lambda a: float(min(a)), "max": lambda a: float(max(a)),
             "count": lambda a: float(len(a)),
             "avg": lambda a: float(np.mean(np.asarray(a, np.float64))),
             "jmin": lambda a: float(min(a))}[comb]
        exp = defaultdict(list)
        for k, rows in per.items():
            if form != "tb":
                v2 = [v for _, v in rows]
                w = 0
                while w * slide + win <= len(v2):
                    exp[k].append(f(v2[w * slide: w * slide + win]))
                    w += 1
                while w * slide < len(v2):   # EOS partial flush
                    exp[k].append(f(v2[w * slide:]))
                    w += 1
            else:
                pane = int(np.gcd(win, slide))
                tss = [t for t, _ in rows]
                t0, tmax = min(tss), max(tss)
                w = max(0, -(-(t0 - win + 1) // slide))
                while (w * slide + win - 1) // pane <= tmax // pane:
                    seg = [v for t, v in rows if w * slide <= t < w * slide + win]
                    exp[k].append(f(seg) if seg else 0.0)
                    w += 1
                while (w * slide) // pane <= tmax // pane:  # EOS partial flush
                    seg = [v for t, v in rows if w * slide <= t < w * slide + win]
                    exp[k].append(f(seg) if seg else 0.0)
                    w += 1
        ok = set(got) == set(exp)
        if ok:
            for k in exp:
                a, bb = sorted(got[k]), sorted(exp[k])
                if len(a) != len(bb) or any(
                        abs(x - y) > 2e-3 * max(1.0, abs(y))
                        for x, y in zip(a, bb)):
                    ok = False
                    break
        print(f"case {ci} {CASES[ci][:2]} {'OK' if ok else 'MISMATCH'}")
        fails += not ok
    print("FAILURES:", fails)
    sys.exit(1 if fails else 0)


if __name__ == "__main__":
    (engine if sys.argv[1:] == ["engine"] else check)()
