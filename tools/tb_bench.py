#!/usr/bin/env python3
"""Quick TB (event-time) GPU window throughput measurement — the flagship
bench is CB; this documents the TB pane machine's rate on the same shape
(win=1000/slide=100 in ts units, monotonic source ts, watermark-driven
pane completion)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import windflow_amd as wf                                   # noqa: E402
from windflow_amd import native_gpu                          # noqa: E402
from windflow_amd.builders_gpu import (Source_GPU_Builder,   # noqa: E402
                                       Ffat_Windows_GPU_Builder,
                                       Sink_GPU_Builder)


def run(n_steps=100, batch=16_777_216, keys=8192, dense=True):
    n = n_steps * batch
    # global ts advances 1/tuple, so each key sees ts every ~keys units:
    # scale the window extent so a key-window holds ~1000 tuples (the CB
    # flagship shape) instead of firing the empty inter-tuple window grid
    win, slide = 1000 * keys, 100 * keys
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, keys, batch, vdt=5))
           .withOutputSchema([5]).withOutputBatchSize(batch).build())
    ff = (Ffat_Windows_GPU_Builder(
        native_gpu.gpu_ffat_windows(native_gpu.COMB_SUM, 0, win, slide,
                                    max_keys=keys, tb=True, lateness=0,
                                    pend_ring_log2=18, dense_keys=dense))
          .withOutputSchema([2]).withOutputBatchSize(2 * batch).build())
    snk = Sink_GPU_Builder(native_gpu.gpu_count_sink()).build()
    g = wf.PipeGraph("tb_bench")
    p = g.add_source(src)
    p.chain(ff)
    p.chain_sink(snk)
    t0 = time.time()
    g.run()
    dt = time.time() - t0
    print(f"TB windows: {n:,} tuples in {dt:.2f}s = "
          f"{n / dt / 1e9:.2f} B tuples/s (windows fired: "
          f"{g.sink_count(snk):,})")


if __name__ == "__main__":
    import argparse
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=100)
    ap.add_argument("--batch", type=int, default=16_777_216)
    ap.add_argument("--keys", type=int, default=8192)
    a = ap.parse_args()
    run(a.steps, a.batch, a.keys)
