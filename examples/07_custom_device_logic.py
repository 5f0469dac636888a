#!/usr/bin/env python3
"""Arbitrary USER device logic via hiprtc (round 2) — the MI355X
counterpart of the reference's `__device__` lambda surface:

  1. custom map/filter expressions,
  2. a custom sliding-window fold (geometric mean via (sum-of-logs, count)
     accumulator — not in any native catalog),
  3. a fused keyed AVG reduce,
  4. a per-tuple device split expression.

Needs an MI355X: run with gpurun / on a ROCm box.
    python examples/07_custom_device_logic.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import windflow_amd as wf                                    # noqa: E402
from windflow_amd import native, native_gpu                  # noqa: E402
from windflow_amd.builders_gpu import (Source_GPU_Builder,   # noqa: E402
                                       Map_GPU_Builder, Filter_GPU_Builder,
                                       Ffat_Windows_GPU_Builder,
                                       Reduce_GPU_Builder, Sink_GPU_Builder)

N, KEYS, BATCH = 4_000_000, 1024, 1_000_000

g = wf.PipeGraph("custom_device_logic")
src = (Source_GPU_Builder(native_gpu.gpu_source(N, KEYS, BATCH, vdt=2))
       .withOutputSchema([2]).withOutputBatchSize(BATCH).build())
mp = g.add_source(src)

# 1. custom expression map + predicate (compiled by hiprtc at operator init)
mp.chain(Map_GPU_Builder(native_gpu.gpu_jit_map("v * 0.5f + 0.25f", 0))
         .withOutputSchema([2]).withOutputBatchSize(BATCH).build())
mp.chain(Filter_GPU_Builder(native_gpu.gpu_jit_filter("v > 0.3f && (key & 3) != 0", 0))
         .withOutputSchema([2]).withOutputBatchSize(BATCH).build())

# 2. custom window fold: geometric mean over a 1000/100 sliding window.
#    lift -> (log v, 1); comb -> fieldwise + (invertible running window);
#    finalize -> exp(sum/count), 0 for empty gap windows.
mp.chain(Ffat_Windows_GPU_Builder(
    lift="__logf(v0);1.0f",
    comb="a0+b0;a1+b1",
    finalize="(f1 > 0.0f) ? __expf(f0 / f1) : 0.0f",
    identity=(0.0, 0.0), invertible=True, max_keys=2 * KEYS)
    .withCBWindows(1000, 100)
    .withOutputSchema([2]).withOutputBatchSize(BATCH).build())

snk = Sink_GPU_Builder(native_gpu.gpu_count_sink()).build()
mp.chain_sink(snk)
g.run()
print(f"geomean windows fired: {g.sink_count(snk)} in {g.elapsed():.3f}s")

# 3. fused keyed AVG reduce (one (sum,count) pass, no two-op composition)
g2 = wf.PipeGraph("fused_avg")
src2 = (Source_GPU_Builder(native_gpu.gpu_source(N, KEYS, BATCH, vdt=2))
        .withOutputSchema([2]).withOutputBatchSize(BATCH).build())
mp2 = g2.add_source(src2)
mp2.chain(Reduce_GPU_Builder(native_gpu.gpu_avg_reduce(0, max_keys=2 * KEYS))
          .withOutputSchema([2]).withOutputBatchSize(BATCH).build())
snk2 = Sink_GPU_Builder(native_gpu.gpu_count_sink()).build()
mp2.chain_sink(snk2)
g2.run()
print(f"per-batch keyed averages: {g2.sink_count(snk2)} rows")

# 4. per-tuple device split: route rows by a value expression
g3 = wf.PipeGraph("device_split")
src3 = (Source_GPU_Builder(native_gpu.gpu_source(N, KEYS, BATCH, vdt=2))
        .withOutputSchema([2]).withOutputBatchSize(BATCH).build())
mp3 = g3.add_source(src3)
mp3.split_gpu(2, expr="v0 < 0.5f ? 0 : 1")
sinks = []
for br in range(2):
    bmp = mp3.select(br)
    s = Sink_GPU_Builder(native_gpu.gpu_count_sink()).build()
    bmp.chain_sink(s)
    sinks.append(s)
g3.run()
print(f"split: low={g3.sink_count(sinks[0])} high={g3.sink_count(sinks[1])}")

# 5. round-2 extras on the same fold surface:
#    - acc="f64": double-precision accumulator fields + column loads
#      (large-magnitude sums; output columns stay F32)
#    - withDenseKeys(): user-asserted integer keys < max_keys skip the
#      hash probe (flagship: 18.1 -> 22 B tuples/s)
g4 = wf.PipeGraph("f64_dense")
src4 = (Source_GPU_Builder(native_gpu.gpu_source(N, KEYS, BATCH, vdt=2))
        .withOutputSchema([2]).withOutputBatchSize(BATCH).build())
mp4 = g4.add_source(src4)
mp4.chain(Ffat_Windows_GPU_Builder(lift="v0;1.0", comb="a0+b0;a1+b1",
                                   finalize="(f1 > 0.0) ? (f0 / f1) : 0.0",
                                   identity=(0, 0), invertible=True,
                                   max_keys=2 * KEYS, acc="f64")
          .withCBWindows(1000, 100).withDenseKeys()
          .withOutputSchema([2]).withOutputBatchSize(2 * BATCH).build())
snk4 = Sink_GPU_Builder(native_gpu.gpu_count_sink()).build()
mp4.chain_sink(snk4)
g4.run()
print(f"f64-accumulator dense-key AVG windows: {g4.sink_count(snk4)}")
