"""Step-by-step GPU validation with prints — isolates kernel faults."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import windflow_amd as wf
from windflow_amd import native, native_gpu
from windflow_amd.builders_gpu import *
from windflow_amd.synth import gen_batch

def run(name, ops, sink_schema, n, expect=None):
    rows = dict(s=0.0, n=0)
    def pysink(cols):
        rows['s'] += float(np.asarray(cols['c0'], dtype=np.float64).sum())
        rows['n'] += len(cols['c0'])
    g = wf.PipeGraph(name)
    mp = g.add_source(ops[0])
    for op in ops[1:]:
        mp.chain(op)
    snk = wf.Sink_Builder(pysink).withParallelism(1).build()
    snk.out_schema = sink_schema
    mp.add_sink(snk)
    g.run()
    print(f"{name}: n={rows['n']} sum={rows['s']:.3f} expect={expect}", flush=True)
    return rows

N, B, K = 100_000, 30_000, 97
print("step 1: generator i64", flush=True)
src = (Source_GPU_Builder(native_gpu.gpu_source(N, K, B, vdt=0))
       .withOutputSchema([0]).withOutputBatchSize(B).build())
ts, key, val = gen_batch(N, 0, 42, K, 0)
run("gen", [src], [0], N, expect=(N, int(val.sum())))

print("step 2: +map", flush=True)
src = (Source_GPU_Builder(native_gpu.gpu_source(N, K, B, vdt=0))
       .withOutputSchema([0]).withOutputBatchSize(B).build())
m = (Map_GPU_Builder(native_gpu.gpu_affine_map(0, 3, 1, dtype=0))
     .withOutputSchema([0]).withOutputBatchSize(B).build())
run("map", [src, m], [0], N, expect=int((val*3+1).sum()))

print("step 3: +filter", flush=True)
src = (Source_GPU_Builder(native_gpu.gpu_source(N, K, B, vdt=0))
       .withOutputSchema([0]).withOutputBatchSize(B).build())
m = (Map_GPU_Builder(native_gpu.gpu_affine_map(0, 3, 1, dtype=0))
     .withOutputSchema([0]).withOutputBatchSize(B).build())
f = (Filter_GPU_Builder(native_gpu.gpu_mod_filter(0, 5, 0))
     .withOutputSchema([0]).withOutputBatchSize(B).build())
v = val*3+1; keep = v[v % 5 != 0]
run("filter", [src, m, f], [0], N, expect=(len(keep), int(keep.sum())))

print("step 4: reduce", flush=True)
src = (Source_GPU_Builder(native_gpu.gpu_source(N, 64, B, vdt=2))
       .withOutputSchema([2]).withOutputBatchSize(B).build())
r = (Reduce_GPU_Builder(native_gpu.gpu_keyed_reduce(native_gpu.COMB_SUM, 0, 256))
     .withOutputSchema([2]).withOutputBatchSize(B).build())
_, _, valf = gen_batch(N, 0, 42, 64, 2)
run("reduce", [src, r], [2], N, expect=float(valf.astype(np.float64).sum()))

print("step 5: ffat ring", flush=True)
src = (Source_GPU_Builder(native_gpu.gpu_source(N, 101, 17_000, vdt=2))
       .withOutputSchema([2]).withOutputBatchSize(17_000).build())
ff = (Ffat_Windows_GPU_Builder(native_gpu.gpu_ffat_windows(native_gpu.COMB_SUM, 0, 40, 10, max_keys=1024))
      .withOutputSchema([2]).withOutputBatchSize(17_000).build())
run("ffat_ring", [src, ff], [2], N)

print("step 6: ffat tree", flush=True)
src = (Source_GPU_Builder(native_gpu.gpu_source(N, 101, 17_000, vdt=2))
       .withOutputSchema([2]).withOutputBatchSize(17_000).build())
ff = (Ffat_Windows_GPU_Builder(native_gpu.gpu_ffat_windows(native_gpu.COMB_SUM, 0, 40, 10, max_keys=1024, use_tree=True))
      .withOutputSchema([2]).withOutputBatchSize(17_000).build())
run("ffat_tree", [src, ff], [2], N)

print("step 7: torch coexistence", flush=True)
import torch
print("torch cuda:", torch.cuda.is_available(), flush=True)
src = (Source_GPU_Builder(native_gpu.gpu_source(N, K, B, vdt=0))
       .withOutputSchema([0]).withOutputBatchSize(B).build())
run("gen_after_torch", [src], [0], N, expect=(N, int(val.sum())))
print("ALL DEBUG STEPS DONE", flush=True)
