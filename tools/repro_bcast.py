import sys
sys.path.insert(0, ".")
import windflow_amd as wf
from windflow_amd import native_gpu
from windflow_amd.builders_gpu import (Source_GPU_Builder, Map_GPU_Builder,
                                       Reduce_GPU_Builder,
                                       Ffat_Windows_GPU_Builder,
                                       Sink_GPU_Builder)

variant = sys.argv[1]
n, n_keys, b, win, slide = 200_000, 101, 50_000, 500, 100
src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=5))
       .withOutputSchema([5]).withOutputBatchSize(b).build())
g = wf.PipeGraph("r")
mp = g.add_source(src)
par = 1 if variant.endswith("p1") else 2
if variant.startswith("ffat"):
    logic = native_gpu.gpu_ffat_windows(
        native_gpu.COMB_SUM, 0, win, slide, max_keys=1024,
        use_tree=(variant == "ffat_tree"), tb=(variant == "ffat_tb"),
        pend_ring_log2=10)
    op = (Ffat_Windows_GPU_Builder(logic).withOutputSchema([2])
          .withOutputBatchSize(2 * b).withParallelism(par).build())
elif variant == "keyedmap":
    op = (Map_GPU_Builder(native_gpu.gpu_keyed_running_sum(0, max_keys=1024))
          .withOutputSchema([5]).withOutputBatchSize(b)
          .withParallelism(2).build())
elif variant == "reduce":
    op = (Reduce_GPU_Builder(native_gpu.gpu_keyed_reduce(
        native_gpu.COMB_SUM, 0, 1024)).withOutputSchema([2])
        .withOutputBatchSize(b).withParallelism(2).build())
snk = Sink_GPU_Builder(native_gpu.gpu_count_sink()).withParallelism(par).build()
mp.add(op)
mp.add(snk)
g.run()
print("OK", variant, g.sink_count(snk), flush=True)
