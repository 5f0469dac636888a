import sys
sys.path.insert(0, ".")
import windflow_amd as wf
from windflow_amd import native_gpu
from windflow_amd.builders_gpu import (Source_GPU_Builder,
                                       Ffat_Windows_GPU_Builder,
                                       Sink_GPU_Builder)

variant = sys.argv[1]
n, n_keys, b, win, slide = 200_000, 101, 50_000, 500, 100
if variant == "small":
    n, b = 20_000, 5_000

src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=5))
       .withOutputSchema([5]).withOutputBatchSize(b).build())
g = wf.PipeGraph("r")
mp = g.add_source(src)
if variant == "sink_only":
    snk = (Sink_GPU_Builder(native_gpu.gpu_count_sink())
           .withParallelism(2).withBroadcast().build())
    mp.add_sink(snk)
else:
    ff = (Ffat_Windows_GPU_Builder(
        native_gpu.gpu_ffat_windows(native_gpu.COMB_SUM, 0, win, slide,
                                    max_keys=1024))
        .withOutputSchema([2]).withOutputBatchSize(2 * b)
        .withParallelism(2).build())
    if variant != "forward":
        ff.broadcast_input = True
    snk = Sink_GPU_Builder(native_gpu.gpu_count_sink()).withParallelism(2).build()
    mp.add(ff)
    mp.add(snk)
g.run()
print("OK", variant, g.sink_count(snk), flush=True)
