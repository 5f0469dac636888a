"""Debug sweep: FFAT CB fire counts vs oracle across key cardinalities."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from collections import Counter

import windflow_amd as wf
from windflow_amd import native_gpu
from windflow_amd.builders_gpu import (Source_GPU_Builder,
                                       Ffat_Windows_GPU_Builder,
                                       Sink_GPU_Builder)
from windflow_amd.synth import gen_batch


def run(n, n_keys, b, win, slide, max_keys):
    src = (Source_GPU_Builder(native_gpu.gpu_source(n, n_keys, b, vdt=2))
           .withOutputSchema([2]).withOutputBatchSize(b).build())
    ff = (Ffat_Windows_GPU_Builder(
        native_gpu.gpu_ffat_windows(native_gpu.COMB_SUM, 0, win, slide,
                                    max_keys=max_keys))
          .withOutputSchema([2]).withOutputBatchSize(2 * b).build())
    g = wf.PipeGraph("dbg")
    p = g.add_source(src)
    p.chain(ff)
    snk = Sink_GPU_Builder(native_gpu.gpu_count_sink()).build()
    p.chain_sink(snk)
    g.run()
    got = g.sink_count(snk)
    _, key, _ = gen_batch(n, 0, 42, n_keys, 2)
    per = Counter(key.tolist())
    P, S = win // __import__('math').gcd(win, slide), slide // __import__('math').gcd(win, slide)
    exp = sum((c - win) // slide + 1 for c in per.values() if c >= win)
    print(f"keys={n_keys:>8} n={n:>8} b={b:>8} win={win} slide={slide} "
          f"got={got} exp={exp} {'OK' if got == exp else 'MISMATCH'}")


if __name__ == "__main__":
    run(200_000, 101, 50_000, 8, 2, 1024)          # known-good shape, tiny win
    run(200_000, 8192, 50_000, 8, 2, 16384)        # 8K keys, tiny win
    run(200_000, 50_000, 50_000, 8, 2, 100_000)    # 50K keys
    run(200_000, 50_000, 200_000, 8, 2, 100_000)   # single batch
    run(1_000_000, 250_000, 250_000, 8, 2, 500_000)
    run(4_000_000, 1_000_000, 1_000_000, 8, 2, 2_000_000)
    run(4_000_000, 1_000_000, 1_000_000, 1000, 100, 2_000_000)
