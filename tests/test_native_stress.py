"""Runs the native C++ engine stress harness (csrc/tests/engine_stress.cpp).
Built plain here; under ThreadSanitizer via
`python build_ext.py --stress --tsan && ./build/engine_stress_tsan` —
race coverage the reference lacks entirely."""
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(300)
def test_engine_stress_plain():
    subprocess.check_call([sys.executable, "build_ext.py", "--stress"], cwd=ROOT)
    out = subprocess.run([os.path.join(ROOT, "build", "engine_stress"), "2"],
                         capture_output=True, text=True, timeout=240)
    assert out.returncode == 0, out.stdout + out.stderr
    assert "engine_stress OK" in out.stdout


@pytest.mark.parametrize("case", range(8))
def test_topology_fuzz(case):
    import windflow_amd as wf
    from windflow_amd import native
    """Random map/filter/flatmap chains with exact sum+count oracle over
    random parallelism/batch/mode (200-config campaign ran clean; note the
    oracle keeps values positive — C++ and Python disagree on % for
    negatives, which is a semantics difference, not a bug)."""
    import random
    rng = random.Random(31337 + case * 29)
    n = rng.choice([2000, 7000, 20000])
    keys = rng.choice([1, 4, 13])
    batch = rng.choice([64, 256, 1024, 4096])
    mode = rng.choice([wf.ExecutionMode.DEFAULT, wf.ExecutionMode.DETERMINISTIC])
    src_par = rng.randint(1, 2)
    vals = list(range(1, n + 1)) * src_par  # each source replica emits all
    g = wf.PipeGraph("topo", mode=mode)
    src = (wf.Source_Builder(native.seq_source(n, keys, batch))
           .withParallelism(src_par).withOutputSchema([0]).build())
    mp = g.add_source(src)
    for _ in range(rng.randint(1, 5)):
        k = rng.choice(["map", "filter", "flatmap"])
        par = rng.randint(1, 4)
        if k == "map":
            a, c = rng.choice([1, 2, 3]), rng.randint(0, 5)
            mp.add(wf.Map_Builder(native.affine_map(0, a, c))
                   .withParallelism(par).withOutputSchema([0]).build())
            vals = [a * v + c for v in vals]
        elif k == "filter":
            m = rng.choice([2, 3, 5])
            c = rng.randint(0, m - 1)
            keep_eq = rng.random() < 0.5
            mp.add(wf.Filter_Builder(native.mod_filter(0, m, c, keep_eq))
                   .withParallelism(par).withOutputSchema([0]).build())
            vals = [v for v in vals if ((v % m == c) == keep_eq)]
        else:
            kk = rng.randint(2, 3)
            mp.add(wf.FlatMap_Builder(native.dup_flatmap(kk))
                   .withParallelism(par).withOutputSchema([0]).build())
            vals = [v for v in vals for _ in range(kk)]
    snk = (wf.Sink_Builder(native.sum_sink(0))
           .withParallelism(rng.randint(1, 2)).build())
    mp.add_sink(snk)
    g.run()
    assert g.sink_sum(snk) == sum(vals)
    assert g.sink_count(snk) == len(vals)


@pytest.mark.parametrize("case", range(4))
def test_chained_topology_fuzz(case):
    """Same random op chains as test_topology_fuzz but FUSED into one
    replica thread via .chain()/.chain_sink() (120-config campaign clean)."""
    import random
    import windflow_amd as wf
    from windflow_amd import native
    rng = random.Random(1_100_000 + case * 31)
    n = rng.choice([2000, 10000])
    keys = rng.choice([1, 4, 13])
    batch = rng.choice([64, 512, 2048])
    vals = list(range(1, n + 1))
    g = wf.PipeGraph("chain")
    mp = g.add_source(wf.Source_Builder(native.seq_source(n, keys, batch))
                      .withParallelism(1).withOutputSchema([0]).build())
    for _ in range(rng.randint(1, 6)):
        k = rng.choice(["map", "filter", "flatmap"])
        if k == "map":
            a, c = rng.choice([1, 2, 3]), rng.randint(0, 5)
            mp.chain(wf.Map_Builder(native.affine_map(0, a, c))
                     .withOutputSchema([0]).build())
            vals = [a * v + c for v in vals]
        elif k == "filter":
            m = rng.choice([2, 3, 5])
            c = rng.randint(0, m - 1)
            ke = rng.random() < 0.5
            mp.chain(wf.Filter_Builder(native.mod_filter(0, m, c, ke))
                     .withOutputSchema([0]).build())
            vals = [v for v in vals if ((v % m == c) == ke)]
        else:
            kk = rng.randint(2, 3)
            mp.chain(wf.FlatMap_Builder(native.dup_flatmap(kk))
                     .withOutputSchema([0]).build())
            vals = [v for v in vals for _ in range(kk)]
    snk = wf.Sink_Builder(native.sum_sink(0)).withParallelism(1).build()
    mp.chain_sink(snk)
    g.run()
    assert g.sink_sum(snk) == sum(vals)
    assert g.sink_count(snk) == len(vals)
