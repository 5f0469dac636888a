"""Randomized differential sweep over the window operators: random
win/slide/key-count/batch-size/parallelism drawn per case, every operator
form compared against the same brute-force oracle (the reference randomizes
degrees but has no oracle)."""
import random
from collections import defaultdict

import pytest

import windflow_amd as wf
from windflow_amd import native
from windflow_amd.builders import (Keyed_Windows_Builder, Parallel_Windows_Builder,
                                   Paned_Windows_Builder, MapReduce_Windows_Builder,
                                   Ffat_Windows_Builder)

import sys, os
sys.path.insert(0, os.path.dirname(__file__))
from test_windows import (run_graph, seq_stream, oracle_cb, oracle_tb,
                          got_counter)  # noqa: E402

BUILDERS = {
    "keyed": lambda f: Keyed_Windows_Builder(func=f),
    "parallel": lambda f: Parallel_Windows_Builder(func=f),
    "paned": lambda f: Paned_Windows_Builder(plq_func=f),
    "mapreduce": lambda f: MapReduce_Windows_Builder(map_func=f),
    "ffat": lambda f: Ffat_Windows_Builder(comb=f),
}


@pytest.mark.parametrize("case", range(18))
def test_window_fuzz(case):
    # widened space (min/count aggs, slide=1, 64 keys, batch > stream): the
    # original narrow draw missed three real bugs — MIN/MAX identity on
    # empty partials, empty-TB-window skip at EOS, and intermediate-batch
    # watermark stamping (all found by a 300-config campaign of this loop)
    rng = random.Random(50000 + case * 17)
    kind = rng.choice(list(BUILDERS))
    agg = rng.choice(["sum", "max", "min", "count"])
    wt = rng.choice(["cb", "tb"])
    slide = rng.choice([1, 3, 5, 10, 25, 60])
    win = slide * rng.randint(1, 8)
    n_keys = rng.choice([1, 2, 3, 11, 64])
    batch = rng.choice([16, 32, 128, 1000, 4096])
    par = rng.randint(1, 4)
    stream = rng.choice([700, 1500, 3100, 7000])
    mode = rng.choice([wf.ExecutionMode.DEFAULT, wf.ExecutionMode.DETERMINISTIC])
    b = BUILDERS[kind]((agg, 0))
    b = (b.withCBWindows(win, slide) if wt == "cb" else b.withTBWindows(win, slide))
    op = b.withParallelism(par).withOutputSchema([0]).build()
    rows = run_graph(op, stream_len=stream, n_keys=n_keys, batch=batch, mode=mode)
    per = seq_stream(stream, n_keys)
    exp = (oracle_cb if wt == "cb" else oracle_tb)(per, win, slide, agg)
    assert got_counter(rows) == exp, (kind, agg, wt, win, slide, n_keys, batch, par)


@pytest.mark.parametrize("case", range(8))
def test_join_fuzz(case):
    """Randomized interval joins: bounds/keys/batch/parallelism/mode drawn
    per case, vs the brute-force pair oracle."""
    from collections import Counter
    from test_windows import join_graph, oracle_join
    rng = random.Random(90000 + case * 13)
    lower = -rng.randint(0, 30)
    upper = rng.randint(0, 30)
    keys = rng.choice([1, 2, 5, 9, 33])
    n = rng.choice([300, 800, 1500, 4000])
    batch = rng.choice([8, 16, 64, 300, 2048])
    par = rng.randint(1, 4)
    kp = rng.random() < 0.5
    rows = join_graph(lambda b: (b.withKPMode() if kp else b.withDPMode())
                      .withParallelism(par),
                      n=n, keys=keys, lower=lower, upper=upper, batch=batch)
    got = Counter((k, a, bb) for _, k, a, bb in rows)
    assert got == oracle_join(n, keys, lower, upper), (lower, upper, keys, n,
                                                       batch, par, kp)
