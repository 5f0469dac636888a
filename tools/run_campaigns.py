#!/usr/bin/env python3
"""Differential fuzz campaigns, runnable at any scale:

    python tools/run_campaigns.py           # ~1 min: 100 configs/domain
    python tools/run_campaigns.py --full    # the round-1 scale (1000+)

Domains: window forms vs oracles, interval joins, op-chain topologies,
backpressure (queue cap 2).  Each failure prints its full config for
distillation into a permanent test (nine bugs were found this way in
round 1 — see README).
"""
import argparse
import os
import random
import sys
from collections import Counter

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tests"))

import windflow_amd as wf                              # noqa: E402
from windflow_amd import native                        # noqa: E402
from test_windows import (run_graph, seq_stream, oracle_cb, oracle_tb,  # noqa: E402
                          got_counter, join_graph, oracle_join)
from test_windows_fuzz import BUILDERS                 # noqa: E402


def windows(n_cases, base_seed):
    fails = []
    for case in range(n_cases):
        rng = random.Random(base_seed + case)
        kind = rng.choice(list(BUILDERS))
        agg = rng.choice(["sum", "max", "min", "count"])
        wt = rng.choice(["cb", "tb"])
        slide = rng.choice([1, 2, 3, 5, 10, 25, 60, 100])
        win = slide * rng.randint(1, 10)
        n_keys = rng.choice([1, 2, 3, 11, 64, 200])
        batch = rng.choice([8, 16, 64, 256, 1000, 4096])
        par = rng.randint(1, 6)
        stream = rng.choice([300, 700, 1500, 3100])
        mode = rng.choice([wf.ExecutionMode.DEFAULT,
                           wf.ExecutionMode.DETERMINISTIC])
        cfg = (case, kind, agg, wt, win, slide, n_keys, batch, par, stream)
        try:
            b = BUILDERS[kind]((agg, 0))
            b = (b.withCBWindows(win, slide) if wt == "cb"
                 else b.withTBWindows(win, slide))
            op = b.withParallelism(par).withOutputSchema([0]).build()
            rows = run_graph(op, stream_len=stream, n_keys=n_keys, batch=batch,
                             mode=mode)
            exp = (oracle_cb if wt == "cb" else oracle_tb)(
                seq_stream(stream, n_keys), win, slide, agg)
            if got_counter(rows) != exp:
                fails.append(("MISMATCH",) + cfg)
        except Exception as e:
            fails.append(("ERROR", str(e)[:100]) + cfg)
    return fails


def joins(n_cases, base_seed):
    fails = []
    for case in range(n_cases):
        rng = random.Random(base_seed + case)
        lower, upper = -rng.randint(0, 30), rng.randint(0, 30)
        keys = rng.choice([1, 2, 5, 9, 33])
        n = rng.choice([300, 800, 1500, 4000])
        batch = rng.choice([8, 16, 64, 300, 2048])
        par = rng.randint(1, 4)
        kp = rng.random() < 0.5
        mode = rng.choice([wf.ExecutionMode.DEFAULT,
                           wf.ExecutionMode.DETERMINISTIC])
        cfg = (case, lower, upper, keys, n, batch, par, kp)
        try:
            rows = join_graph(lambda b: (b.withKPMode() if kp else b.withDPMode())
                              .withParallelism(par), n=n, keys=keys,
                              lower=lower, upper=upper, batch=batch, mode=mode)
            got = Counter((k, a, bb) for _, k, a, bb in rows)
            if got != oracle_join(n, keys, lower, upper):
                fails.append(("MISMATCH",) + cfg)
        except Exception as e:
            fails.append(("ERROR", str(e)[:100]) + cfg)
    return fails


def topologies(n_cases, base_seed):
    fails = []
    for case in range(n_cases):
        rng = random.Random(base_seed + case)
        n = rng.choice([2000, 7000, 20000])
        keys = rng.choice([1, 4, 13])
        batch = rng.choice([64, 256, 1024, 4096])
        mode = rng.choice([wf.ExecutionMode.DEFAULT,
                           wf.ExecutionMode.DETERMINISTIC])
        src_par = rng.randint(1, 2)
        vals = list(range(1, n + 1)) * src_par
        g = wf.PipeGraph("topo", mode=mode)
        mp = g.add_source(wf.Source_Builder(native.seq_source(n, keys, batch))
                          .withParallelism(src_par).withOutputSchema([0]).build())
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
                ke = rng.random() < 0.5
                mp.add(wf.Filter_Builder(native.mod_filter(0, m, c, ke))
                       .withParallelism(par).withOutputSchema([0]).build())
                vals = [v for v in vals if ((v % m == c) == ke)]
            else:
                kk = rng.randint(2, 3)
                mp.add(wf.FlatMap_Builder(native.dup_flatmap(kk))
                       .withParallelism(par).withOutputSchema([0]).build())
                vals = [v for v in vals for _ in range(kk)]
        snk = (wf.Sink_Builder(native.sum_sink(0))
               .withParallelism(rng.randint(1, 2)).build())
        mp.add_sink(snk)
        try:
            g.run()
            if g.sink_sum(snk) != sum(vals) or g.sink_count(snk) != len(vals):
                fails.append(("MISMATCH", case))
        except Exception as e:
            fails.append(("ERROR", str(e)[:100], case))
    return fails


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--full", action="store_true")
    ap.add_argument("--seed", type=int, default=5_000_000)
    args = ap.parse_args()
    n = 1000 if args.full else 100
    total = 0
    for name, fn in (("windows", windows), ("joins", joins),
                     ("topologies", topologies)):
        fails = fn(n, args.seed)
        print(f"{name}: {n} configs, {len(fails)} failures")
        for f in fails[:5]:
            print("   ", f)
        total += len(fails)
    # backpressure: rerun the window domain at queue cap 2
    os.environ["WFA_QUEUE_CAP"] = "2"
    fails = windows(max(20, n // 5), args.seed + 777)
    print(f"windows@cap2: {max(20, n // 5)} configs, {len(fails)} failures")
    total += len(fails)
    print("TOTAL FAILURES:", total)
    sys.exit(1 if total else 0)


if __name__ == "__main__":
    main()
