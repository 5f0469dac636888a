#!/usr/bin/env python3
"""windflow_amd flagship benchmark — driver contract.

Headline metric (BASELINE.json): tuples/sec (whole node) on a keyed FFAT
sliding window, count-based win=1000 slide=100, bf16 tuple values with
random keys, synthetic GPU-resident stream.  One rank per GPU; for N>1
the driver launches via torch.distributed.run (RCCL backend).

A "step" = one micro-batch of --batch tuples through the pipeline
(device source -> keyed FFAT window -> sink).  W warmup steps run first
(untimed), then exactly K steps are timed, bracketed by barrier +
torch.cuda.synchronize on both sides; value = whole-job aggregate
tuples/sec (max step time over ranks).
"""
import argparse
import json
import os
import sys
import time

ROOT = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, ROOT)


def build_ffat_graph(n_tuples, batch, n_keys, win, slide, rank, world, device,
                     dense=True):
    import windflow_amd as wf
    from windflow_amd import native_gpu
    from windflow_amd.builders_gpu import (Source_GPU_Builder,
                                           Ffat_Windows_GPU_Builder,
                                           Sink_GPU_Builder)
    src = (Source_GPU_Builder(
        native_gpu.gpu_source(n_tuples, n_keys, batch, vdt=5, seed=42 + rank))
        .withOutputSchema([5]).withOutputBatchSize(batch)
        .withDevice(device).build())
    # keys are random-ORDER but dense-VALUED integers in [0, n_keys): the
    # withDenseKeys declaration (slot = key, no hash probe) is the honest
    # MI355X-first choice for this stream; --no-dense-keys measures the
    # general hashed path (both recorded in BASELINE.md)
    ff = (Ffat_Windows_GPU_Builder(
        native_gpu.gpu_ffat_windows(native_gpu.COMB_SUM, 0, win, slide,
                                    max_keys=2 * n_keys, dense_keys=dense))
        .withOutputSchema([2]).withOutputBatchSize(batch)
        .withDevice(device).build())
    snk = (Sink_GPU_Builder(native_gpu.gpu_count_sink())
           .withDevice(device).build())
    g = wf.PipeGraph("bench_ffat")
    mp = g.add_source(src)
    mp.chain(ff)
    mp.chain_sink(snk)
    return g, snk


def build_a2a_graph(n_tuples, batch, n_keys, rank, world, device, dist_cfg,
                    dense=True):
    """Driver config #4: Map_GPU -> RCCL keyby all-to-all -> Reduce_GPU,
    one rank per GPU."""
    import windflow_amd as wf
    from windflow_amd import native_gpu
    from windflow_amd.builders_gpu import (Source_GPU_Builder, Map_GPU_Builder,
                                           KeyBy_Exchange_GPU_Builder,
                                           Reduce_GPU_Builder, Sink_GPU_Builder)
    src = (Source_GPU_Builder(
        native_gpu.gpu_source(n_tuples, n_keys, batch, vdt=2, seed=42 + rank))
        .withOutputSchema([2]).withOutputBatchSize(batch)
        .withDevice(device).build())
    mp_ = (Map_GPU_Builder(native_gpu.gpu_affine_map(0, 2.0, 0.5, dtype=2))
           .withOutputSchema([2]).withOutputBatchSize(batch)
           .withDevice(device).build())
    ex = (KeyBy_Exchange_GPU_Builder(native_gpu.gpu_keyby_exchange())
          .withOutputSchema([2]).withOutputBatchSize(3 * batch)
          .withDevice(device).build())
    rd = (Reduce_GPU_Builder(
        native_gpu.gpu_keyed_reduce(native_gpu.COMB_SUM, 0, 4 * n_keys,
                                    dense_keys=dense))
        .withOutputSchema([2]).withOutputBatchSize(3 * batch)
        .withDevice(device).build())
    snk = Sink_GPU_Builder(native_gpu.gpu_count_sink()).withDevice(device).build()
    g = wf.PipeGraph("bench_a2a")
    g.set_dist(*dist_cfg)
    mp = g.add_source(src)
    mp.chain(mp_)
    mp.chain(ex)
    mp.chain(rd)
    mp.chain_sink(snk)
    return g, snk


def build_ffat_x_graph(n_tuples, batch, n_keys, win, slide, rank, world, device,
                       dist_cfg, dense=True):
    """Keyed FFAT with a true cross-GPU keyby: source -> RCCL all-to-allv
    exchange -> FFAT window.  Every key's window state lives on exactly one
    rank (hash(key) % world)."""
    import windflow_amd as wf
    from windflow_amd import native_gpu
    from windflow_amd.builders_gpu import (Source_GPU_Builder,
                                           KeyBy_Exchange_GPU_Builder,
                                           Ffat_Windows_GPU_Builder,
                                           Sink_GPU_Builder)
    src = (Source_GPU_Builder(
        native_gpu.gpu_source(n_tuples, n_keys, batch, vdt=5, seed=42 + rank))
        .withOutputSchema([5]).withOutputBatchSize(batch)
        .withDevice(device).build())
    ex = (KeyBy_Exchange_GPU_Builder(native_gpu.gpu_keyby_exchange())
          .withOutputSchema([5]).withOutputBatchSize(3 * batch)
          .withDevice(device).build())
    ff = (Ffat_Windows_GPU_Builder(
        native_gpu.gpu_ffat_windows(native_gpu.COMB_SUM, 0, win, slide,
                                    max_keys=2 * n_keys, dense_keys=dense))
        .withOutputSchema([2]).withOutputBatchSize(3 * batch)
        .withDevice(device).build())
    snk = Sink_GPU_Builder(native_gpu.gpu_count_sink()).withDevice(device).build()
    g = wf.PipeGraph("bench_ffat_x")
    g.set_dist(*dist_cfg)
    mp = g.add_source(src)
    mp.chain(ex)
    mp.chain(ff)
    mp.chain_sink(snk)
    return g, snk


def build_mapfilter_graph(n_tuples, batch, n_keys, rank, world, device):
    """Driver config #2: Map_GPU + Filter_GPU chained, bf16 tuples."""
    import windflow_amd as wf
    from windflow_amd import native_gpu
    from windflow_amd.builders_gpu import (Source_GPU_Builder, Map_GPU_Builder,
                                           Filter_GPU_Builder, Sink_GPU_Builder)
    src = (Source_GPU_Builder(
        native_gpu.gpu_source(n_tuples, n_keys, batch, vdt=5, seed=42 + rank))
        .withOutputSchema([5]).withOutputBatchSize(batch)
        .withDevice(device).build())
    mp_ = (Map_GPU_Builder(native_gpu.gpu_affine_map(0, 1.5, 0.25, dtype=5))
           .withOutputSchema([5]).withOutputBatchSize(batch)
           .withDevice(device).build())
    fl = (Filter_GPU_Builder(native_gpu.gpu_gt_filter(0, 0.5, dtype=5))
          .withOutputSchema([5]).withOutputBatchSize(batch)
          .withDevice(device).build())
    snk = Sink_GPU_Builder(native_gpu.gpu_count_sink()).withDevice(device).build()
    g = wf.PipeGraph("bench_mapfilter")
    mp = g.add_source(src)
    mp.chain(mp_)
    mp.chain(fl)
    mp.chain_sink(snk)
    return g, snk


def build_join_graph(n_tuples, batch, n_keys, par):
    """Driver config #5 shape (CPU side): two keyed sources -> interval
    join (KP) -> paned windows -> sink, watermark collectors throughout."""
    import windflow_amd as wf
    from windflow_amd import native
    from windflow_amd.builders import (Interval_Join_Builder,
                                       Paned_Windows_Builder)
    g = wf.PipeGraph("bench_join")
    sa = (wf.Source_Builder(native.seq_source(n_tuples, n_keys, batch))
          .withParallelism(1).withOutputSchema([0])
          .withOutputBatchSize(batch).build())
    # B's offset must be a multiple of n_keys: key = v % n_keys, so a
    # misaligned offset shifts every B key off A's and the join matches
    # NOTHING (round-2 validity fix — the join was running empty)
    off_b = (1_000_000_000 // n_keys) * n_keys
    sb = (wf.Source_Builder(native.seq_source(n_tuples, n_keys, batch,
                                              value_offset=off_b))
          .withParallelism(1).withOutputSchema([0])
          .withOutputBatchSize(batch).build())
    mpA = g.add_source(sa)
    mpB = g.add_source(sb)
    mp = mpA.merge(mpB)
    mp.add(Interval_Join_Builder().withBoundaries(-2, 2).withKPMode()
           .withValueCols(0).withParallelism(par)
           .withOutputSchema([0, 0]).withOutputBatchSize(batch).build())
    mp.add(Paned_Windows_Builder(plq_func=("sum", 0))
           .withTBWindows(1000, 100).withParallelism(par)
           .withOutputSchema([0]).withOutputBatchSize(batch).build())
    snk = wf.Sink_Builder(native.count_sink()).withParallelism(1).build()
    mp.add_sink(snk)
    return g, snk


def build_cpu_graph(n_tuples, batch, par=1):
    """Config #1: Source -> Map -> Filter -> Sink.  par=1 chains all four
    into ONE thread (round-1 shape); par>1 runs `par` source replicas
    each chained with its own map+filter+sink replica (rebalanced
    stream shards, zero cross-thread shuffles — operator replication,
    SURVEY §2.10)."""
    import windflow_amd as wf
    from windflow_amd import native
    g = wf.PipeGraph("bench_cpu")
    per = n_tuples // par
    pipes = []
    snks = []
    for r in range(par):
        src = (wf.Source_Builder(native.seq_source(per, 1000, batch,
                                                   value_offset=r * per))
               .withParallelism(1).withOutputSchema([0])
               .withOutputBatchSize(batch).build())
        mp = g.add_source(src)
        mp.chain(wf.Map_Builder(native.affine_map(0, 3, 1)).withParallelism(1)
                 .withOutputSchema([0]).withOutputBatchSize(batch).build())
        mp.chain(wf.Filter_Builder(native.mod_filter(0, 5, 0)).withParallelism(1)
                 .withOutputSchema([0]).withOutputBatchSize(batch).build())
        snk = wf.Sink_Builder(native.count_sink()).withParallelism(1).build()
        mp.chain_sink(snk)
        snks.append(snk)
        pipes.append(mp)
    return g, snks[0]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    # defaults sized so the driver's own record carries a sustained claim
    # (~1.7 B tuples, ~100 ms timed region) while finishing in seconds
    ap.add_argument("--steps", type=int, default=200)
    ap.add_argument("--warmup", type=int, default=20)
    # default micro-batch = measured throughput plateau (BASELINE.md sweep:
    # 16.2 B t/s @4M, 19.8 @8M, 19.8 @16M; p99 2.2 ms at 8M)
    ap.add_argument("--batch", type=int, default=16_777_216)
    ap.add_argument("--keys", type=int, default=8192,
                    help="distinct keys per rank")
    ap.add_argument("--par", type=int, default=1,
                    help="CPU configs: operator parallelism")
    ap.add_argument("--no-dense-keys", action="store_true",
                    help="ffat config: use the hashed key->slot path even "
                         "though the synthetic keys are dense integers")
    ap.add_argument("--win", type=int, default=1000)
    ap.add_argument("--slide", type=int, default=100)
    ap.add_argument("--config", choices=["ffat", "ffat_x", "a2a", "mapfilter", "cpu", "join"], default="ffat")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    dist = None
    torch = None
    dist_cfg = None
    if args.config not in ("cpu", "join"):
        import torch  # noqa: F811
        if world > 1:
            import torch.distributed as dist  # noqa: F811
            # gloo for the out-of-band bootstrap (rccl id broadcast +
            # barriers): torch's own nccl group would refuse multiple ranks
            # per device, but the ENGINE's RCCL communicator is the thing
            # under test; device collectives all run through it.
            backend = os.environ.get("WFA_DIST_BACKEND", "gloo")
            dist.init_process_group(backend, rank=rank, world_size=world)
            torch.cuda.set_device(local_rank)
        if args.config in ("a2a", "ffat_x"):
            if world > 1:
                from windflow_amd.dist import init_from_torch
                dist_cfg = init_from_torch()
            else:
                from windflow_amd import _core
                dist_cfg = (0, 1, _core.rccl_unique_id())

    def sync():
        if torch is not None and torch.cuda.is_available():
            torch.cuda.synchronize()

    def barrier():
        if dist is not None:
            dist.barrier()

    K, W, B = args.steps, args.warmup, args.batch
    p99_us = None

    if args.config in ("cpu", "join"):
        B = min(B, 65536)
        build = ((lambda n, b: build_cpu_graph(n, b, args.par))
                 if args.config == "cpu"
                 else lambda n, b: build_join_graph(n, b, args.keys,
                                                    args.par if args.par > 1 else 4))
        gw, _ = build(W * B, B)
        gw.run()
        g, snk = build(K * B, B)
        t0 = time.time()
        g.run()
        dt = time.time() - t0
        n_gpus = 0
        lat = sorted(g.engine.sink_latencies(g._sink_map[id(snk)]))
        p99_us = lat[min(len(lat) - 1, int(0.99 * len(lat)))] if lat else None
    else:
        def builder(steps):
            if args.config == "mapfilter":
                return build_mapfilter_graph(steps * B, B, args.keys, rank,
                                             world, local_rank)
            if args.config == "a2a":
                return build_a2a_graph(steps * B, B, args.keys, rank, world,
                                       local_rank, dist_cfg,
                                       dense=not args.no_dense_keys)
            if args.config == "ffat_x":
                return build_ffat_x_graph(steps * B, B, args.keys, args.win,
                                          args.slide, rank, world, local_rank,
                                          dist_cfg,
                                          dense=not args.no_dense_keys)
            return build_ffat_graph(steps * B, B, args.keys, args.win,
                                    args.slide, rank, world, local_rank,
                                    dense=not args.no_dense_keys)
        # warmup engine (also JIT-warms pools/streams/arena)
        if W > 0:
            gw, _ = builder(W)
            gw.run()
        g, snk = builder(K)
        g.prepare()   # threads spawned + streams/pools warm, gated
        sync()
        barrier()
        t0 = time.time()
        g.run_gated()
        sync()
        t1 = time.time()
        barrier()
        dt = t1 - t0
        n_gpus = world
        lat = sorted(g.engine.sink_latencies(g._sink_map[id(snk)]))
        p99_us = lat[min(len(lat) - 1, int(0.99 * len(lat)))] if lat else None

    # max step time over ranks (tensor device must match the backend)
    if dist is not None:
        dev = "cuda" if dist.get_backend() == "nccl" else "cpu"
        t = torch.tensor([dt], dtype=torch.float64, device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        dt = float(t.item())

    total_tuples = K * B * max(world, 1)
    value = total_tuples / dt
    if rank == 0:
        out = {
            "metric": "tuples_per_sec",
            "value": value,
            "unit": "tuples/s",
            "n_gpus": 0 if args.config in ("cpu", "join") else n_gpus,
            "steps": K,
            "warmup": W,
            "ms_per_step": dt / K * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": {"ffat": "bf16", "ffat_x": "bf16", "a2a": "f32",
                      "mapfilter": "bf16", "cpu": "int64",
                      "join": "int64"}[args.config],
            "data": "synthetic",
            "config": {
                "model": {"ffat": "keyed_ffat_cb_window",
                          "ffat_x": "rccl_keyby_ffat_cb_window",
                          "a2a": "map_gpu_rccl_keyby_reduce_gpu",
                          "mapfilter": "map_gpu_filter_gpu_chained",
                          "cpu": "cpu_source_map_filter_sink",
                          "join": "interval_join_paned_windows_cpu"}[args.config],
                "global_batch": B * max(world, 1),
                "win": args.win,
                "slide": args.slide,
                "keys_per_rank": args.keys,
                "parallelism": f"keyed-dp{max(world,1)}",
                "p99_batch_latency_us": p99_us,
                "dense_keys": (args.config in ("ffat", "ffat_x", "a2a")
                               and not args.no_dense_keys),
            },
        }
        print(json.dumps(out))
    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
