"""Worker for the multi-process CPU keyby-exchange test (gloo, world 2).

Each rank runs: seq_source -> keyby exchange (all-to-all over gloo) -> sink.
Asserts: (1) every received row's hash(key) % world == my rank (keys are
partitioned), (2) the global sum of exchanged values equals the sum both
sources generated (nothing lost or duplicated).
"""
import os
import sys

import numpy as np
import torch
import torch.distributed as td

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import windflow_amd as wf                     # noqa: E402
from windflow_amd import native, _core        # noqa: E402
from windflow_amd.dist import cpu_keyby_exchange, _hash_keys  # noqa: E402


def main():
    td.init_process_group("gloo")
    rank, world = td.get_rank(), td.get_world_size()

    # RCCL bootstrap helper: id bytes must broadcast identically (this is
    # what wires the 8-GPU exchange in bench.py)
    from windflow_amd.dist import init_from_torch
    r, w, rid = init_from_torch()
    assert (r, w) == (rank, world)
    ok = [None]
    if rank == 0:
        td.broadcast_object_list([rid], src=0)
        ok[0] = True
    else:
        got = [None]
        td.broadcast_object_list(got, src=0)
        assert got[0] == rid, "rccl id bytes differ across ranks"
    # config via env (fuzzable); n % batch == 0: equal rounds per rank
    n = int(os.environ.get("WFZ_N", 20000))
    n_keys = int(os.environ.get("WFZ_KEYS", 13))
    batch = int(os.environ.get("WFZ_BATCH", 500))

    got = {"sum": 0, "rows": 0, "bad_dest": 0}

    def sink(cols):
        keys = np.asarray(cols["key"], dtype=np.uint64)
        vals = np.asarray(cols["c0"])
        got["sum"] += int(vals.sum())
        got["rows"] += len(vals)
        got["bad_dest"] += int(((_hash_keys(keys) % np.uint64(world))
                                != np.uint64(rank)).sum())

    g = wf.PipeGraph("dist_cpu")
    src = (wf.Source_Builder(native.seq_source(n, n_keys, batch, value_offset=rank))
           .withParallelism(1).withOutputSchema([0]).withOutputBatchSize(batch)
           .build())
    mp = g.add_source(src)
    mp.add(wf.FlatMap_Builder(cpu_keyby_exchange(n_cols=1))
           .withParallelism(1).withOutputSchema([0]).withOutputBatchSize(batch)
           .build())
    mp.add_sink(wf.Sink_Builder(sink).withParallelism(1).build())
    g.run()

    t = torch.tensor([got["sum"], got["rows"], got["bad_dest"]], dtype=torch.int64)
    td.all_reduce(t)
    exp_sum = sum(sum(range(1 + r, n + 1 + r)) for r in range(world))
    assert t[2].item() == 0, f"misrouted rows: {t[2].item()}"
    assert t[1].item() == n * world, (t[1].item(), n * world)
    assert t[0].item() == exp_sum, (t[0].item(), exp_sum)
    if rank == 0:
        print("DIST_CPU_OK", t.tolist())
    td.destroy_process_group()


if __name__ == "__main__":
    main()
