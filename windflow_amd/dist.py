"""Distributed helpers: one process per GPU, RCCL over xGMI.

The engine's gpu_keyby_exchange stage needs an ncclUniqueId shared by all
ranks; torch.distributed (already initialized by the launcher) is the
out-of-band channel.  CPU pipelines get a pure-Python exchange stage over
the gloo backend with identical bucketing semantics (hash(key) % world via
_core.hash_key), used by the multi-process CPU tests.
"""
import numpy as np

from . import _core


def init_from_torch():
    """-> (rank, world, rccl_id) using the current torch.distributed group."""
    import torch.distributed as td
    rank, world = td.get_rank(), td.get_world_size()
    obj = [_core.rccl_unique_id() if rank == 0 else None]
    td.broadcast_object_list(obj, src=0)
    return rank, world, obj[0]


def _hash_keys(keys):
    # must match KeyByEmitter::mix / wfa_bucket_by_key (splitmix64 finalizer)
    k = keys.astype(np.uint64) + np.uint64(0x9E3779B97F4A7C15)
    k = (k ^ (k >> np.uint64(30))) * np.uint64(0xBF58476D1CE4E5B9)
    k = (k ^ (k >> np.uint64(27))) * np.uint64(0x94D049BB133111EB)
    return k ^ (k >> np.uint64(31))


def cpu_keyby_exchange(n_cols=1):
    """Per-batch transform callable performing a gloo all-to-all keyby
    shuffle of host batches: every rank receives exactly the rows with
    hash(key) % world == rank.  All ranks must process the same number of
    batches (synchronized synthetic sources)."""
    import torch
    import torch.distributed as td

    world = td.get_world_size()
    rank = td.get_rank()

    def exchange(cols):
        keys = np.asarray(cols['key'], dtype=np.uint64)
        dest = (_hash_keys(keys) % np.uint64(world)).astype(np.int64)
        arrays = [np.asarray(cols['ts'], dtype=np.int64), keys.view(np.int64)]
        arrays += [np.asarray(cols[f'c{i}']) for i in range(n_cols)]
        counts = np.bincount(dest, minlength=world)
        # gloo has no all_to_all: allgather the count matrix, then paired
        # isend/irecv per peer (the RCCL path uses grouped send/recv too)
        cnt_all = [torch.zeros(world, dtype=torch.int64) for _ in range(world)]
        td.all_gather(cnt_all, torch.tensor(counts, dtype=torch.int64))
        na = len(arrays)
        send_rows = [torch.from_numpy(np.stack(
            [a[dest == p].view(np.int64) for a in arrays], axis=1)
            .reshape(-1)) for p in range(world)]
        recv_rows = [torch.zeros(int(cnt_all[p][rank]) * na, dtype=torch.int64)
                     for p in range(world)]
        reqs = []
        for p in range(world):
            if p == rank:
                recv_rows[p] = send_rows[p]
                continue
            if len(send_rows[p]):
                reqs.append(td.isend(send_rows[p], dst=p))
            if len(recv_rows[p]):
                reqs.append(td.irecv(recv_rows[p], src=p))
        for r in reqs:
            r.wait()
        out_ts, out_key, out_cols = [], [], [[] for _ in range(n_cols)]
        for p in range(world):
            rows = recv_rows[p].numpy().reshape(-1, na)
            out_ts.append(rows[:, 0])
            out_key.append(rows[:, 1].view(np.uint64))
            for c in range(n_cols):
                out_cols[c].append(rows[:, 2 + c])
        out = {'ts': np.concatenate(out_ts),
               'key': np.concatenate(out_key)}
        for c in range(n_cols):
            out[f'c{c}'] = np.concatenate(out_cols[c])
        return out

    return exchange
