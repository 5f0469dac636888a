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
        # Rows travel as i64 words; each payload column is widened to an
        # exact 8-byte carrier (f32->f64, i32/u32/u16/u8->i64, f64/u64 bit
        # views) and restored to its ORIGINAL dtype on receive — round 1
        # left float payloads as int64 bit patterns (silent corruption).
        col_dtypes = []
        for i in range(n_cols):
            a = np.asarray(cols[f'c{i}'])
            col_dtypes.append(a.dtype)
            if a.dtype.itemsize == 8:
                a = a.view(np.int64)
            elif np.issubdtype(a.dtype, np.floating):
                a = a.astype(np.float64).view(np.int64)
            else:
                a = a.astype(np.int64)
            arrays.append(a)
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
            raw = np.ascontiguousarray(np.concatenate(out_cols[c]))
            dt = col_dtypes[c]
            if dt.itemsize == 8:
                out[f'c{c}'] = raw.view(dt)
            elif np.issubdtype(dt, np.floating):
                out[f'c{c}'] = raw.view(np.float64).astype(dt)
            else:
                out[f'c{c}'] = raw.astype(dt)
        return out

    return exchange
