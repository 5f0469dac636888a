"""Numpy reference of the device-side synthetic generator (k_gen in
csrc/hip/kernels.hip) — used by GPU numerics tests as the CPU oracle."""
import numpy as np


def _mix64(k):
    k = (k + np.uint64(0x9e3779b97f4a7c15)) & np.uint64(0xFFFFFFFFFFFFFFFF)
    k = ((k ^ (k >> np.uint64(30))) * np.uint64(0xbf58476d1ce4e5b9)) & np.uint64(0xFFFFFFFFFFFFFFFF)
    k = ((k ^ (k >> np.uint64(27))) * np.uint64(0x94d049bb133111eb)) & np.uint64(0xFFFFFFFFFFFFFFFF)
    return k ^ (k >> np.uint64(31))


def f32_to_bf16_np(f):
    u = f.astype(np.float32).view(np.uint32)
    lsb = (u >> np.uint32(16)) & np.uint32(1)
    u = u + np.uint32(0x7FFF) + lsb
    return (u >> np.uint32(16)).astype(np.uint16)


def bf16_to_f32_np(h):
    return (h.astype(np.uint32) << np.uint32(16)).view(np.float32)


def gen_batch(n, start, seed, n_keys, vdt):
    """Matches wfa_gen_batch exactly (bit-for-bit)."""
    old = np.seterr(over='ignore')
    try:
        i = np.arange(start, start + n, dtype=np.uint64)
        h = _mix64(np.uint64(seed) ^ i)
        ts = i.astype(np.int64)
        key = h % np.uint64(n_keys)
        h2 = _mix64(h)
        if vdt == 0:
            val = (h2 % np.uint64(1000)).astype(np.int64)
        else:
            f = (h2 >> np.uint64(40)).astype(np.float32) * np.float32(1.0 / 16777216.0)
            if vdt == 2:
                val = f
            else:  # bf16 stored as u16
                val = f32_to_bf16_np(f)
    finally:
        np.seterr(**old)
    return ts, key, val
