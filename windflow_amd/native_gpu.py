"""GPU native logic specs — map onto csrc/engine/gpu_ops.cpp factories and
the gfx950 kernels in csrc/hip/.  comb ids: 0 sum, 1 min, 2 max, 3 count.
Value dtypes: 0 I64, 2 F32, 5 U16(bf16)."""
from .operators import NativeLogic

COMB_SUM, COMB_MIN, COMB_MAX, COMB_COUNT = 0, 1, 2, 3


def gpu_source(stream_len, n_keys=1024, batch=1 << 20, vdt=5, seed=42):
    """Device-resident synthetic source (ts, key, value[vdt])."""
    return NativeLogic("gpu_source", "", [], [stream_len, n_keys, batch, vdt, seed])


def gpu_affine_map(col=0, a=1.0, b=0.0, dtype=5):
    """x = a*x + b on device; dtype picks the kernel (i64/f32/bf16)."""
    spec = {0: 1, 2: 2, 5: 3}[dtype]
    return NativeLogic("gpu_map", "", [float(a), float(b)], [spec, col])


def gpu_square_map(col=0):
    return NativeLogic("gpu_map", "", [0.0, 0.0], [4, col])


def gpu_mod_filter(col=0, m=3, c=0):
    """keep x % m != c (i64)."""
    return NativeLogic("gpu_filter", "", [float(m), float(c)], [1, col])


def gpu_gt_filter(col=0, thr=0.5, dtype=2):
    """keep x > thr (f32) / x >= thr (bf16)."""
    spec = 2 if dtype == 2 else 3
    return NativeLogic("gpu_filter", "", [float(thr), 0.0], [spec, col])


def gpu_jit_map(expr, col=0):
    """Custom device map, runtime-compiled with hiprtc for the local arch
    (parity with the reference's __device__ lambda MAP_GPU).  `expr` is a C
    expression over `v` (the column value, mutable type), `ts` (i64) and
    `key` (u64); the result is written back to the column, e.g.
    gpu_jit_map("v * v + 1.0f", col=0)."""
    return NativeLogic("gpu_jit_map", expr, [], [col])


def gpu_jit_filter(expr, col=0):
    """Custom device predicate (reference FILTER_GPU lambda): keep rows
    where `expr` is true, e.g. gpu_jit_filter("v > 0.5f && key % 2 == 0")."""
    return NativeLogic("gpu_jit_filter", expr, [], [col])


def gpu_keyed_running_sum(col=0, max_keys=1 << 16):
    """stateful map: per-key running sum written in place (key-order walk)."""
    return NativeLogic("gpu_map_keyed", "", [0.0, 0.0], [1, col, max_keys])


def gpu_keyed_ema(col=0, alpha=0.9, max_keys=1 << 16):
    """stateful map: x = ema(state, x) per key."""
    return NativeLogic("gpu_map_keyed", "", [float(alpha), 0.0], [2, col, max_keys])


def gpu_keyed_running_count(col=0, max_keys=1 << 16):
    return NativeLogic("gpu_map_keyed", "", [0.0, 0.0], [3, col, max_keys])


def gpu_keyed_dedup(col=0, max_keys=1 << 16):
    """stateful filter: drop consecutive per-key duplicates."""
    return NativeLogic("gpu_filter_keyed", "", [0.0, 0.0], [1, col, max_keys])


def gpu_keyed_every_kth(k=2, col=0, max_keys=1 << 16):
    """stateful filter: keep every k-th tuple per key."""
    return NativeLogic("gpu_filter_keyed", "", [float(k), 0.0], [2, col, max_keys])


def gpu_keyed_reduce(comb=COMB_SUM, col=0, max_keys=1 << 16):
    """per-batch keyed reduction -> one (key, agg, ts_max) per distinct key."""
    return NativeLogic("gpu_reduce", "", [], [comb, col, max_keys])


def gpu_reduce_all(comb=COMB_SUM, col=0):
    """per-batch UNKEYED full reduction -> one (value, ts_max) tuple per
    batch (reference reduce_gpu.hpp:269 thrust::reduce path); deterministic
    two-stage tree, no sort and no atomics."""
    return NativeLogic("gpu_reduce_all", "", [], [comb, col])


def gpu_ffat_windows(comb=COMB_SUM, col=0, win=1000, slide=100,
                     max_keys=1 << 16, use_tree=False, tb=False, lateness=0,
                     pend_ring_log2=0):
    """Keyed sliding window over panes (pane = gcd(win, slide)).
    CB (default): windows fire every `slide` tuples per key; use_tree
    selects the FlatFAT-arena path (O(log) combine for large win/slide
    ratios and non-invertible combines).
    TB (tb=True): event-time windows [w*slide, w*slide+win) on ts; panes
    complete at watermark - lateness; pending out-of-order panes live in a
    2^pend_ring_log2 ring per key (default 2^16)."""
    if win < 1 or slide < 1:
        raise ValueError("window length and slide must be >= 1")
    return NativeLogic("gpu_ffat", "", [],
                       [comb, col, win, slide, max_keys, 1 if use_tree else 0,
                        1 if tb else 0, lateness, pend_ring_log2])


def gpu_keyby_exchange():
    """Inter-GPU keyby shuffle: rows bucketed by hash(key) % world on device,
    exchanged with a grouped RCCL send/recv all-to-allv over xGMI.  One
    replica per rank; requires PipeGraph.set_dist(rank, world, rccl_id)."""
    return NativeLogic("gpu_exchange", "", [], [])


def gpu_gram_windows(win, max_keys=1 << 14):
    """MFMA windowed Gram aggregator (MI355X extension): per-key tumbling
    windows of `win` tuples over 16 F32 payload columns; the window
    aggregate sum(v * v^T) — online covariance / Gram — is GEMM-shaped and
    runs on the matrix cores (v_mfma_f32_16x16x4_f32).  Emits 16 rows per
    fired window: (key, ts=last tuple, c0=gwid, c1..c16 = one Gram row).
    Output schema must be [I64] + [F32]*16."""
    return NativeLogic("gpu_gram", "", [], [win, max_keys])


def gpu_count_sink():
    return NativeLogic("gpu_count_sink", "", [], [])


def gpu_to_host():
    return NativeLogic("gpu_to_host", "", [], [])
