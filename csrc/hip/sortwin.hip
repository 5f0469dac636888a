// gfx950 keyed-stream machinery: key->dense-slot hash table, stable 4-bit
// LSD radix sort, segment extraction, segmented reduce, and the batched
// multi-key sliding-window fold.
//
// MI355X-native redesign of the reference's keyed GPU path:
//   Extract_Dests/Compute_Mapping + thrust sort/unique
//       (keyby_emitter_gpu.hpp:68-99,518-583)        -> hash table + radix sort
//   Reduce_GPU thrust sort_by_key/reduce_by_key
//       (reduce_gpu.hpp:239-269)                     -> wfa_segment_reduce
//   Ffat_Replica_GPU per-key host loop + per-key FlatFAT_GPU streams
//       (ffat_replica_gpu.hpp:423-1047, flatfat_gpu.hpp) ->
//       wfa_ffat_cb_fold / wfa_ffat_tree_fold: ONE kernel advances every
//       key's window state over the batch's segments (the reference's
//       per-key host loop is its main scaling flaw — SURVEY.md §7 step 6).
#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>

#include "wfa_kernels.h"

// WFA_DEBUG_SYNC: sync + error-print after each sort sub-kernel (fault
// localization; see gpu_common.hpp dbg_sync)
static int wfa_sort_dbg() {
    static int v = -1;
    if (v < 0) v = getenv("WFA_DEBUG_SYNC") ? 1 : 0;
    return v;
}
static void sdbg(hipStream_t st, const char* what, int pass) {
    if (!wfa_sort_dbg()) return;
    hipError_t e = hipStreamSynchronize(st);
    hipError_t e2 = hipGetLastError();
    fprintf(stderr, "[sortdbg] p%d %s sync=%s last=%s\n", pass, what,
            hipGetErrorString(e), hipGetErrorString(e2));
    fflush(stderr);
}

#define WFA_THREADS 256
#define WFA_MAX_BLOCKS 2048
#define RS_IPT 8
#define RS_PER_BLOCK (WFA_THREADS * RS_IPT)
#ifndef RS8_IPT
#define RS8_IPT 16   // items per thread in the 8-bit sort (A/B via -DRS8_IPT=N)
#endif
#define RS8_PER_WAVE (64 * RS8_IPT)
#define RS8_PER_BLOCK (WFA_THREADS * RS8_IPT)

static inline int64_t nblk(int64_t n, int64_t per_thread = 1) {
    int64_t b = (n + WFA_THREADS * per_thread - 1) / (WFA_THREADS * per_thread);
    return b < 1 ? 1 : (b > WFA_MAX_BLOCKS ? WFA_MAX_BLOCKS : b);
}

__device__ __forceinline__ float wfa_val_at(const void* v, int vdt, int64_t i) {
    // vdt 2 = f32, 5 = bf16 (u16), 6 = bf16 packed in the low 16 bits of the
    // SORTED u32 slot array (value-in-key: the fold reads values
    // sequentially instead of gathering one cache line per tuple);
    // 7 = f32 carried through the sort as a second payload (also read
    // sequentially at the SORTED position); i64 inputs pre-cast by group()
    if (vdt == 7) return ((const float*)v)[i];
    if (vdt == 6) {
        union { uint32_t u; float f; } c;
        c.u = ((const uint32_t*)v)[i] << 16;
        return c.f;
    }
    if (vdt == 5) {
        uint32_t u = ((const uint16_t*)v)[i];
        union { uint32_t u; float f; } c;
        c.u = u << 16;
        return c.f;
    }
    return ((const float*)v)[i];
}

__device__ __forceinline__ uint64_t mix64s(uint64_t k) {
    k += 0x9e3779b97f4a7c15ULL;
    k = (k ^ (k >> 30)) * 0xbf58476d1ce4e5b9ULL;
    k = (k ^ (k >> 27)) * 0x94d049bb133111ebULL;
    return k ^ (k >> 31);
}

// ===== key -> dense slot (open addressing, device-scope atomics) =====
#define WFA_EMPTY_KEY (~0ULL)

// Entries are PACKED 16 B (key at [2p], slot at [2p+1]) so the common
// first-probe hit touches ONE cache line instead of two (tkeys+tslots
// halves split across arrays cost ~2x the L2 traffic per lookup).
__global__ void k_key_to_slot(const uint64_t* key, int64_t n, uint64_t* tab,
                              uint32_t* n_slots, int64_t cap,
                              uint32_t* slot_out, uint64_t* slot_to_key) {
    const uint64_t mask = (uint64_t)cap - 1;  // cap is a power of two
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (int64_t)blockDim.x) {
        uint64_t k = key[i];
        uint64_t p = mix64s(k) & mask;
        // (A/B-measured: a plain 16 B vector-load fast path with atomic
        // fallback changed nothing — the probe is line-fetch/latency bound,
        // not atomic bound — so the simple atomic protocol stays.)
        for (;;) {
            uint64_t cur = __hip_atomic_load(&tab[2 * p], __ATOMIC_RELAXED,
                                             __HIP_MEMORY_SCOPE_AGENT);
            if (cur == k) break;
            if (cur == WFA_EMPTY_KEY) {
                uint64_t expected = WFA_EMPTY_KEY;
                bool won = __hip_atomic_compare_exchange_strong(
                    &tab[2 * p], &expected, k, __ATOMIC_RELAXED, __ATOMIC_RELAXED,
                    __HIP_MEMORY_SCOPE_AGENT);
                if (won) {  // we inserted
                    uint32_t slot = atomicAdd(n_slots, 1u);
                    slot_to_key[slot] = k;
                    __hip_atomic_store(&tab[2 * p + 1], (uint64_t)slot,
                                       __ATOMIC_RELAXED,
                                       __HIP_MEMORY_SCOPE_AGENT);
                    break;
                }
                if (expected == k) break;  // raced: same key inserted by other
                // raced: different key took the cell — keep probing
            }
            p = (p + 1) & mask;
        }
        // wait for the slot id to be published (same 16 B line as the key)
        uint64_t s;
        do {
            s = __hip_atomic_load(&tab[2 * p + 1], __ATOMIC_RELAXED,
                                  __HIP_MEMORY_SCOPE_AGENT);
        } while (s == ~0ULL);
        slot_out[i] = (uint32_t)s;
    }
}

// value-in-key variant: slot_out[i] = slot << 16 | bf16(value).  Only used
// when the value column is already bf16 (no precision change) and
// max_keys <= 65535; the radix passes then sort on bits 16.. only.
__global__ void k_key_to_slot_v(const uint64_t* key, int64_t n, uint64_t* tab,
                                uint32_t* n_slots, int64_t cap,
                                uint32_t* slot_out, uint64_t* slot_to_key,
                                const uint16_t* val) {
    const uint64_t mask = (uint64_t)cap - 1;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (int64_t)blockDim.x) {
        uint64_t k = key[i];
        uint64_t p = mix64s(k) & mask;
        for (;;) {
            uint64_t cur = __hip_atomic_load(&tab[2 * p], __ATOMIC_RELAXED,
                                             __HIP_MEMORY_SCOPE_AGENT);
            if (cur == k) break;
            if (cur == WFA_EMPTY_KEY) {
                uint64_t expected = WFA_EMPTY_KEY;
                bool won = __hip_atomic_compare_exchange_strong(
                    &tab[2 * p], &expected, k, __ATOMIC_RELAXED, __ATOMIC_RELAXED,
                    __HIP_MEMORY_SCOPE_AGENT);
                if (won) {
                    uint32_t slot = atomicAdd(n_slots, 1u);
                    slot_to_key[slot] = k;
                    __hip_atomic_store(&tab[2 * p + 1], (uint64_t)slot,
                                       __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
                    break;
                }
                if (expected == k) break;
            }
            p = (p + 1) & mask;
        }
        uint64_t sl;
        do {
            sl = __hip_atomic_load(&tab[2 * p + 1], __ATOMIC_RELAXED,
                                   __HIP_MEMORY_SCOPE_AGENT);
        } while (sl == ~0ULL);
        slot_out[i] = ((uint32_t)sl << 16) | (uint32_t)val[i];
    }
}

extern "C" void wfa_key_to_slot_v(wfa_stream_t s, const uint64_t* key, int64_t n,
                                  uint64_t* table_packed, uint32_t* n_slots,
                                  int64_t table_cap, uint32_t* slot_out,
                                  uint64_t* slot_to_key, const uint16_t* val) {
    hipLaunchKernelGGL(k_key_to_slot_v, dim3(nblk(n)), dim3(WFA_THREADS), 0,
                       (hipStream_t)s, key, n, table_packed, n_slots, table_cap,
                       slot_out, slot_to_key, val);
}

// Dense-key fast path (user-asserted integer keys < max_keys): slot = key,
// no hash probe (the probe is ~25% of the flagship chain).  slot_to_key is
// identity (prefilled); n_slots tracked as max(key)+1; keys out of range
// raise the overflow flag (host checks and fails loudly).
__global__ void k_key_dense(const uint64_t* key, int64_t n, int64_t max_keys,
                            uint32_t* slot_out, uint32_t* n_slots,
                            uint32_t* overflow, const uint16_t* val) {
    __shared__ uint32_t blk_max;
    if (threadIdx.x == 0) blk_max = 0;
    __syncthreads();
    uint32_t my_max = 0;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (int64_t)blockDim.x) {
        uint64_t k = key[i];
        if (k >= (uint64_t)max_keys) {
            atomicOr(overflow, 1u);
            k = 0;
        }
        uint32_t sl = (uint32_t)k;
        slot_out[i] = val ? ((sl << 16) | (uint32_t)val[i]) : sl;
        if (sl + 1 > my_max) my_max = sl + 1;
    }
    for (int o = 32; o; o >>= 1)
        my_max = max(my_max, (uint32_t)__shfl_down(my_max, o, 64));
    if ((threadIdx.x & 63) == 0) atomicMax(&blk_max, my_max);
    __syncthreads();
    if (threadIdx.x == 0) atomicMax(n_slots, blk_max);
}


// hashed key->slot probe FUSED with the sort's pass-0 histogram (same
// trick as k_key_dense_h; the probe is latency-bound, the LDS histogram
// rides along for free and the sort skips one full read of slot_out)
__global__ void k_key_to_slot_h(const uint64_t* key, int64_t n, uint64_t* tab,
                                uint32_t* n_slots, int64_t cap,
                                uint32_t* slot_out, uint64_t* slot_to_key,
                                const uint16_t* val, uint32_t* hist,
                                int64_t nblocks, int shift) {
    __shared__ uint32_t h[WFA_THREADS / 64][256];
    const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    for (int d = threadIdx.x; d < 256; d += blockDim.x)
        for (int w = 0; w < WFA_THREADS / 64; ++w) h[w][d] = 0;
    __syncthreads();
    const uint64_t mask = (uint64_t)cap - 1;
    int64_t waveBase = (int64_t)blockIdx.x * RS8_PER_BLOCK + wave * RS8_PER_WAVE;
    for (int j = 0; j < RS8_IPT; ++j) {
        int64_t i = waveBase + j * 64 + lane;
        if (i >= n) continue;
        uint64_t k = key[i];
        uint64_t p = mix64s(k) & mask;
        for (;;) {
            uint64_t cur = __hip_atomic_load(&tab[2 * p], __ATOMIC_RELAXED,
                                             __HIP_MEMORY_SCOPE_AGENT);
            if (cur == k) break;
            if (cur == WFA_EMPTY_KEY) {
                uint64_t expected = WFA_EMPTY_KEY;
                bool won = __hip_atomic_compare_exchange_strong(
                    &tab[2 * p], &expected, k, __ATOMIC_RELAXED,
                    __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
                if (won) {
                    uint32_t slot = atomicAdd(n_slots, 1u);
                    slot_to_key[slot] = k;
                    __hip_atomic_store(&tab[2 * p + 1], (uint64_t)slot,
                                       __ATOMIC_RELAXED,
                                       __HIP_MEMORY_SCOPE_AGENT);
                    break;
                }
                if (expected == k) break;
            }
            p = (p + 1) & mask;
        }
        uint64_t sl;
        do {
            sl = __hip_atomic_load(&tab[2 * p + 1], __ATOMIC_RELAXED,
                                   __HIP_MEMORY_SCOPE_AGENT);
        } while (sl == ~0ULL);
        uint32_t packed = val ? (((uint32_t)sl << 16) | (uint32_t)val[i])
                              : (uint32_t)sl;
        slot_out[i] = packed;
        atomicAdd(&h[wave][(packed >> shift) & 255], 1u);
    }
    __syncthreads();
    for (int d = threadIdx.x; d < 256; d += blockDim.x) {
        uint32_t t = 0;
        for (int w = 0; w < WFA_THREADS / 64; ++w) t += h[w][d];
        hist[(int64_t)d * nblocks + blockIdx.x] = t;
    }
}

extern "C" void wfa_key_to_slot_h(wfa_stream_t s, const uint64_t* key,
                                  int64_t n, uint64_t* table_packed,
                                  uint32_t* n_slots, int64_t table_cap,
                                  uint32_t* slot_out, uint64_t* slot_to_key,
                                  const uint16_t* val, uint32_t* hist,
                                  int shift) {
    int64_t nb = (n + RS8_PER_BLOCK - 1) / RS8_PER_BLOCK;
    hipLaunchKernelGGL(k_key_to_slot_h, dim3(nb), dim3(WFA_THREADS), 0,
                       (hipStream_t)s, key, n, table_packed, n_slots,
                       table_cap, slot_out, slot_to_key, val, hist, nb, shift);
}

// keyby-exchange dest bucketing FUSED with the dest-sort's pass-0
// histogram (same pattern): dest = splitmix64(key) % world, and the
// single 8-bit dest-sort pass then skips its count read entirely.
// Runs on the exchange's META stream, overlapped with the previous
// batch's sort/gather (hist ping-pongs with the dest array).
__global__ void k_bucket_h(const uint64_t* key, int64_t n, int world,
                           uint32_t* dest, uint32_t* hist, int64_t nblocks) {
    __shared__ uint32_t h[WFA_THREADS / 64][256];
    const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    for (int d = threadIdx.x; d < 256; d += blockDim.x)
        for (int w = 0; w < WFA_THREADS / 64; ++w) h[w][d] = 0;
    __syncthreads();
    int64_t waveBase = (int64_t)blockIdx.x * RS8_PER_BLOCK + wave * RS8_PER_WAVE;
#pragma unroll
    for (int j = 0; j < RS8_IPT; ++j) {
        int64_t i = waveBase + j * 64 + lane;
        if (i >= n) continue;
        uint32_t d = (uint32_t)(mix64s(key[i]) % (uint64_t)world);
        dest[i] = d;
        atomicAdd(&h[wave][d], 1u);
    }
    __syncthreads();
    for (int d = threadIdx.x; d < 256; d += blockDim.x) {
        uint32_t t = 0;
        for (int w = 0; w < WFA_THREADS / 64; ++w) t += h[w][d];
        hist[(int64_t)d * nblocks + blockIdx.x] = t;
    }
}

extern "C" void wfa_bucket_by_key_h(wfa_stream_t s, const uint64_t* key,
                                    int64_t n, int world, uint32_t* dest,
                                    uint32_t* hist) {
    int64_t nb = (n + RS8_PER_BLOCK - 1) / RS8_PER_BLOCK;
    hipLaunchKernelGGL(k_bucket_h, dim3(nb), dim3(WFA_THREADS), 0,
                       (hipStream_t)s, key, n, world, dest, hist, nb);
}

// dense key->slot FUSED with the radix sort's pass-0 per-block histogram:
// the sort's first count pass re-reads the whole slot array; since this
// kernel just wrote it, accumulate the same LDS histogram here (sort
// blocking: RS8_PER_BLOCK elements per block, hist[d*nblocks + b]) and
// let the sort skip its pass-0 hist kernel (one 4n-byte read saved).
__global__ void k_key_dense_h(const uint64_t* key, int64_t n, int64_t max_keys,
                              uint32_t* slot_out, uint32_t* n_slots,
                              uint32_t* overflow, const uint16_t* val,
                              uint32_t* hist, int64_t nblocks, int shift) {
    __shared__ uint32_t h[WFA_THREADS / 64][256];
    __shared__ uint32_t blk_max;
    const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    if (threadIdx.x == 0) blk_max = 0;
    for (int d = threadIdx.x; d < 256; d += blockDim.x)
        for (int w = 0; w < WFA_THREADS / 64; ++w) h[w][d] = 0;
    __syncthreads();
    uint32_t my_max = 0;
    int64_t waveBase = (int64_t)blockIdx.x * RS8_PER_BLOCK + wave * RS8_PER_WAVE;
#pragma unroll
    for (int j = 0; j < RS8_IPT; ++j) {
        int64_t i = waveBase + j * 64 + lane;
        if (i >= n) continue;
        uint64_t k = key[i];
        if (k >= (uint64_t)max_keys) {
            atomicOr(overflow, 1u);
            k = 0;
        }
        uint32_t sl = (uint32_t)k;
        uint32_t packed = val ? ((sl << 16) | (uint32_t)val[i]) : sl;
        slot_out[i] = packed;
        atomicAdd(&h[wave][(packed >> shift) & 255], 1u);
        if (sl + 1 > my_max) my_max = sl + 1;
    }
    for (int o = 32; o; o >>= 1)
        my_max = max(my_max, (uint32_t)__shfl_down(my_max, o, 64));
    if (lane == 0) atomicMax(&blk_max, my_max);
    __syncthreads();
    for (int d = threadIdx.x; d < 256; d += blockDim.x) {
        uint32_t t = 0;
        for (int w = 0; w < WFA_THREADS / 64; ++w) t += h[w][d];
        hist[(int64_t)d * nblocks + blockIdx.x] = t;
    }
    if (threadIdx.x == 0) atomicMax(n_slots, blk_max);
}

extern "C" void wfa_key_dense_h(wfa_stream_t s, const uint64_t* key, int64_t n,
                                int64_t max_keys, uint32_t* slot_out,
                                uint32_t* n_slots, uint32_t* overflow,
                                const uint16_t* val, uint32_t* hist,
                                int shift) {
    int64_t nb = (n + RS8_PER_BLOCK - 1) / RS8_PER_BLOCK;
    hipLaunchKernelGGL(k_key_dense_h, dim3(nb), dim3(WFA_THREADS), 0,
                       (hipStream_t)s, key, n, max_keys, slot_out, n_slots,
                       overflow, val, hist, nb, shift);
}

__global__ void k_iota_u64(uint64_t* p, int64_t n) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (int64_t)blockDim.x)
        p[i] = (uint64_t)i;
}

extern "C" void wfa_key_dense(wfa_stream_t s, const uint64_t* key, int64_t n,
                              int64_t max_keys, uint32_t* slot_out,
                              uint32_t* n_slots, uint32_t* overflow,
                              const uint16_t* val) {
    hipLaunchKernelGGL(k_key_dense, dim3(nblk(n)), dim3(WFA_THREADS), 0,
                       (hipStream_t)s, key, n, max_keys, slot_out, n_slots,
                       overflow, val);
}

extern "C" void wfa_iota_u64(wfa_stream_t s, uint64_t* p, int64_t n) {
    hipLaunchKernelGGL(k_iota_u64, dim3(nblk(n)), dim3(WFA_THREADS), 0,
                       (hipStream_t)s, p, n);
}

extern "C" void wfa_key_to_slot(wfa_stream_t s, const uint64_t* key, int64_t n,
                                uint64_t* table_packed, uint32_t* n_slots,
                                int64_t table_cap, uint32_t* slot_out,
                                uint64_t* slot_to_key) {
    hipLaunchKernelGGL(k_key_to_slot, dim3(nblk(n)), dim3(WFA_THREADS), 0,
                       (hipStream_t)s, key, n, table_packed, n_slots,
                       table_cap, slot_out, slot_to_key);
}

// ===== stable LSD radix sort, 4-bit digits =====
extern "C" int64_t wfa_sort_nblocks(int64_t n) {
    return (n + RS_PER_BLOCK - 1) / RS_PER_BLOCK;
}

// scratch requirement (u32 elements) covering the 4-bit and 8-bit paths
extern "C" int64_t wfa_sort_hist_u32(int64_t cap) {
    int64_t nb4 = (cap + RS_PER_BLOCK - 1) / RS_PER_BLOCK;
    int64_t nb8 = (cap + RS8_PER_BLOCK - 1) / RS8_PER_BLOCK;
    int64_t a = 16 * nb4 + 16;
    int64_t b = 256 * nb8 + 512;
    return (a > b ? a : b) + 64;
}

__global__ void k_rs_hist(const uint32_t* keys, int64_t n, int shift, uint32_t* hist,
                          int64_t nblocks) {
    __shared__ uint32_t h[16];
    if (threadIdx.x < 16) h[threadIdx.x] = 0;
    __syncthreads();
    int64_t base = (int64_t)blockIdx.x * RS_PER_BLOCK + (int64_t)threadIdx.x * RS_IPT;
#pragma unroll
    for (int j = 0; j < RS_IPT; ++j) {
        int64_t i = base + j;
        if (i < n) atomicAdd(&h[(keys[i] >> shift) & 15], 1u);
    }
    __syncthreads();
    if (threadIdx.x < 16) hist[(int64_t)threadIdx.x * nblocks + blockIdx.x] = h[threadIdx.x];
}

// per-digit exclusive scan: 16 independent blocks, block d scans its own
// hist[d*nblocks .. d*nblocks+nblocks) and records the digit total; the
// scatter kernel turns digit totals into global digit bases with a
// 16-element prefix.  Replaces one serial global scan of 16*nblocks.
__global__ void k_rs_scan16(uint32_t* hist, int64_t nblocks, uint32_t* dt) {
    uint32_t* a = hist + (int64_t)blockIdx.x * nblocks;
    __shared__ uint32_t tot[256];
    const int64_t chunk = (nblocks + 255) / 256;
    const int64_t b0 = (int64_t)threadIdx.x * chunk;
    const int64_t b1 = min(nblocks, b0 + chunk);
    uint32_t s = 0;
    for (int64_t i = b0; i < b1; ++i) s += a[i];
    tot[threadIdx.x] = s;
    __syncthreads();
    for (int off = 1; off < 256; off <<= 1) {
        uint32_t t = (threadIdx.x >= off) ? tot[threadIdx.x - off] : 0;
        __syncthreads();
        tot[threadIdx.x] += t;
        __syncthreads();
    }
    uint32_t run = tot[threadIdx.x] - s;
    for (int64_t i = b0; i < b1; ++i) {
        uint32_t v = a[i];
        a[i] = run;
        run += v;
    }
    if (threadIdx.x == 255) dt[blockIdx.x] = tot[255];
}

// single-block exclusive scan: each thread owns a contiguous chunk
// (sequential sum -> 1024-wide block scan -> sequential rewrite).
__global__ void k_rs_scan(uint32_t* a, int64_t m) {  // single block, 1024 thr
    __shared__ uint32_t tot[1024];
    const int64_t chunk = (m + 1023) / 1024;
    const int64_t b0 = (int64_t)threadIdx.x * chunk;
    const int64_t b1 = min(m, b0 + chunk);
    uint32_t s = 0;
    for (int64_t i = b0; i < b1; ++i) s += a[i];
    tot[threadIdx.x] = s;
    __syncthreads();
    for (int off = 1; off < 1024; off <<= 1) {
        uint32_t t = (threadIdx.x >= off) ? tot[threadIdx.x - off] : 0;
        __syncthreads();
        tot[threadIdx.x] += t;
        __syncthreads();
    }
    uint32_t run = tot[threadIdx.x] - s;  // exclusive base for this chunk
    for (int64_t i = b0; i < b1; ++i) {
        uint32_t v = a[i];
        a[i] = run;
        run += v;
    }
}

// Stable scatter: per-thread contiguous items; LDS vector-scan of 16-digit
// count vectors across the block's threads.
__global__ void k_rs_scatter(const uint32_t* keys, const uint32_t* vals, int64_t n,
                             int shift, const uint32_t* hist, int64_t nblocks,
                             const uint32_t* dt, uint32_t* keys_out,
                             uint32_t* vals_out, const uint32_t* vals2,
                             uint32_t* vals2_out) {
    __shared__ uint32_t tc[WFA_THREADS][17];  // [thread][digit] padded
    __shared__ uint32_t gbase[16];
    __shared__ uint32_t dbase[16];
    if (threadIdx.x == 0) {
        uint32_t run = 0;
        for (int d = 0; d < 16; ++d) {
            dbase[d] = run;
            run += dt[d];
        }
    }
    int64_t base = (int64_t)blockIdx.x * RS_PER_BLOCK + (int64_t)threadIdx.x * RS_IPT;
    uint32_t k_[RS_IPT];
    uint32_t v_[RS_IPT];
    uint32_t v2_[RS_IPT];
    int cnt_local[16];
#pragma unroll
    for (int d = 0; d < 16; ++d) cnt_local[d] = 0;
    int nit = 0;
#pragma unroll
    for (int j = 0; j < RS_IPT; ++j) {
        int64_t i = base + j;
        if (i < n) {
            k_[j] = keys[i];
            v_[j] = vals ? vals[i] : (uint32_t)i;  // null = implicit iota
            if (vals2) v2_[j] = vals2[i];
            cnt_local[(k_[j] >> shift) & 15]++;
            nit = j + 1;
        }
    }
#pragma unroll
    for (int d = 0; d < 16; ++d) tc[threadIdx.x][d] = cnt_local[d];
    if (threadIdx.x < 16)
        gbase[threadIdx.x] = hist[(int64_t)threadIdx.x * nblocks + blockIdx.x];
    __syncthreads();
    if (threadIdx.x < 16) gbase[threadIdx.x] += dbase[threadIdx.x];
    __syncthreads();
    // Hillis-Steele inclusive scan over threads of the 16-vector
    for (int off = 1; off < WFA_THREADS; off <<= 1) {
        uint32_t t[16];
#pragma unroll
        for (int d = 0; d < 16; ++d)
            t[d] = (threadIdx.x >= off) ? tc[threadIdx.x - off][d] : 0;
        __syncthreads();
#pragma unroll
        for (int d = 0; d < 16; ++d) tc[threadIdx.x][d] += t[d];
        __syncthreads();
    }
    // exclusive prefix for this thread = inclusive - own count
    uint32_t pre[16];
#pragma unroll
    for (int d = 0; d < 16; ++d) pre[d] = tc[threadIdx.x][d] - cnt_local[d];
    // scatter in order; rank among own items recomputed (static indexing)
    for (int j = 0; j < nit; ++j) {
        uint32_t d = (k_[j] >> shift) & 15;
        uint32_t own = 0;
        for (int j2 = 0; j2 < j; ++j2) own += (((k_[j2] >> shift) & 15) == d);
        uint32_t pos = gbase[d] + pre[d] + own;
        keys_out[pos] = k_[j];
        vals_out[pos] = v_[j];
        if (vals2) vals2_out[pos] = v2_[j];
    }
}

// ===== 8-bit-digit radix pass (halves the pass count for slot sorts) =====
// Item order inside a block: per-wave contiguous ranges, lane-strided
// rounds (waveBase + j*64 + lane) — coalesced loads AND a stable rank
// order (wave w's range precedes wave w+1's; within a wave, round j
// precedes j+1; within a round, lane order).  Per-item stable rank =
// LDS per-wave digit counter before this round + lane rank inside the
// round's same-digit ballot group.

extern "C" int64_t wfa_sort8_nblocks(int64_t n) {
    return (n + RS8_PER_BLOCK - 1) / RS8_PER_BLOCK;
}

__global__ void k_rs8_hist(const uint32_t* keys, int64_t n, int shift,
                           uint32_t* hist, int64_t nblocks) {
    // one LDS histogram copy per wave: hot digits contend within 64 lanes
    // instead of 256 threads
    __shared__ uint32_t h[WFA_THREADS / 64][256];
    const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    for (int d = threadIdx.x; d < 256; d += blockDim.x)
        for (int w = 0; w < WFA_THREADS / 64; ++w) h[w][d] = 0;
    __syncthreads();
    int64_t waveBase = (int64_t)blockIdx.x * RS8_PER_BLOCK + wave * RS8_PER_WAVE;
#pragma unroll
    for (int j = 0; j < RS8_IPT; ++j) {
        int64_t i = waveBase + j * 64 + lane;
        if (i < n) atomicAdd(&h[wave][(keys[i] >> shift) & 255], 1u);
    }
    __syncthreads();
    for (int d = threadIdx.x; d < 256; d += blockDim.x) {
        uint32_t t = 0;
        for (int w = 0; w < WFA_THREADS / 64; ++w) t += h[w][d];
        hist[(int64_t)d * nblocks + blockIdx.x] = t;
    }
}

// per-digit scan over blocks (256 blocks, one digit each) + digit totals
__global__ void k_rs8_scan(uint32_t* hist, int64_t nblocks, uint32_t* dt) {
    uint32_t* a = hist + (int64_t)blockIdx.x * nblocks;
    __shared__ uint32_t tot[256];
    const int64_t chunk = (nblocks + 255) / 256;
    const int64_t b0 = (int64_t)threadIdx.x * chunk;
    const int64_t b1 = min(nblocks, b0 + chunk);
    uint32_t s = 0;
    for (int64_t i = b0; i < b1; ++i) s += a[i];
    tot[threadIdx.x] = s;
    __syncthreads();
    for (int off = 1; off < 256; off <<= 1) {
        uint32_t t = (threadIdx.x >= off) ? tot[threadIdx.x - off] : 0;
        __syncthreads();
        tot[threadIdx.x] += t;
        __syncthreads();
    }
    uint32_t run = tot[threadIdx.x] - s;
    for (int64_t i = b0; i < b1; ++i) {
        uint32_t v = a[i];
        a[i] = run;
        run += v;
    }
    if (threadIdx.x == 255) dt[blockIdx.x] = tot[255];
}

__global__ void k_rs8_dbase(const uint32_t* dt, uint32_t* dbase) {  // 1 block
    __shared__ uint32_t tot[256];
    tot[threadIdx.x] = dt[threadIdx.x];
    __syncthreads();
    for (int off = 1; off < 256; off <<= 1) {
        uint32_t t = (threadIdx.x >= off) ? tot[threadIdx.x - off] : 0;
        __syncthreads();
        tot[threadIdx.x] += t;
        __syncthreads();
    }
    dbase[threadIdx.x] = tot[threadIdx.x] - dt[threadIdx.x];
}

__global__ void k_rs8_scatter(const uint32_t* keys, const uint32_t* vals,
                              int64_t n, int shift, const uint32_t* hist,
                              int64_t nblocks, const uint32_t* dbase,
                              uint32_t* keys_out, uint32_t* vals_out,
                              const uint32_t* vals2, uint32_t* vals2_out) {
    __shared__ uint32_t gbase[256];                    // digit base, this block
    __shared__ uint32_t waveCnt[WFA_THREADS / 64][256];
    __shared__ uint32_t wavePre[WFA_THREADS / 64][256];
    const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    const uint64_t lt = ((uint64_t)1 << lane) - 1;
    for (int d = threadIdx.x; d < 256; d += blockDim.x) {
        gbase[d] = hist[(int64_t)d * nblocks + blockIdx.x] + dbase[d];
        for (int w = 0; w < WFA_THREADS / 64; ++w) waveCnt[w][d] = 0;
    }
    __syncthreads();
    int64_t waveBase = (int64_t)blockIdx.x * RS8_PER_BLOCK + wave * RS8_PER_WAVE;
    uint32_t mk[RS8_IPT], mv[RS8_IPT], mr[RS8_IPT];  // key, val, (digit<<24|rank)
    uint32_t mv2[RS8_IPT];
    int nit = 0;
#pragma unroll
    for (int j = 0; j < RS8_IPT; ++j) {
        int64_t i = waveBase + j * 64 + lane;
        bool valid = i < n;
        uint32_t k = valid ? keys[i] : 0;
        uint32_t v = valid ? (vals ? vals[i] : (uint32_t)i) : 0;
        uint32_t d = (k >> shift) & 255;
        // same-digit ballot group among valid lanes of this round
        uint64_t mask = __ballot(valid);
#pragma unroll
        for (int bit = 0; bit < 8; ++bit) {
            uint64_t b = __ballot((d >> bit) & 1);
            mask &= ((d >> bit) & 1) ? b : ~b;
        }
        if (valid) {
            uint32_t pre = waveCnt[wave][d];
            uint32_t rank = pre + (uint32_t)__popcll(mask & lt);
            mk[j] = k;
            mv[j] = v;
            if (vals2) mv2[j] = vals2[i];
            mr[j] = (d << 24) | (rank & 0xFFFFFF);
            // group leader bumps the wave's digit counter
            if ((mask & lt) == 0) waveCnt[wave][d] = pre + (uint32_t)__popcll(mask);
            nit = j + 1;
        }
    }
    __syncthreads();
    // prefix of wave digit counts across waves
    for (int d = threadIdx.x; d < 256; d += blockDim.x) {
        uint32_t run = 0;
        for (int w = 0; w < WFA_THREADS / 64; ++w) {
            wavePre[w][d] = run;
            run += waveCnt[w][d];
        }
    }
    __syncthreads();
    for (int j = 0; j < nit; ++j) {
        uint32_t d = mr[j] >> 24;
        uint32_t pos = gbase[d] + wavePre[wave][d] + (mr[j] & 0xFFFFFF);
        keys_out[pos] = mk[j];
        vals_out[pos] = mv[j];
        if (vals2) vals2_out[pos] = mv2[j];
    }
}

// LDS-staged variant: items are reordered in LDS (grouped by digit) before
// the global write, so each wave stores contiguous runs per digit instead
// of 4 B scattered words.  Used when there is no second payload (the hot
// keyed-operator path).
template <bool CARRY2>
__global__ void k_rs8_scatter_lds(const uint32_t* keys, const uint32_t* vals,
                                  int64_t n, int shift, const uint32_t* hist,
                                  int64_t nblocks, const uint32_t* dbase,
                                  uint32_t* keys_out, uint32_t* vals_out,
                                  const uint32_t* vals2, uint32_t* vals2_out) {
    __shared__ uint32_t gbase[256];
    __shared__ uint32_t waveCnt[WFA_THREADS / 64][256];
    __shared__ uint32_t wavePre[WFA_THREADS / 64][256];
    __shared__ uint32_t localBase[256];
    __shared__ uint32_t sk[RS8_PER_BLOCK];
    __shared__ uint32_t sv[RS8_PER_BLOCK];
    // second payload LDS only exists in the CARRY2 instantiation (a
    // runtime-null variant still cost 16 KB static LDS = occupancy)
    __shared__ uint32_t sv2[CARRY2 ? RS8_PER_BLOCK : 1];
    __shared__ uint8_t sd[RS8_PER_BLOCK];
    const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    const uint64_t lt = ((uint64_t)1 << lane) - 1;
    for (int d = threadIdx.x; d < 256; d += blockDim.x) {
        gbase[d] = hist[(int64_t)d * nblocks + blockIdx.x] + dbase[d];
        for (int w = 0; w < WFA_THREADS / 64; ++w) waveCnt[w][d] = 0;
    }
    __syncthreads();
    int64_t blockStart = (int64_t)blockIdx.x * RS8_PER_BLOCK;
    int64_t waveBase = blockStart + wave * RS8_PER_WAVE;
    uint32_t mk[RS8_IPT], mv[RS8_IPT], mv2[RS8_IPT], mr[RS8_IPT];
    int nit = 0;
#pragma unroll
    for (int j = 0; j < RS8_IPT; ++j) {
        int64_t i = waveBase + j * 64 + lane;
        bool valid = i < n;
        uint32_t k = valid ? keys[i] : 0;
        uint32_t v = valid ? (vals ? vals[i] : (uint32_t)i) : 0;
        uint32_t d = (k >> shift) & 255;
        uint64_t mask = __ballot(valid);
#pragma unroll
        for (int bit = 0; bit < 8; ++bit) {
            uint64_t b = __ballot((d >> bit) & 1);
            mask &= ((d >> bit) & 1) ? b : ~b;
        }
        if (valid) {
            uint32_t pre = waveCnt[wave][d];
            uint32_t rank = pre + (uint32_t)__popcll(mask & lt);
            mk[j] = k;
            mv[j] = v;
            if (CARRY2) mv2[j] = vals2[i];
            mr[j] = (d << 24) | (rank & 0xFFFFFF);
            if ((mask & lt) == 0) waveCnt[wave][d] = pre + (uint32_t)__popcll(mask);
            nit = j + 1;
        }
    }
    __syncthreads();
    // wave prefix + block-local digit bases (layout grouped by digit)
    __shared__ uint32_t blockCnt[256];
    for (int d = threadIdx.x; d < 256; d += blockDim.x) {
        uint32_t run = 0;
        for (int w = 0; w < WFA_THREADS / 64; ++w) {
            wavePre[w][d] = run;
            run += waveCnt[w][d];
        }
        blockCnt[d] = run;
    }
    __syncthreads();
    if (threadIdx.x < 256) {  // exclusive prefix over digits (256 threads)
        __shared__ uint32_t tot[256];
        uint32_t c = blockCnt[threadIdx.x];
        tot[threadIdx.x] = c;
        __syncthreads();
        for (int off = 1; off < 256; off <<= 1) {
            uint32_t t = (threadIdx.x >= off) ? tot[threadIdx.x - off] : 0;
            __syncthreads();
            tot[threadIdx.x] += t;
            __syncthreads();
        }
        localBase[threadIdx.x] = tot[threadIdx.x] - c;
    }
    __syncthreads();
    for (int j = 0; j < nit; ++j) {
        uint32_t d = mr[j] >> 24;
        uint32_t lpos = localBase[d] + wavePre[wave][d] + (mr[j] & 0xFFFFFF);
        sk[lpos] = mk[j];
        sv[lpos] = mv[j];
        if (CARRY2) sv2[lpos] = mv2[j];
        sd[lpos] = (uint8_t)d;
    }
    __syncthreads();
    const int nblk_items = (int)min((int64_t)RS8_PER_BLOCK, n - blockStart);
    for (int p = threadIdx.x; p < nblk_items; p += blockDim.x) {
        uint32_t d = sd[p];
        uint32_t g = gbase[d] + (uint32_t)p - localBase[d];
        keys_out[g] = sk[p];
        vals_out[g] = sv[p];
        if (CARRY2) vals2_out[g] = sv2[p];
    }
}

static void sort_pairs2_impl(wfa_stream_t s, uint32_t* slot, uint32_t* idx,
                             uint32_t* slot_tmp, uint32_t* idx_tmp,
                             uint32_t* val2, uint32_t* val2_tmp,
                             uint32_t* hist, int64_t n, int bits,
                             uint32_t** out_slot, uint32_t** out_idx,
                             uint32_t** out_val2, int implicit_iota,
                             int base_shift, int skip_hist0) {
    hipStream_t st = (hipStream_t)s;
    uint32_t *ka = slot, *va = idx, *kb = slot_tmp, *vb = idx_tmp;
    uint32_t *wa = val2, *wb = val2_tmp;
    // implicit iota: the first pass reads no payload (vals==nullptr means
    // payload = global index) — saves the iota kernel + one 4n-byte read
    bool first = true;
    if (bits > 4) {
        // 8-bit digits: fewer passes; hist is sized for 16*nblocks4 + 16
        // which covers 256*nblocks8 + 512 (RS8_PER_BLOCK = 2*RS_PER_BLOCK)
        int64_t nb = wfa_sort8_nblocks(n);
        uint32_t* dt = hist + 256 * nb;
        uint32_t* dbase = dt + 256;
        int passes = (bits + 7) / 8;
        for (int p = 0; p < passes; ++p) {
            int shift = base_shift + 8 * p;
            if (!(p == 0 && skip_hist0)) {
                hipLaunchKernelGGL(k_rs8_hist, dim3(nb), dim3(WFA_THREADS), 0, st,
                                   ka, n, shift, hist, nb);
            }
            sdbg(st, "hist", p);
            hipLaunchKernelGGL(k_rs8_scan, dim3(256), dim3(256), 0, st, hist, nb, dt);
            sdbg(st, "scan", p);
            hipLaunchKernelGGL(k_rs8_dbase, dim3(1), dim3(256), 0, st, dt, dbase);
            sdbg(st, "dbase", p);
            uint32_t* va_eff = (first && implicit_iota) ? nullptr : va;
            if (val2)
                hipLaunchKernelGGL(k_rs8_scatter_lds<true>, dim3(nb),
                                   dim3(WFA_THREADS), 0, st, ka, va_eff, n, shift,
                                   hist, nb, dbase, kb, vb, wa, wb);
            else
                hipLaunchKernelGGL(k_rs8_scatter_lds<false>, dim3(nb),
                                   dim3(WFA_THREADS), 0, st, ka, va_eff, n, shift,
                                   hist, nb, dbase, kb, vb, wa, wb);
            sdbg(st, "scatter", p);
            first = false;
            uint32_t* t;
            t = ka; ka = kb; kb = t;
            t = va; va = vb; vb = t;
            t = wa; wa = wb; wb = t;
        }
        *out_slot = ka;
        *out_idx = va;
        if (out_val2) *out_val2 = wa;
        return;
    }
    int64_t nblocks = wfa_sort_nblocks(n);
    uint32_t* dt = hist + 16 * nblocks;  // hist is sized 16*nblocks + 16
    int passes = (bits + 3) / 4;
    (void)base_shift;  // 4-bit path is only used for tiny key spaces
    for (int p = 0; p < passes; ++p) {
        int shift = 4 * p;
        hipLaunchKernelGGL(k_rs_hist, dim3(nblocks), dim3(WFA_THREADS), 0, st, ka, n,
                           shift, hist, nblocks);
        hipLaunchKernelGGL(k_rs_scan16, dim3(16), dim3(256), 0, st, hist, nblocks, dt);
        hipLaunchKernelGGL(k_rs_scatter, dim3(nblocks), dim3(WFA_THREADS), 0, st, ka,
                           (first && implicit_iota) ? nullptr : va, n, shift, hist,
                           nblocks, dt, kb, vb, wa, wb);
        first = false;
        uint32_t* t;
        t = ka; ka = kb; kb = t;
        t = va; va = vb; vb = t;
        t = wa; wa = wb; wb = t;
    }
    *out_slot = ka;
    *out_idx = va;
    if (out_val2) *out_val2 = wa;
}

extern "C" void wfa_sort_pairs2(wfa_stream_t s, uint32_t* slot, uint32_t* idx,
                                uint32_t* slot_tmp, uint32_t* idx_tmp,
                                uint32_t* val2, uint32_t* val2_tmp,
                                uint32_t* hist, int64_t n, int bits,
                                uint32_t** out_slot, uint32_t** out_idx,
                                uint32_t** out_val2, int implicit_iota,
                                int base_shift) {
    sort_pairs2_impl(s, slot, idx, slot_tmp, idx_tmp, val2, val2_tmp, hist, n,
                     bits, out_slot, out_idx, out_val2, implicit_iota,
                     base_shift, 0);
}

// pass-0 histogram already produced by the slot-writing kernel
// (k_key_dense_h) with the same blocking — skip the first count pass
extern "C" void wfa_sort_pairs2_ph(wfa_stream_t s, uint32_t* slot, uint32_t* idx,
                                   uint32_t* slot_tmp, uint32_t* idx_tmp,
                                   uint32_t* val2, uint32_t* val2_tmp,
                                   uint32_t* hist, int64_t n, int bits,
                                   uint32_t** out_slot, uint32_t** out_idx,
                                   uint32_t** out_val2, int implicit_iota,
                                   int base_shift) {
    sort_pairs2_impl(s, slot, idx, slot_tmp, idx_tmp, val2, val2_tmp, hist, n,
                     bits, out_slot, out_idx, out_val2, implicit_iota,
                     base_shift, bits > 4 ? 1 : 0);
}

extern "C" void wfa_sort_pairs(wfa_stream_t s, uint32_t* slot, uint32_t* idx,
                               uint32_t* slot_tmp, uint32_t* idx_tmp, uint32_t* hist,
                               int64_t n, int bits, uint32_t** out_slot,
                               uint32_t** out_idx) {
    wfa_sort_pairs2(s, slot, idx, slot_tmp, idx_tmp, nullptr, nullptr, hist, n,
                    bits, out_slot, out_idx, nullptr, 0, 0);
}

// ===== segment extraction =====
__global__ void k_seg_count(const uint32_t* slot, int64_t n, uint32_t* blk_cnt,
                            int shr) {
    int64_t base = (int64_t)blockIdx.x * RS_PER_BLOCK + (int64_t)threadIdx.x * RS_IPT;
    uint32_t c = 0;
#pragma unroll
    for (int j = 0; j < RS_IPT; ++j) {
        int64_t i = base + j;
        if (i < n) c += (i == 0) || ((slot[i] >> shr) != (slot[i - 1] >> shr));
    }
    __shared__ uint32_t red[WFA_THREADS / 64];
    for (int off = 32; off; off >>= 1) c += __shfl_down(c, off, 64);
    if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = c;
    __syncthreads();
    if (threadIdx.x == 0) {
        uint32_t t = 0;
        for (int w = 0; w < WFA_THREADS / 64; ++w) t += red[w];
        blk_cnt[blockIdx.x] = t;
    }
}

__global__ void k_seg_scatter(const uint32_t* slot, int64_t n, const uint32_t* blk_off,
                              uint32_t* seg_start, uint32_t* seg_slot, int shr) {
    int64_t base = (int64_t)blockIdx.x * RS_PER_BLOCK + (int64_t)threadIdx.x * RS_IPT;
    uint32_t c = 0;
#pragma unroll
    for (int j = 0; j < RS_IPT; ++j) {
        int64_t i = base + j;
        if (i < n) c += (i == 0) || ((slot[i] >> shr) != (slot[i - 1] >> shr));
    }
    __shared__ uint32_t tc[WFA_THREADS];
    tc[threadIdx.x] = c;
    __syncthreads();
    for (int off = 1; off < WFA_THREADS; off <<= 1) {
        uint32_t t = (threadIdx.x >= off) ? tc[threadIdx.x - off] : 0;
        __syncthreads();
        tc[threadIdx.x] += t;
        __syncthreads();
    }
    uint32_t w = blk_off[blockIdx.x] + tc[threadIdx.x] - c;
    for (int j = 0; j < RS_IPT; ++j) {
        int64_t i = base + j;
        if (i >= n) continue;
        if ((i == 0) || ((slot[i] >> shr) != (slot[i - 1] >> shr))) {
            seg_start[w] = (uint32_t)i;
            seg_slot[w] = slot[i] >> shr;
            ++w;
        }
    }
}

__global__ void k_nseg_total(const uint32_t* slot_sorted, int64_t n,
                             const uint32_t* scan_tmp, int64_t nb, int64_t* d_nseg,
                             int shr) {
    // one wave recounts the last block's boundaries in parallel
    if (blockIdx.x == 0 && threadIdx.x < 64) {
        int64_t base = (nb - 1) * (int64_t)RS_PER_BLOCK;
        uint32_t c = 0;
        for (int64_t i = base + threadIdx.x; i < n; i += 64)
            c += (i == 0) || ((slot_sorted[i] >> shr) != (slot_sorted[i - 1] >> shr));
        for (int off = 32; off; off >>= 1) c += __shfl_down(c, off, 64);
        if (threadIdx.x == 0) *d_nseg = (int64_t)scan_tmp[nb - 1] + c;
    }
}

// Bounded-slot fast path: every slot has at most ONE boundary in the
// sorted sequence, so boundaries scatter directly into a per-slot table
// (one 4 B/elem read of the sorted slots, no second pass, no grid scan),
// then a single-workgroup ordered compaction over max_slots entries
// builds the dense (seg_start, seg_slot) lists.  Used when max_slots is
// small (<= 64 K: table fits L2); the general two-pass path below covers
// unbounded slot counts (e.g. all-unique keys).
__global__ void k_seg_by_slot(const uint32_t* slot, int64_t n, uint32_t* by_slot,
                              int shr) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (int64_t)blockDim.x) {
        uint32_t sl = slot[i] >> shr;
        if (i == 0 || (slot[i - 1] >> shr) != sl) by_slot[sl] = (uint32_t)i;
    }
}

// multi-block compaction: count per block-range -> micro-scan -> ordered
// write (a single-workgroup scan left 255 CUs idle for ~20 us/batch)
#define SEGC_BLOCKS 32
__global__ void k_seg_count_dense(const uint32_t* by_slot, int64_t max_slots,
                                  uint32_t* blk_tot) {
    const int64_t per = (max_slots + SEGC_BLOCKS - 1) / SEGC_BLOCKS;
    const int64_t lo = blockIdx.x * per;
    const int64_t hi = min(lo + per, max_slots);
    uint32_t c = 0;
    for (int64_t s = lo + threadIdx.x; s < hi; s += blockDim.x)
        c += by_slot[s] != 0xFFFFFFFFu;
    __shared__ uint32_t red[WFA_THREADS / 64];
    for (int off = 32; off; off >>= 1) c += __shfl_down(c, off, 64);
    if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = c;
    __syncthreads();
    if (threadIdx.x == 0) {
        uint32_t t = 0;
        for (int w = 0; w < (int)blockDim.x / 64; ++w) t += red[w];
        blk_tot[blockIdx.x] = t;
    }
}

__global__ void k_seg_scan_dense(uint32_t* blk_tot, int64_t* d_nseg) {
    if (threadIdx.x == 0) {
        uint32_t acc = 0;
        for (int b = 0; b < SEGC_BLOCKS; ++b) {
            uint32_t t = blk_tot[b];
            blk_tot[b] = acc;
            acc += t;
        }
        *d_nseg = (int64_t)acc;
    }
}

__global__ void k_seg_write_dense(const uint32_t* by_slot, int64_t max_slots,
                                  const uint32_t* blk_base, uint32_t* seg_start,
                                  uint32_t* seg_slot) {
    const int64_t per = (max_slots + SEGC_BLOCKS - 1) / SEGC_BLOCKS;
    const int64_t lo = blockIdx.x * per;
    const int64_t hi = min(lo + per, max_slots);
    __shared__ uint32_t wsum[WFA_THREADS / 64 + 1];
    __shared__ uint32_t base;
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    const int nw = (int)blockDim.x / 64;
    if (threadIdx.x == 0) base = blk_base[blockIdx.x];
    __syncthreads();
    for (int64_t c = lo; c < hi; c += blockDim.x) {
        int64_t s = c + threadIdx.x;
        uint32_t v = (s < hi) ? by_slot[s] : 0xFFFFFFFFu;
        bool pred = v != 0xFFFFFFFFu;
        uint64_t m = __ballot(pred);
        uint32_t before = __popcll(m & ((1ULL << lane) - 1));
        if (lane == 0) wsum[wid] = (uint32_t)__popcll(m);
        __syncthreads();
        if (threadIdx.x == 0) {
            uint32_t acc = 0;
            for (int w = 0; w < nw; ++w) {
                uint32_t t = wsum[w];
                wsum[w] = acc;
                acc += t;
            }
            wsum[nw] = acc;
        }
        __syncthreads();
        if (pred) {
            uint32_t w = base + wsum[wid] + before;
            seg_start[w] = v;
            seg_slot[w] = (uint32_t)s;
        }
        __syncthreads();
        if (threadIdx.x == 0) base += wsum[nw];
        __syncthreads();
    }
}

extern "C" void wfa_segments_dense(wfa_stream_t s, const uint32_t* slot_sorted,
                                   int64_t n, uint32_t* by_slot, int64_t max_slots,
                                   uint32_t* scratch32, uint32_t* seg_start,
                                   uint32_t* seg_slot, int64_t* d_nseg, int shr) {
    hipStream_t st = (hipStream_t)s;
    (void)hipMemsetAsync(by_slot, 0xFF, 4 * max_slots, st);
    hipLaunchKernelGGL(k_seg_by_slot, dim3(nblk(n)), dim3(WFA_THREADS), 0, st,
                       slot_sorted, n, by_slot, shr);
    hipLaunchKernelGGL(k_seg_count_dense, dim3(SEGC_BLOCKS), dim3(WFA_THREADS),
                       0, st, by_slot, max_slots, scratch32);
    hipLaunchKernelGGL(k_seg_scan_dense, dim3(1), dim3(64), 0, st, scratch32,
                       d_nseg);
    hipLaunchKernelGGL(k_seg_write_dense, dim3(SEGC_BLOCKS), dim3(WFA_THREADS),
                       0, st, by_slot, max_slots, scratch32, seg_start,
                       seg_slot);
}

extern "C" void wfa_segments(wfa_stream_t s, const uint32_t* slot_sorted, int64_t n,
                             uint32_t* scan_tmp, uint32_t* seg_start,
                             uint32_t* seg_slot, int64_t* d_nseg, int shr) {
    hipStream_t st = (hipStream_t)s;
    int64_t nb = (n + RS_PER_BLOCK - 1) / RS_PER_BLOCK;
    hipLaunchKernelGGL(k_seg_count, dim3(nb), dim3(WFA_THREADS), 0, st, slot_sorted, n,
                       scan_tmp, shr);
    hipLaunchKernelGGL(k_rs_scan, dim3(1), dim3(1024), 0, st, scan_tmp, nb);
    hipLaunchKernelGGL(k_seg_scatter, dim3(nb), dim3(WFA_THREADS), 0, st, slot_sorted,
                       n, scan_tmp, seg_start, seg_slot, shr);
    hipLaunchKernelGGL(k_nseg_total, dim3(1), dim3(64), 0, st, slot_sorted, n, scan_tmp,
                       nb, d_nseg, shr);
}

// ===== segmented reduce (Reduce_GPU per-batch semantics) =====
__global__ void k_seg_reduce(const uint32_t* seg_start, const uint32_t* seg_slot,
                             const int64_t* d_nseg, int64_t n, const void* v_orig,
                             const uint32_t* idx_sorted,
                             const int64_t* ts_orig, int vdt, int comb,
                             const uint64_t* slot_to_key, uint64_t* out_key,
                             void* out_val, int64_t* out_ts, int64_t* d_out_n) {
    int64_t nseg = *d_nseg;
    for (int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; j < nseg;
         j += gridDim.x * (int64_t)blockDim.x) {
        int64_t b = seg_start[j];
        int64_t e = (j + 1 < nseg) ? seg_start[j + 1] : n;
        int64_t tmax = ts_orig ? ts_orig[idx_sorted[b]] : 0;
        if (vdt == 0) {
            const int64_t* v = (const int64_t*)v_orig;
            int64_t acc = (comb == 3) ? 0 : v[idx_sorted[b]];
            if (comb == 3) acc = e - b;
            else
                for (int64_t i = b + 1; i < e; ++i) {
                    int64_t x = v[idx_sorted[i]];
                    acc = (comb == 0) ? acc + x : (comb == 1 ? min(acc, x) : max(acc, x));
                }
            ((int64_t*)out_val)[j] = acc;
        } else {
            const float* v = (const float*)v_orig;
            float acc = (comb == 3) ? (float)(e - b) : v[idx_sorted[b]];
            if (comb != 3)
                for (int64_t i = b + 1; i < e; ++i) {
                    float x = v[idx_sorted[i]];
                    acc = (comb == 0) ? acc + x : (comb == 1 ? fminf(acc, x) : fmaxf(acc, x));
                }
            ((float*)out_val)[j] = acc;
        }
        if (ts_orig)
            for (int64_t i = b + 1; i < e; ++i) tmax = max(tmax, ts_orig[idx_sorted[i]]);
        out_key[j] = slot_to_key ? slot_to_key[seg_slot[j]] : (uint64_t)seg_slot[j];
        if (out_ts) out_ts[j] = tmax;
    }
    if (blockIdx.x == 0 && threadIdx.x == 0) *d_out_n = nseg;
}

extern "C" void wfa_segment_reduce(wfa_stream_t s, const uint32_t* seg_start,
                                   const uint32_t* seg_slot, const int64_t* d_nseg,
                                   int64_t n, const void* v_orig,
                                   const uint32_t* idx_sorted,
                                   const int64_t* ts_orig, int vdt, int comb,
                                   const uint64_t* slot_to_key, uint64_t* out_key,
                                   void* out_val, int64_t* out_ts, int64_t* d_out_n) {
    hipLaunchKernelGGL(k_seg_reduce, dim3(WFA_MAX_BLOCKS / 8), dim3(WFA_THREADS), 0,
                       (hipStream_t)s, seg_start, seg_slot, d_nseg, n, v_orig,
                       idx_sorted, ts_orig, vdt, comb, slot_to_key, out_key, out_val,
                       out_ts, d_out_n);
}

// wave-per-segment variant: 64 lanes stride one segment, wave-reduce the
// value and the ts max (one thread per segment serializes ~100-element
// loops with a random ts gather per element — measured 819 us per 4M
// batch at 32 K keys; lanes hide the gather latency 64-wide).
__global__ void k_seg_reduce_wave(const uint32_t* seg_start, const uint32_t* seg_slot,
                                  const int64_t* d_nseg, int64_t n,
                                  const void* v_orig, const uint32_t* idx_sorted,
                                  const int64_t* ts_orig, int vdt, int comb,
                                  int ts_last, const uint64_t* slot_to_key,
                                  uint64_t* out_key,
                                  void* out_val, int64_t* out_ts, int64_t* d_out_n) {
    const int64_t nseg = *d_nseg;
    const int lane = threadIdx.x & 63;
    const int64_t wid = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const int64_t nw = ((int64_t)gridDim.x * blockDim.x) >> 6;
    for (int64_t j = wid; j < nseg; j += nw) {
        const int64_t b = seg_start[j];
        const int64_t e = (j + 1 < nseg) ? seg_start[j + 1] : n;
        float acc = (comb == 1) ? INFINITY : (comb == 2 ? -INFINITY : 0.f);
        int64_t tmax = INT64_MIN;
        for (int64_t i = b + lane; i < e; i += 64) {
            const uint32_t r = idx_sorted[i];
            // vdt >= 6: value rides the sort (VIK / carried payload) and is
            // read at the SORTED position — sequential, not gathered
            float x = wfa_val_at(v_orig, vdt, vdt >= 6 ? i : r);
            acc = (comb == 0 || comb == 3)
                      ? acc + x
                      : (comb == 1 ? fminf(acc, x) : fmaxf(acc, x));
            // ts_last: the batch's ts column is nondecreasing in row order
            // and the stable sort keeps per-segment rows in arrival order,
            // so the segment max is its LAST row — skip the random gather
            if (ts_orig && !ts_last) tmax = max(tmax, ts_orig[r]);
        }
        for (int o = 32; o; o >>= 1) {
            float ov = __shfl_down(acc, o, 64);
            acc = (comb == 0 || comb == 3)
                      ? acc + ov
                      : (comb == 1 ? fminf(acc, ov) : fmaxf(acc, ov));
            tmax = max(tmax, __shfl_down(tmax, o, 64));
        }
        if (lane == 0) {
            ((float*)out_val)[j] = (comb == 3) ? (float)(e - b) : acc;
            out_key[j] = slot_to_key ? slot_to_key[seg_slot[j]] : (uint64_t)seg_slot[j];
            if (out_ts)
                out_ts[j] = (ts_orig && ts_last) ? ts_orig[idx_sorted[e - 1]]
                                                 : tmax;
        }
    }
    if (blockIdx.x == 0 && threadIdx.x == 0) *d_out_n = nseg;
}

extern "C" void wfa_segment_reduce_wave(
    wfa_stream_t s, const uint32_t* seg_start, const uint32_t* seg_slot,
    const int64_t* d_nseg, int64_t n, const void* v_orig, int vdt,
    const uint32_t* idx_sorted, const int64_t* ts_orig, int comb, int ts_last,
    const uint64_t* slot_to_key, uint64_t* out_key, void* out_val,
    int64_t* out_ts, int64_t* d_out_n) {
    hipLaunchKernelGGL(k_seg_reduce_wave, dim3(WFA_MAX_BLOCKS), dim3(WFA_THREADS), 0,
                       (hipStream_t)s, seg_start, seg_slot, d_nseg, n, v_orig,
                       idx_sorted, ts_orig, vdt, comb, ts_last, slot_to_key,
                       out_key, out_val, out_ts, d_out_n);
}


// ===== deterministic window-output offsets =====
// Counting fires per segment is closed-form from the per-slot pane state,
// so output positions come from an exclusive scan instead of 40K serial
// atomicAdds on one cursor word (~88 atomics/us saturation — measured
// 455 us per 4M-tuple batch before this pass existed).
__global__ void k_fire_count(const uint32_t* seg_start, const uint32_t* seg_slot,
                             const int64_t* d_nseg, int64_t n, int64_t pane_len,
                             int64_t P, int64_t S, const uint32_t* st_fill,
                             const uint32_t* st_head, uint32_t* nf,
                             const uint32_t* idx_sorted, const int64_t* ts_orig,
                             int64_t* st_last, uint32_t* pane_nf) {
    const int64_t nseg = *d_nseg;
    for (int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; j < nseg;
         j += gridDim.x * (int64_t)blockDim.x) {
        const uint32_t slot = seg_slot[j];
        const int64_t b = seg_start[j];
        const int64_t e = (j + 1 < nseg) ? seg_start[j + 1] : n;
        const uint64_t fill = st_fill[slot];
        const uint64_t head0 = st_head[slot];
        const uint64_t ncomp = (fill + (uint64_t)(e - b)) / (uint64_t)pane_len;
        auto F = [&](uint64_t x) {
            return x < (uint64_t)P ? 0ull : (x - (uint64_t)P) / (uint64_t)S + 1ull;
        };
        nf[j] = (uint32_t)(F(head0 + ncomp) - F(head0));
        if (pane_nf) pane_nf[j] = (uint32_t)ncomp;
        if (st_last) {  // fused per-key last-arrival ts (EOS flush emit ts)
            const int64_t t = ts_orig[idx_sorted[e - 1]];
            if (t > st_last[slot]) st_last[slot] = t;
        }
    }
}

// exclusive scan of nf[0..*d_nseg) + total -> *d_out_n (single block)
__global__ void k_fire_scan(uint32_t* nf, const int64_t* d_nseg, int64_t* d_out_n) {
    const int64_t m = *d_nseg;
    __shared__ uint32_t tot[1024];
    const int64_t chunk = (m + 1023) / 1024;
    const int64_t b0 = (int64_t)threadIdx.x * chunk;
    const int64_t b1 = min(m, b0 + chunk);
    uint32_t s = 0;
    for (int64_t i = b0; i < b1; ++i) s += nf[i];
    tot[threadIdx.x] = s;
    __syncthreads();
    for (int off = 1; off < 1024; off <<= 1) {
        uint32_t t = (threadIdx.x >= off) ? tot[threadIdx.x - off] : 0;
        __syncthreads();
        tot[threadIdx.x] += t;
        __syncthreads();
    }
    uint32_t run = tot[threadIdx.x] - s;
    for (int64_t i = b0; i < b1; ++i) {
        uint32_t v = nf[i];
        nf[i] = run;
        run += v;
    }
    if (threadIdx.x == 1023 && d_out_n) *d_out_n = tot[1023];
}

extern "C" void wfa_ffat_fire_offsets(wfa_stream_t s, const uint32_t* seg_start,
                                      const uint32_t* seg_slot, const int64_t* d_nseg,
                                      int64_t n, int64_t pane_len, int64_t P,
                                      int64_t S, const uint32_t* st_fill,
                                      const uint32_t* st_head, uint32_t* nf,
                                      int64_t* d_out_n, const uint32_t* idx_sorted,
                                      const int64_t* ts_orig, int64_t* st_last) {
    hipStream_t st = (hipStream_t)s;
    hipLaunchKernelGGL(k_fire_count, dim3(WFA_MAX_BLOCKS / 8), dim3(WFA_THREADS), 0,
                       st, seg_start, seg_slot, d_nseg, n, pane_len, P, S, st_fill,
                       st_head, nf, idx_sorted, ts_orig, st_last,
                       (uint32_t*)nullptr);
    hipLaunchKernelGGL(k_fire_scan, dim3(1), dim3(1024), 0, st, nf, d_nseg, d_out_n);
}

// variant also producing the per-segment COMPLETED-PANE scan (pane_base)
// + total (*d_total_panes) for the pane-wave fold below
extern "C" void wfa_ffat_fire_offsets_pw(
    wfa_stream_t s, const uint32_t* seg_start, const uint32_t* seg_slot,
    const int64_t* d_nseg, int64_t n, int64_t pane_len, int64_t P, int64_t S,
    const uint32_t* st_fill, const uint32_t* st_head, uint32_t* nf,
    int64_t* d_out_n, const uint32_t* idx_sorted, const int64_t* ts_orig,
    int64_t* st_last, uint32_t* pane_base, int64_t* d_total_panes) {
    hipStream_t st = (hipStream_t)s;
    hipLaunchKernelGGL(k_fire_count, dim3(WFA_MAX_BLOCKS / 8), dim3(WFA_THREADS),
                       0, st, seg_start, seg_slot, d_nseg, n, pane_len, P, S,
                       st_fill, st_head, nf, idx_sorted, ts_orig, st_last,
                       pane_base);
    hipLaunchKernelGGL(k_fire_scan, dim3(1), dim3(1024), 0, st, nf, d_nseg, d_out_n);
    hipLaunchKernelGGL(k_fire_scan, dim3(1), dim3(1024), 0, st, pane_base, d_nseg,
                       d_total_panes);
}

// ===== batched multi-key CB sliding-window fold (pane ring) =====
// One thread advances one key's window state over its segment.  Sum uses a
// running window total (invertible); min/max recombine the P-pane ring on
// fire.  All keys of the batch progress in ONE kernel launch — the
// reference's per-key host loop + per-key stream (ffat_replica_gpu.hpp:
// 829-867) becomes a dense device-side state machine.
__global__ void k_ffat_cb(const uint32_t* seg_start, const uint32_t* seg_slot,
                          const int64_t* d_nseg, int64_t n, const void* v_f32,
                          int vdt, const uint32_t* idx_sorted, const int64_t* ts_orig,
                          int64_t pane_len, int64_t P,
                          int64_t S, int comb, int ring_log2, int64_t* st_count,
                          uint32_t* st_fill, float* st_acc, float* ring,
                          uint32_t* st_head, float* st_wsum,
                          const uint64_t* slot_to_key, const uint32_t* fire_base,
                          uint64_t* out_key,
                          float* out_val, int64_t* out_ts,
                          int64_t out_cap) {
    const int64_t nseg = *d_nseg;
    const uint32_t R = 1u << ring_log2;
    const uint32_t Rm = R - 1;
    const float ident = (comb == 1) ? INFINITY : (comb == 2 ? -INFINITY : 0.f);
    for (int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; j < nseg;
         j += gridDim.x * (int64_t)blockDim.x) {
        const uint32_t slot = seg_slot[j];
        int64_t i = seg_start[j];
        const int64_t e = (j + 1 < nseg) ? seg_start[j + 1] : n;
        uint32_t fill = st_fill[slot];
        float acc = st_acc[slot];
        uint32_t head = st_head[slot];
        float wsum = st_wsum[slot];
        int64_t w = fire_base[j];
        float* rg = ring + (size_t)slot * R;
        for (; i < e; ++i) {
            float x = (comb == 3) ? 1.0f
                                  : wfa_val_at(v_f32, vdt, vdt == 6 ? i : idx_sorted[i]);
            acc = (comb == 1) ? fminf(acc, x)
                              : (comb == 2 ? fmaxf(acc, x) : acc + x);
            if (++fill == (uint32_t)pane_len) {
                // pane complete
                if (comb == 0 || comb == 3) {
                    wsum += acc;
                    if (head >= (uint32_t)P) wsum -= rg[(head - (uint32_t)P) & Rm];
                }
                rg[head & Rm] = acc;
                acc = ident;
                fill = 0;
                ++head;
                if (head >= (uint32_t)P && ((head - (uint32_t)P) % (uint32_t)S) == 0) {
                    float res;
                    if (comb == 0 || comb == 3) {
                        res = wsum;
                    } else {
                        res = rg[(head - 1) & Rm];
                        for (uint32_t q = 2; q <= (uint32_t)P; ++q) {
                            float pv = rg[(head - q) & Rm];
                            res = (comb == 1) ? fminf(res, pv) : fmaxf(res, pv);
                        }
                    }
                    if (w < out_cap) {
                        out_key[w] = slot_to_key[slot];
                        out_val[w] = res;
                        out_ts[w] = ts_orig ? ts_orig[idx_sorted[i]] : 0;
                    }
                    ++w;
                }
            }
        }
        st_fill[slot] = fill;
        st_acc[slot] = acc;
        st_head[slot] = head;
        st_wsum[slot] = wsum;
        st_count[slot] += e - seg_start[j];
    }
}

// Wave-per-segment variant: 64 lanes cooperatively stream one key's
// segment in coalesced 256 B chunks; per-chunk pane partials come from
// masked wave reduces; the (tiny) per-pane state machine runs redundantly
// lane-uniform, with lane 0 doing the writes.  For pane_len >= 32 this
// replaces the thread-per-segment kernel (which at 8 K keys puts only 32
// waves on 256 CUs — 1.5 % occupancy, latency-bound).
__global__ void k_ffat_cb_wave(const uint32_t* seg_start, const uint32_t* seg_slot,
                               const int64_t* d_nseg, int64_t n,
                               const void* v_f32, int vdt,
                               const uint32_t* idx_sorted,
                               const int64_t* ts_orig,
                               int64_t pane_len, int64_t P, int64_t S, int comb,
                               int ring_log2, int64_t* st_count, uint32_t* st_fill,
                               float* st_acc, float* ring, uint32_t* st_head,
                               float* st_wsum, const uint64_t* slot_to_key,
                               const uint32_t* fire_base,
                               uint64_t* out_key, float* out_val, int64_t* out_ts,
                               int64_t out_cap) {
    const int64_t nseg = *d_nseg;
    const uint32_t R = 1u << ring_log2;
    const uint32_t Rm = R - 1;
    const float ident = (comb == 1) ? INFINITY : (comb == 2 ? -INFINITY : 0.f);
    const int lane = threadIdx.x & 63;
    const int64_t wid = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const int64_t nw = ((int64_t)gridDim.x * blockDim.x) >> 6;
    const uint32_t L = (uint32_t)pane_len;
#define WCOMB(a, b) ((comb == 1) ? fminf(a, b) : (comb == 2 ? fmaxf(a, b) : (a) + (b)))
    for (int64_t j = wid; j < nseg; j += nw) {
        const uint32_t slot = seg_slot[j];
        const int64_t i0 = seg_start[j];
        const int64_t e = (j + 1 < nseg) ? seg_start[j + 1] : n;
        uint32_t fill = st_fill[slot];
        float acc = st_acc[slot];
        uint32_t head = st_head[slot];
        float wsum = st_wsum[slot];
        int64_t w = fire_base[j];
        float* rg = ring + (size_t)slot * R;
        for (int64_t pos = i0; pos < e; pos += 64) {
            const uint32_t nchunk = (uint32_t)min((int64_t)64, e - pos);
            float v = (lane < (int)nchunk)
                          ? ((comb == 3)
                                 ? 1.0f
                                 : wfa_val_at(v_f32, vdt,
                                              vdt == 6 ? pos + lane
                                                       : idx_sorted[pos + lane]))
                          : ident;
            uint32_t rel = (lane < (int)nchunk) ? (fill + (uint32_t)lane) / L : ~0u;
            const uint32_t maxrel = (fill + nchunk - 1) / L;
            const uint32_t ncomplete = (fill + nchunk) / L;  // panes finished here
            for (uint32_t r = 0; r <= maxrel; ++r) {
                float pv = (rel == r) ? v : ident;
                for (int o = 32; o; o >>= 1) pv = WCOMB(pv, __shfl_xor(pv, o, 64));
                acc = WCOMB(acc, pv);
                if (r < ncomplete) {  // pane boundary crossed inside this chunk
                    if (lane == 0) rg[head & Rm] = acc;
                    if (comb == 0 || comb == 3) {
                        wsum += acc;
                        if (head >= (uint32_t)P) wsum -= rg[(head - (uint32_t)P) & Rm];
                    }
                    ++head;
                    const float closed = acc;
                    acc = ident;
                    if (head >= (uint32_t)P &&
                        ((head - (uint32_t)P) % (uint32_t)S) == 0) {
                        float res;
                        if (comb == 0 || comb == 3) {
                            res = wsum;
                        } else {
                            // lane-parallel recombine of the last P panes
                            float part = ident;
                            for (uint32_t q = lane; q < (uint32_t)P; q += 64) {
                                float x = (q == 0) ? closed : rg[(head - 1 - q) & Rm];
                                part = WCOMB(part, x);
                            }
                            for (int o = 32; o; o >>= 1)
                                part = WCOMB(part, __shfl_xor(part, o, 64));
                            res = part;
                        }
                        if (lane == 0 && w < out_cap) {
                            out_key[w] = slot_to_key[slot];
                            out_val[w] = res;
                            int64_t last = pos + (int64_t)((r + 1) * L - fill) - 1;
                            out_ts[w] = ts_orig ? ts_orig[idx_sorted[last]] : 0;
                        }
                        ++w;
                    }
                }
            }
            fill = fill + nchunk - ncomplete * L;
        }
        if (lane == 0) {
            st_fill[slot] = fill;
            st_acc[slot] = acc;
            st_head[slot] = head;
            st_wsum[slot] = wsum;
            st_count[slot] += e - i0;
        }
    }
#undef WCOMB
}

// ===== pane-wave CB fold =====
// The serial per-key state machine above walks TUPLES; at pane_len >= 32
// almost all of that walk is pane-partial accumulation, which is
// independent per pane.  Decomposition: (A) one wave per completed pane
// computes its partial from the sorted values into a dense temp array
// (pane 0 folds the carried open-pane accumulator), plus one wave per
// segment for the new open tail -> st_acc_new; (B) one thread per segment
// replays the tiny per-PANE machine (ring/wsum/fires) over the temps —
// 1/pane_len of the old serial work.  A only READS state; B writes it.
__global__ void k_pane_partials(const uint32_t* seg_start,
                                const uint32_t* seg_slot,
                                const int64_t* d_nseg,
                                const int64_t* d_total_panes, int64_t n,
                                const void* v_f32, int vdt,
                                const uint32_t* idx_sorted, int64_t pane_len,
                                int comb, const uint32_t* st_fill,
                                const float* st_acc, const uint32_t* pane_base,
                                float* temp, float* st_acc_new) {
    const int64_t nseg = *d_nseg;
    const int64_t total = *d_total_panes;
    const float ident = (comb == 1) ? INFINITY : (comb == 2 ? -INFINITY : 0.f);
    const int lane = threadIdx.x & 63;
    const int64_t wid = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const int64_t nw = ((int64_t)gridDim.x * blockDim.x) >> 6;
#define PWCOMB(a, b) ((comb == 1) ? fminf(a, b) : (comb == 2 ? fmaxf(a, b) : (a) + (b)))
    for (int64_t g = wid; g < total + nseg; g += nw) {
        int64_t j, p;
        bool tail = g >= total;
        if (tail) {
            j = g - total;
            p = -1;
        } else {
            int64_t lo = 0, hi = nseg - 1;
            while (lo < hi) {
                int64_t mid = (lo + hi + 1) >> 1;
                if ((int64_t)pane_base[mid] <= g) lo = mid;
                else hi = mid - 1;
            }
            j = lo;
            p = g - (int64_t)pane_base[j];
        }
        const uint32_t slot = seg_slot[j];
        const int64_t i0 = seg_start[j];
        const int64_t e = (j + 1 < nseg) ? seg_start[j + 1] : n;
        const int64_t fill = st_fill[slot];
        const int64_t ncomp = (fill + (e - i0)) / pane_len;
        int64_t start, cnt;
        float carry = ident;
        if (tail) {
            // new open pane: rows after the last completed pane
            start = i0 + (ncomp == 0 ? 0 : ncomp * pane_len - fill);
            cnt = e - start;
            if (ncomp == 0) carry = st_acc[slot];  // pane still open
        } else if (p == 0) {
            start = i0;
            cnt = pane_len - fill;
            carry = st_acc[slot];
        } else {
            start = i0 + p * pane_len - fill;
            cnt = pane_len;
        }
        float part = ident;
        for (int64_t q = lane; q < cnt; q += 64) {
            float x = (comb == 3)
                          ? 1.0f
                          : wfa_val_at(v_f32, vdt,
                                       vdt == 6 ? start + q : idx_sorted[start + q]);
            part = PWCOMB(part, x);
        }
        for (int o = 32; o; o >>= 1) part = PWCOMB(part, __shfl_xor(part, o, 64));
        part = PWCOMB(part, carry);
        if (lane == 0) {
            if (tail) st_acc_new[slot] = part;
            else temp[g] = part;
        }
    }
#undef PWCOMB
}

__global__ void k_pane_advance(const uint32_t* seg_start,
                               const uint32_t* seg_slot, const int64_t* d_nseg,
                               int64_t n, const uint32_t* idx_sorted,
                               const int64_t* ts_orig, int64_t pane_len,
                               int64_t P, int64_t S, int comb, int ring_log2,
                               int64_t* st_count, uint32_t* st_fill,
                               float* st_acc, const float* st_acc_new,
                               float* ring, uint32_t* st_head, float* st_wsum,
                               const uint64_t* slot_to_key,
                               const uint32_t* fire_base,
                               const uint32_t* pane_base, const float* temp,
                               uint64_t* out_key, float* out_val,
                               int64_t* out_ts, int64_t out_cap) {
    const int64_t nseg = *d_nseg;
    const uint32_t R = 1u << ring_log2;
    const uint32_t Rm = R - 1;
    for (int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; j < nseg;
         j += gridDim.x * (int64_t)blockDim.x) {
        const uint32_t slot = seg_slot[j];
        const int64_t i0 = seg_start[j];
        const int64_t e = (j + 1 < nseg) ? seg_start[j + 1] : n;
        const int64_t fill = st_fill[slot];
        const int64_t len = e - i0;
        const int64_t ncomp = (fill + len) / pane_len;
        uint32_t head = st_head[slot];
        float wsum = st_wsum[slot];
        int64_t w = fire_base[j];
        float* rg = ring + (size_t)slot * R;
        const float* tp = temp + pane_base[j];
        for (int64_t q = 0; q < ncomp; ++q) {
            const float acc = tp[q];
            if (comb == 0 || comb == 3) {
                wsum += acc;
                if (head >= (uint32_t)P) wsum -= rg[(head - (uint32_t)P) & Rm];
            }
            rg[head & Rm] = acc;
            ++head;
            if (head >= (uint32_t)P && ((head - (uint32_t)P) % (uint32_t)S) == 0) {
                float res;
                if (comb == 0 || comb == 3) {
                    res = wsum;
                } else {
                    res = rg[(head - 1) & Rm];
                    for (uint32_t z = 2; z <= (uint32_t)P; ++z) {
                        float pv = rg[(head - z) & Rm];
                        res = (comb == 1) ? fminf(res, pv) : fmaxf(res, pv);
                    }
                }
                if (w < out_cap) {
                    out_key[w] = slot_to_key[slot];
                    out_val[w] = res;
                    const int64_t last = i0 + (q + 1) * pane_len - fill - 1;
                    out_ts[w] = ts_orig ? ts_orig[idx_sorted[last]] : 0;
                }
                ++w;
            }
        }
        st_fill[slot] = (uint32_t)(fill + len - ncomp * pane_len);
        st_acc[slot] = st_acc_new[slot];
        st_head[slot] = head;
        st_wsum[slot] = wsum;
        st_count[slot] += len;
    }
}

extern "C" void wfa_ffat_cb_fold_pw(
    wfa_stream_t s, const uint32_t* seg_start, const uint32_t* seg_slot,
    const int64_t* d_nseg, const int64_t* d_total_panes, int64_t n,
    const void* v_f32, int vdt, const uint32_t* idx_sorted,
    const int64_t* ts_orig, int64_t pane_len, int64_t P, int64_t S, int comb,
    int ring_log2, int64_t* st_count, uint32_t* st_fill, float* st_acc,
    float* st_acc_new, float* ring, uint32_t* st_head, float* st_wsum,
    const uint64_t* slot_to_key, const uint32_t* fire_base,
    const uint32_t* pane_base, float* temp, uint64_t* out_key, float* out_val,
    int64_t* out_ts, int64_t out_cap) {
    hipStream_t st = (hipStream_t)s;
    hipLaunchKernelGGL(k_pane_partials, dim3(WFA_MAX_BLOCKS), dim3(WFA_THREADS),
                       0, st, seg_start, seg_slot, d_nseg, d_total_panes, n,
                       v_f32, vdt, idx_sorted, pane_len, comb, st_fill, st_acc,
                       pane_base, temp, st_acc_new);
    hipLaunchKernelGGL(k_pane_advance, dim3(WFA_MAX_BLOCKS / 8),
                       dim3(WFA_THREADS), 0, st, seg_start, seg_slot, d_nseg, n,
                       idx_sorted, ts_orig, pane_len, P, S, comb, ring_log2,
                       st_count, st_fill, st_acc, st_acc_new, ring, st_head,
                       st_wsum, slot_to_key, fire_base, pane_base, temp,
                       out_key, out_val, out_ts, out_cap);
}

extern "C" void wfa_ffat_cb_fold(wfa_stream_t s, const uint32_t* seg_start,
                                 const uint32_t* seg_slot, const int64_t* d_nseg,
                                 int64_t n, const void* v_f32, int vdt,
                                 const uint32_t* idx_sorted, const int64_t* ts_orig,
                                 int64_t pane_len, int64_t P,
                                 int64_t S, int comb, int ring_log2, int64_t* st_count,
                                 uint32_t* st_fill, float* st_acc, float* ring,
                                 uint32_t* st_head, float* st_wsum,
                                 const uint64_t* slot_to_key,
                                 const uint32_t* fire_base, uint64_t* out_key,
                                 float* out_val, int64_t* out_ts, int64_t out_cap) {
    if (pane_len >= 32)
        hipLaunchKernelGGL(k_ffat_cb_wave, dim3(WFA_MAX_BLOCKS), dim3(WFA_THREADS), 0,
                           (hipStream_t)s, seg_start, seg_slot, d_nseg, n, v_f32, vdt,
                           idx_sorted, ts_orig, pane_len, P, S, comb, ring_log2,
                           st_count, st_fill, st_acc, ring, st_head, st_wsum,
                           slot_to_key, fire_base, out_key, out_val, out_ts, out_cap);
    else
        hipLaunchKernelGGL(k_ffat_cb, dim3(WFA_MAX_BLOCKS / 4), dim3(WFA_THREADS), 0,
                           (hipStream_t)s, seg_start, seg_slot, d_nseg, n, v_f32, vdt,
                           idx_sorted, ts_orig, pane_len, P, S, comb, ring_log2,
                           st_count, st_fill, st_acc, ring, st_head, st_wsum,
                           slot_to_key, fire_base, out_key, out_val, out_ts, out_cap);
}

// ===== MFMA windowed Gram aggregation (matrix-core combine) =====
// Per-key tumbling windows over 16-dim f32 tuple vectors; the window
// aggregate Σ v·vᵀ (online covariance / Gram) is GEMM-shaped, so it runs
// on the matrix cores: C(16x16) += A(16x4)·B(4x16) with A = Vᵀ, B = V via
// v_mfma_f32_16x16x4_f32 (exact f32, guide §3).  Lane layout (CDNA):
//   A[i=l&15][k=l>>4], B[k=l>>4][j=l&15]  -> a = b = V[k][l&15]
//   C/D: col = l&15, row = (l>>4)*4 + reg
// One wave per key segment; the 4-reg accumulator persists across batches
// in a per-slot arena laid out in the SAME lane mapping.
typedef __attribute__((ext_vector_type(4))) float f32x4;

// inverse permutation: inv[idx_sorted[j]] = j (sorted position of row r)
__global__ void k_inv_perm(const uint32_t* idx_sorted, int64_t n, uint32_t* inv) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (int64_t)blockDim.x)
        inv[idx_sorted[i]] = (uint32_t)i;
}

// transpose-gather the 16 value columns into SORTED row-major staging:
// staged[inv[r]*16 + c] = colp[c][r].  Thread per SOURCE row: the 16
// column reads are coalesced across the wave, and the 64 B row write is
// one full cache line at a random offset — this replaces the per-MFMA-step
// random gather that made the fold memory-bound (measured 2.4 ms/8M,
// ~1.8 TFLOP/s; see profiles/gram_mfma_r02.md).
// 4 lanes cooperate per row: each writes one float4, so the 4 stores of
// one row land in ONE 64 B line transaction (a scalar version did 16
// separate 4 B writes per row -> L2 transaction-bound, 2.7 ms/8M)
__global__ void k_gram_stage(const float* const* colp, const uint32_t* inv,
                             int64_t n, float* staged) {
    const int sub = threadIdx.x & 3;  // which float4 of the 16-f row
    const float* c0 = colp[4 * sub + 0];
    const float* c1 = colp[4 * sub + 1];
    const float* c2 = colp[4 * sub + 2];
    const float* c3 = colp[4 * sub + 3];
    const int64_t rbase = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 2;
    const int64_t rstride = ((int64_t)gridDim.x * blockDim.x) >> 2;
    for (int64_t r = rbase; r < n; r += rstride) {
        float4 v;
        v.x = c0[r];
        v.y = c1[r];
        v.z = c2[r];
        v.w = c3[r];
        ((float4*)staged)[(int64_t)inv[r] * 4 + sub] = v;
    }
}

__global__ void k_gram(const uint32_t* seg_start, const uint32_t* seg_slot,
                       const int64_t* d_nseg, int64_t n,
                       const float* staged,  // [n][16] sorted row-major
                       const uint32_t* idx_sorted, const int64_t* ts_orig,
                       int64_t win, uint32_t* st_fill, float* st_acc /*256/slot*/,
                       uint32_t* st_head, const uint64_t* slot_to_key,
                       const uint32_t* fire_base, uint64_t* out_key,
                       int64_t* out_gwid, float* const* out_colp /*16*/,
                       int64_t* out_ts, int64_t out_cap) {
    const int64_t nseg = *d_nseg;
    const int lane = threadIdx.x & 63;
    const int64_t wid = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const int64_t nw = ((int64_t)gridDim.x * blockDim.x) >> 6;
    const int krow = lane >> 4;   // 0..3 (K slice of this step)
    const int dim = lane & 15;    // matrix column
    for (int64_t j = wid; j < nseg; j += nw) {
        const uint32_t slot = seg_slot[j];
        const int64_t i0 = seg_start[j];
        const int64_t e = (j + 1 < nseg) ? seg_start[j + 1] : n;
        int64_t cnt = st_fill[slot];
        uint32_t head = st_head[slot];
        f32x4 acc;
        float* sa = st_acc + (size_t)slot * 256;
#pragma unroll
        for (int r = 0; r < 4; ++r) acc[r] = sa[lane * 4 + r];
        int64_t w = fire_base[j];
        int64_t pos = i0;
        int64_t last_ts = 0;
        while (pos < e) {
            int64_t chunk = min((int64_t)4, min(e - pos, win - cnt));
            float v = 0.f;
            if (krow < chunk) v = staged[(pos + krow) * 16 + dim];
            acc = __builtin_amdgcn_mfma_f32_16x16x4f32(v, v, acc, 0, 0, 0);
            last_ts = ts_orig[idx_sorted[pos + chunk - 1]];
            pos += chunk;
            cnt += chunk;
            if (cnt == win) {
                // fire: 16 output rows (one Gram row each)
                const int64_t base = w * 16;
                if (base + 16 <= out_cap) {
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        int row = (lane >> 4) * 4 + r;
                        out_colp[dim][base + row] = acc[r];
                        if (dim == 0) {
                            out_key[base + row] = slot_to_key[slot];
                            out_gwid[base + row] = head;
                            out_ts[base + row] = last_ts;
                        }
                    }
                }
                ++w;
                ++head;
#pragma unroll
                for (int r = 0; r < 4; ++r) acc[r] = 0.f;
                cnt = 0;
            }
        }
        st_fill[slot] = (uint32_t)cnt;
        st_head[slot] = head;
#pragma unroll
        for (int r = 0; r < 4; ++r) sa[lane * 4 + r] = acc[r];
    }
}

// window-per-wave variant: one wave per FIRED WINDOW (k_gram_win, reads
// per-slot state, never writes) then one wave per segment for the tail
// (k_gram_tail, updates state) — TWO launches so state reads order before
// writes.  At low key counts the per-segment serial MFMA chain starves
// the chip (profiles/gram_mfma_r02.md); fired windows have independent
// accumulators and spread across all wave slots.
__global__ void k_gram_win(const uint32_t* seg_start, const uint32_t* seg_slot,
                           const int64_t* d_nseg, const int64_t* d_total,
                           int64_t n, const float* staged,
                           const uint32_t* idx_sorted, const int64_t* ts_orig,
                           int64_t win, const uint32_t* st_fill,
                           const float* st_acc, const uint32_t* st_head,
                           const uint64_t* slot_to_key,
                           const uint32_t* fire_base, uint64_t* out_key,
                           int64_t* out_gwid, float* const* out_colp,
                           int64_t* out_ts, int64_t out_cap) {
    const int64_t nseg = *d_nseg;
    const int64_t total = *d_total;
    const int lane = threadIdx.x & 63;
    const int64_t wid = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const int64_t nw = ((int64_t)gridDim.x * blockDim.x) >> 6;
    const int krow = lane >> 4;
    const int dim = lane & 15;
    for (int64_t p = wid; p < total; p += nw) {
        // binary-search the owning segment in the fire-offset scan
        int64_t lo = 0, hi = nseg - 1;
        while (lo < hi) {
            int64_t mid = (lo + hi + 1) >> 1;
            if ((int64_t)fire_base[mid] <= p) lo = mid;
            else hi = mid - 1;
        }
        const int64_t j = lo;
        const int64_t w = p - (int64_t)fire_base[j];
        const uint32_t slot = seg_slot[j];
        const int64_t i0 = seg_start[j];
        const int64_t fill = st_fill[slot];
        f32x4 acc;
        int64_t start, cnt;
        if (w == 0) {
            start = i0;
            cnt = win - fill;
            const float* sa = st_acc + (size_t)slot * 256;
#pragma unroll
            for (int r = 0; r < 4; ++r) acc[r] = sa[lane * 4 + r];
        } else {
            start = i0 + w * win - fill;
            cnt = win;
#pragma unroll
            for (int r = 0; r < 4; ++r) acc[r] = 0.f;
        }
        for (int64_t q = 0; q < cnt; q += 4) {
            float v = 0.f;
            if (krow < cnt - q) v = staged[(start + q + krow) * 16 + dim];
            acc = __builtin_amdgcn_mfma_f32_16x16x4f32(v, v, acc, 0, 0, 0);
        }
        const int64_t base = p * 16;
        if (base + 16 <= out_cap) {
            const int64_t last_ts = ts_orig[idx_sorted[start + cnt - 1]];
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = krow * 4 + r;
                out_colp[dim][base + row] = acc[r];
                if (dim == 0) {
                    out_key[base + row] = slot_to_key[slot];
                    out_gwid[base + row] = (int64_t)st_head[slot] + w;
                    out_ts[base + row] = last_ts;
                }
            }
        }
    }
}

__global__ void k_gram_tail(const uint32_t* seg_start, const uint32_t* seg_slot,
                            const int64_t* d_nseg, const int64_t* d_total,
                            int64_t n, const float* staged, int64_t win,
                            uint32_t* st_fill, float* st_acc, uint32_t* st_head,
                            const uint32_t* fire_base) {
    const int64_t nseg = *d_nseg;
    const int64_t total = *d_total;
    const int lane = threadIdx.x & 63;
    const int64_t wid = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const int64_t nw = ((int64_t)gridDim.x * blockDim.x) >> 6;
    const int krow = lane >> 4;
    const int dim = lane & 15;
    for (int64_t j = wid; j < nseg; j += nw) {
        const uint32_t slot = seg_slot[j];
        const int64_t i0 = seg_start[j];
        const int64_t e = (j + 1 < nseg) ? seg_start[j + 1] : n;
        const int64_t fill = st_fill[slot];
        const int64_t fires =
            ((j + 1 < nseg) ? (int64_t)fire_base[j + 1] : total) -
            (int64_t)fire_base[j];
        const int64_t consumed = fires == 0 ? 0 : fires * win - fill;
        float* sa = st_acc + (size_t)slot * 256;
        f32x4 acc;
        if (fires == 0) {
#pragma unroll
            for (int r = 0; r < 4; ++r) acc[r] = sa[lane * 4 + r];
        } else {
#pragma unroll
            for (int r = 0; r < 4; ++r) acc[r] = 0.f;
        }
        for (int64_t q = i0 + consumed; q < e; q += 4) {
            float v = 0.f;
            if (q + krow < e) v = staged[(q + krow) * 16 + dim];
            acc = __builtin_amdgcn_mfma_f32_16x16x4f32(v, v, acc, 0, 0, 0);
        }
#pragma unroll
        for (int r = 0; r < 4; ++r) sa[lane * 4 + r] = acc[r];
        if (lane == 0) {
            st_fill[slot] = (uint32_t)(fill + (e - i0) - fires * win);
            st_head[slot] += (uint32_t)fires;
        }
    }
}

__global__ void k_scale16(int64_t* d_out_n) { *d_out_n *= 16; }

extern "C" void wfa_gram_prep(wfa_stream_t s, const float* const* colp,
                              const uint32_t* idx_sorted, int64_t n,
                              uint32_t* inv_scratch, float* staged) {
    hipStream_t st = (hipStream_t)s;
    hipLaunchKernelGGL(k_inv_perm, dim3(nblk(n)), dim3(WFA_THREADS), 0, st,
                       idx_sorted, n, inv_scratch);
    hipLaunchKernelGGL(k_gram_stage, dim3(nblk(n)), dim3(WFA_THREADS), 0, st,
                       colp, inv_scratch, n, staged);
}

extern "C" void wfa_gram_fold(wfa_stream_t s, const uint32_t* seg_start,
                              const uint32_t* seg_slot, const int64_t* d_nseg,
                              int64_t n, const float* const* colp,
                              const uint32_t* idx_sorted, const int64_t* ts_orig,
                              int64_t win, uint32_t* st_fill, float* st_acc,
                              uint32_t* st_head, const uint64_t* slot_to_key,
                              const uint32_t* fire_base, uint64_t* out_key,
                              int64_t* out_gwid, float* const* out_colp,
                              int64_t* out_ts, int64_t out_cap,
                              int64_t* d_out_n, uint32_t* inv_scratch,
                              float* staged) {
    hipStream_t st = (hipStream_t)s;
    (void)colp; (void)inv_scratch;  // staging done by wfa_gram_prep
    static int seg_mode = -1;
    if (seg_mode < 0) {
        const char* e = getenv("WFA_GRAM_SEG");
        seg_mode = (e && e[0] == '1') ? 1 : 0;
    }
    if (seg_mode) {
        // legacy wave-per-segment fold (A/B reference)
        hipLaunchKernelGGL(k_gram, dim3(WFA_MAX_BLOCKS / 2), dim3(WFA_THREADS),
                           0, st, seg_start, seg_slot, d_nseg, n, staged,
                           idx_sorted, ts_orig, win, st_fill, st_acc, st_head,
                           slot_to_key, fire_base, out_key, out_gwid, out_colp,
                           out_ts, out_cap);
    } else {
        hipLaunchKernelGGL(k_gram_win, dim3(WFA_MAX_BLOCKS), dim3(WFA_THREADS),
                           0, st, seg_start, seg_slot, d_nseg, d_out_n, n,
                           staged, idx_sorted, ts_orig, win, st_fill, st_acc,
                           st_head, slot_to_key, fire_base, out_key, out_gwid,
                           out_colp, out_ts, out_cap);
        hipLaunchKernelGGL(k_gram_tail, dim3(WFA_MAX_BLOCKS / 2),
                           dim3(WFA_THREADS), 0, st, seg_start, seg_slot,
                           d_nseg, d_out_n, n, staged, win, st_fill, st_acc,
                           st_head, fire_base);
    }
    hipLaunchKernelGGL(k_scale16, dim3(1), dim3(1), 0, st, d_out_n);
}

// ===== stateful map/filter: per-key device state, key-order walk =====
// Reference Stateful_MAPGPU_Kernel / Stateful_FILTERGPU_Kernel
// (map_gpu.hpp:80-102, filter_gpu.hpp:92-115): one worker per distinct key
// walks that key's tuples IN ORDER applying func(tuple, state).  There the
// keyed states are objects in a shared tbb map guarded by a spinlock; here
// state is a dense per-slot arena (f64) indexed by the hash-slot machinery,
// one thread per segment, results written back to original row positions.
// map specs:    1 running_sum (x=state+=x)  2 ema (state=a*state+(1-a)x, x=state)
//               3 running_count (x=++state)
// filter specs: 1 dedup_consecutive (keep if x != state; state=x)
//               2 every_kth (keep when ++state % k == 0)
// typed load/store for the stateful walk (round-2 fix: round 1 treated
// every non-i64 dtype as f32 — a bf16 column got 4-byte accesses, twice
// the buffer, silently corrupting neighboring device allocations)
__device__ __forceinline__ double st_load(const void* col, int dt, uint32_t r) {
    switch (dt) {
        case 0: return (double)((const int64_t*)col)[r];
        case 1: return ((const double*)col)[r];
        case 2: return (double)((const float*)col)[r];
        case 3: return (double)((const uint64_t*)col)[r];
        case 4: return (double)((const int*)col)[r];
        default: {  // bf16
            union { uint32_t u; float f; } c;
            c.u = (uint32_t)((const uint16_t*)col)[r] << 16;
            return (double)c.f;
        }
    }
}
__device__ __forceinline__ void st_store(void* col, int dt, uint32_t r, double x) {
    switch (dt) {
        case 0: ((int64_t*)col)[r] = (int64_t)x; break;
        case 1: ((double*)col)[r] = x; break;
        case 2: ((float*)col)[r] = (float)x; break;
        case 3: ((uint64_t*)col)[r] = (uint64_t)x; break;
        case 4: ((int*)col)[r] = (int)x; break;
        default: {  // bf16 round-to-nearest-even
            union { uint32_t u; float f; } c;
            c.f = (float)x;
            uint32_t lsb = (c.u >> 16) & 1;
            c.u += 0x7fff + lsb;
            ((uint16_t*)col)[r] = (uint16_t)(c.u >> 16);
        }
    }
}

__global__ void k_stateful(const uint32_t* seg_start, const uint32_t* seg_slot,
                           const int64_t* d_nseg, int64_t n,
                           const uint32_t* idx_sorted, void* col, int dt,
                           int spec, int is_filter, double a, double b,
                           double* state, uint32_t* flags) {
    const int64_t nseg = *d_nseg;
    for (int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; j < nseg;
         j += gridDim.x * (int64_t)blockDim.x) {
        const uint32_t slot = seg_slot[j];
        const int64_t e = (j + 1 < nseg) ? seg_start[j + 1] : n;
        double st = state[slot];
        for (int64_t i = seg_start[j]; i < e; ++i) {
            const uint32_t r = idx_sorted[i];
            double x = st_load(col, dt, r);
            if (!is_filter) {
                switch (spec) {
                    case 1: st += x; x = st; break;
                    case 2: st = a * st + (1.0 - a) * x; x = st; break;
                    case 3: st += 1.0; x = st; break;
                }
                st_store(col, dt, r, x);
            } else {
                uint32_t keep = 1;
                switch (spec) {
                    case 1: keep = (x != st); st = x; break;
                    case 2: st += 1.0; keep = ((int64_t)st % (int64_t)a) == 0; break;
                }
                flags[r] = keep;
            }
        }
        state[slot] = st;
    }
}

extern "C" void wfa_stateful_apply(wfa_stream_t s, const uint32_t* seg_start,
                                   const uint32_t* seg_slot, const int64_t* d_nseg,
                                   int64_t n, const uint32_t* idx_sorted, void* col,
                                   int dt, int spec, int is_filter, double a,
                                   double b, double* state, uint32_t* flags) {
    hipLaunchKernelGGL(k_stateful, dim3(WFA_MAX_BLOCKS / 8), dim3(WFA_THREADS), 0,
                       (hipStream_t)s, seg_start, seg_slot, d_nseg, n, idx_sorted,
                       col, dt, spec, is_filter, a, b, state, flags);
}

// ===== time-based (TB) keyed windows: pane lift + watermark advance =====
// Event-time redesign of the reference's TB path (ffat_replica_gpu.hpp:
// 869-1047: lift -> thrust (key,pane) reduce -> PendingPanes_Queue ->
// per-key FlatFAT): tuples accumulate into a per-slot circular pending-pane
// buffer (pane id = ts / pane_len, absolute); the watermark completes panes
// up to limit = (wm - lateness)/pane_len - 1; completed panes (gaps =
// identity) advance the same ring/wsum window machine as the CB fold.
// Keys are aligned to the absolute window grid at first touch (pend_base =
// first window containing the first tuple), matching the CPU FfatCpuLogic.

// pass 1: per segment, accumulate tuples into the pending ring
__global__ void k_tb_lift(const uint32_t* seg_start, const uint32_t* seg_slot,
                          const int64_t* d_nseg, int64_t n, const void* v_f32,
                          int vdt, const uint32_t* idx_sorted, const int64_t* ts_orig,
                          int64_t pane_len, int64_t P, int64_t S, int comb,
                          int pend_log2, float* pend, int64_t* pend_base,
                          int64_t* last_pane, uint32_t* ignored,
                          uint32_t* overflow) {
    const int64_t nseg = *d_nseg;
    const uint32_t Rp = 1u << pend_log2;
    const uint32_t Pm = Rp - 1;
    for (int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; j < nseg;
         j += gridDim.x * (int64_t)blockDim.x) {
        const uint32_t slot = seg_slot[j];
        const int64_t e = (j + 1 < nseg) ? seg_start[j + 1] : n;
        float* pd = pend + (size_t)slot * Rp;
        int64_t base = pend_base[slot];
        int64_t lastp = last_pane[slot];
        uint32_t ign = 0;
        for (int64_t i = seg_start[j]; i < e; ++i) {
            const int64_t ts = ts_orig[idx_sorted[i]];
            const int64_t p = ts / pane_len;
            if (base < 0) {
                // first tuple of this key: align to the absolute window grid
                int64_t w0 = ts - P * pane_len + 1;
                w0 = w0 <= 0 ? 0 : (w0 + S * pane_len - 1) / (S * pane_len);
                base = w0 * S;
            }
            if (p < base) {
                ++ign;  // late beyond completed panes
                continue;
            }
            if (p - base >= (int64_t)Rp) {
                atomicAdd(overflow, 1u);
                continue;
            }
            const float x = (comb == 3)
                                ? 1.0f
                                : wfa_val_at(v_f32, vdt,
                                             vdt == 6 ? i : idx_sorted[i]);
            float* cell = &pd[(uint64_t)p & Pm];
            *cell = (comb == 1) ? fminf(*cell, x)
                                : (comb == 2 ? fmaxf(*cell, x) : *cell + x);
            if (p > lastp) lastp = p;
        }
        pend_base[slot] = base;
        last_pane[slot] = lastp;
        if (ign) atomicAdd(ignored, ign);
    }
}

// pass 2: fires per SLOT (watermark advances every key, batch or not)
__global__ void k_tb_count(const uint32_t* n_slots, int64_t limit_pane,
                           const int64_t* pend_base, const int64_t* last_pane,
                           const uint32_t* st_head, int64_t P, int64_t S,
                           uint32_t* nf) {
    const int64_t ns = *n_slots;
    for (int64_t s = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; s < ns;
         s += gridDim.x * (int64_t)blockDim.x) {
        int64_t base = pend_base[s];
        int64_t ncomp = 0;
        if (base >= 0) {
            int64_t hi = min(limit_pane, last_pane[s]);
            ncomp = hi >= base ? hi - base + 1 : 0;
        }
        auto F = [&](uint64_t x) {
            return x < (uint64_t)P ? 0ull : (x - (uint64_t)P) / (uint64_t)S + 1ull;
        };
        const uint64_t h0 = st_head[s];
        nf[s] = (uint32_t)(F(h0 + (uint64_t)ncomp) - F(h0));
    }
}

__global__ void k_tb_scan(uint32_t* nf, const uint32_t* n_slots, int64_t* d_out_n) {
    const int64_t m = *n_slots;
    __shared__ uint32_t tot[1024];
    const int64_t chunk = (m + 1023) / 1024;
    const int64_t b0 = (int64_t)threadIdx.x * chunk;
    const int64_t b1 = min(m, b0 + chunk);
    uint32_t s = 0;
    for (int64_t i = b0; i < b1; ++i) s += nf[i];
    tot[threadIdx.x] = s;
    __syncthreads();
    for (int off = 1; off < 1024; off <<= 1) {
        uint32_t t = (threadIdx.x >= off) ? tot[threadIdx.x - off] : 0;
        __syncthreads();
        tot[threadIdx.x] += t;
        __syncthreads();
    }
    uint32_t run = tot[threadIdx.x] - s;
    for (int64_t i = b0; i < b1; ++i) {
        uint32_t v = nf[i];
        nf[i] = run;
        run += v;
    }
    if (threadIdx.x == 1023 && d_out_n) *d_out_n = tot[1023];
}

// pass 3: per slot, insert completed panes into the window ring and fire
__global__ void k_tb_advance(const uint32_t* n_slots, int64_t limit_pane,
                             int64_t pane_len, int64_t P, int64_t S, int comb,
                             int ring_log2, int pend_log2, float* pend,
                             int64_t* pend_base, const int64_t* last_pane,
                             uint32_t* st_head, float* st_wsum, float* ring,
                             const uint64_t* slot_to_key, const uint32_t* nf,
                             uint64_t* out_key, float* out_val, int64_t* out_ts,
                             int64_t out_cap) {
    const int64_t ns = *n_slots;
    const uint32_t R = 1u << ring_log2;
    const uint32_t Rm = R - 1;
    const uint32_t Rp = 1u << pend_log2;
    const uint32_t Pm = Rp - 1;
    const float ident = (comb == 1) ? INFINITY : (comb == 2 ? -INFINITY : 0.f);
    for (int64_t s = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; s < ns;
         s += gridDim.x * (int64_t)blockDim.x) {
        int64_t base = pend_base[s];
        if (base < 0) continue;
        const int64_t hi = min(limit_pane, last_pane[s]);
        if (hi < base) continue;
        float* pd = pend + (size_t)s * Rp;
        float* rg = ring + (size_t)s * R;
        uint32_t head = st_head[s];
        float wsum = st_wsum[s];
        int64_t w = nf[s];
        for (int64_t q = base; q <= hi; ++q) {
            float pane = pd[(uint64_t)q & Pm];
            pd[(uint64_t)q & Pm] = ident;  // consumed
            rg[head & Rm] = pane;
            if (comb == 0 || comb == 3) {
                wsum += pane;
                if (head >= (uint32_t)P) wsum -= rg[(head - (uint32_t)P) & Rm];
            }
            ++head;
            if (head >= (uint32_t)P && ((head - (uint32_t)P) % (uint32_t)S) == 0) {
                float res;
                if (comb == 0 || comb == 3) {
                    res = wsum;
                } else {
                    res = rg[(head - 1) & Rm];
                    for (uint32_t q2 = 2; q2 <= (uint32_t)P; ++q2) {
                        float pv = rg[(head - q2) & Rm];
                        res = (comb == 1) ? fminf(res, pv) : fmaxf(res, pv);
                    }
                }
                if (w < out_cap) {
                    out_key[w] = slot_to_key[s];
                    out_val[w] = res;
                    out_ts[w] = (q + 1) * pane_len - 1;  // window end - 1
                }
                ++w;
            }
        }
        pend_base[s] = hi + 1;
        st_head[s] = head;
        st_wsum[s] = wsum;
    }
}

// Wave-per-segment TB lift for MONOTONIC-ts batches: pane boundaries are
// found by lane-uniform binary search (log2(seg_len) ts reads per PANE
// instead of one random gather per ROW), and pane partials come from
// coalesced wave reduces.  The thread-per-segment kernel below walks
// tuples serially — TB measured 11x slower than CB at the flagship shape
// because of it (tools/tb_bench.py).  Gated on Batch::ts_mono; arbitrary
// ts orders take the serial kernel.
__global__ void k_tb_lift_wave(const uint32_t* seg_start, const uint32_t* seg_slot,
                               const int64_t* d_nseg, int64_t n,
                               const void* v_f32, int vdt,
                               const uint32_t* idx_sorted,
                               const int64_t* ts_orig, int64_t pane_len,
                               int64_t P, int64_t S, int comb, int pend_log2,
                               float* pend, int64_t* pend_base,
                               int64_t* last_pane, uint32_t* ignored,
                               uint32_t* overflow) {
    const int64_t nseg = *d_nseg;
    const uint32_t Rp = 1u << pend_log2;
    const uint32_t Pm = Rp - 1;
    const float ident = (comb == 1) ? INFINITY : (comb == 2 ? -INFINITY : 0.f);
    const int lane = threadIdx.x & 63;
    const int64_t wid = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const int64_t nw = ((int64_t)gridDim.x * blockDim.x) >> 6;
#define TLCOMB(a, b) ((comb == 1) ? fminf(a, b) : (comb == 2 ? fmaxf(a, b) : (a) + (b)))
    for (int64_t j = wid; j < nseg; j += nw) {
        const uint32_t slot = seg_slot[j];
        int64_t i = seg_start[j];
        const int64_t e = (j + 1 < nseg) ? seg_start[j + 1] : n;
        float* pd = pend + (size_t)slot * Rp;
        int64_t base = pend_base[slot];
        int64_t lastp = last_pane[slot];
        uint32_t ign = 0;
        while (i < e) {
            const int64_t t0 = ts_orig[idx_sorted[i]];
            const int64_t p = t0 / pane_len;
            if (base < 0) {
                int64_t w0 = t0 - P * pane_len + 1;
                w0 = w0 <= 0 ? 0 : (w0 + S * pane_len - 1) / (S * pane_len);
                base = w0 * S;
            }
            // first index with ts >= (p+1)*pane_len (ts nondecreasing in
            // arrival order; stable sort preserved it per segment)
            int64_t lo = i + 1, hi = e;
            const int64_t tlim = (p + 1) * pane_len;
            while (lo < hi) {
                int64_t mid = (lo + hi) >> 1;
                if (ts_orig[idx_sorted[mid]] < tlim) lo = mid + 1;
                else hi = mid;
            }
            const int64_t pe = lo;
            if (p < base) {
                ign += (uint32_t)(pe - i);
                i = pe;
                continue;
            }
            if (p - base >= (int64_t)Rp) {
                if (lane == 0) atomicAdd(overflow, 1u);
                i = pe;
                continue;
            }
            float part = ident;
            for (int64_t q = i + lane; q < pe; q += 64) {
                float x = (comb == 3)
                              ? 1.0f
                              : wfa_val_at(v_f32, vdt,
                                           vdt == 6 ? q : idx_sorted[q]);
                part = TLCOMB(part, x);
            }
            for (int o = 32; o; o >>= 1)
                part = TLCOMB(part, __shfl_xor(part, o, 64));
            if (lane == 0) {
                float* cell = &pd[(uint64_t)p & Pm];
                *cell = TLCOMB(*cell, part);
            }
            if (p > lastp) lastp = p;
            i = pe;
        }
        if (lane == 0) {
            pend_base[slot] = base;
            last_pane[slot] = lastp;
            if (ign) atomicAdd(ignored, ign);
        }
    }
#undef TLCOMB
}

// Pane-parallel TB lift for LOW segment counts (< ~4096): wave-per-segment
// starves the chip, so assign one wave per (segment, pane-in-span).
// k_tb_span: per segment, pane span + first pane from the first/last ts
// (monotonic).  k_tb_lift_pw: wave g -> binary search owning segment in
// the span scan, then binary search its pane's row range and wave-reduce.
// Panes with no rows write nothing.  Base/lastp/ignored bookkeeping runs
// in the span kernel (one thread per segment, reads only boundary ts).
__global__ void k_tb_span(const uint32_t* seg_start, const uint32_t* seg_slot,
                          const int64_t* d_nseg, int64_t n,
                          const uint32_t* idx_sorted, const int64_t* ts_orig,
                          int64_t pane_len, int64_t P, int64_t S,
                          int64_t* pend_base, int64_t* last_pane,
                          uint32_t* span_nf, int64_t* pfirst) {
    const int64_t nseg = *d_nseg;
    for (int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; j < nseg;
         j += gridDim.x * (int64_t)blockDim.x) {
        const uint32_t slot = seg_slot[j];
        const int64_t b = seg_start[j];
        const int64_t e = (j + 1 < nseg) ? seg_start[j + 1] : n;
        const int64_t t0 = ts_orig[idx_sorted[b]];
        const int64_t t1 = ts_orig[idx_sorted[e - 1]];
        int64_t base = pend_base[slot];
        if (base < 0) {
            int64_t w0 = t0 - P * pane_len + 1;
            w0 = w0 <= 0 ? 0 : (w0 + S * pane_len - 1) / (S * pane_len);
            base = w0 * S;
            pend_base[slot] = base;
        }
        const int64_t p0 = t0 / pane_len, p1 = t1 / pane_len;
        pfirst[j] = p0;
        span_nf[j] = (uint32_t)(p1 - p0 + 1);
        if (p1 > last_pane[slot]) last_pane[slot] = p1;
    }
}

__global__ void k_tb_lift_pw(const uint32_t* seg_start, const uint32_t* seg_slot,
                             const int64_t* d_nseg, const int64_t* d_total,
                             int64_t n, const void* v_f32, int vdt,
                             const uint32_t* idx_sorted, const int64_t* ts_orig,
                             int64_t pane_len, int comb, int pend_log2,
                             float* pend, const int64_t* pend_base,
                             const uint32_t* span_base, const int64_t* pfirst,
                             uint32_t* ignored, uint32_t* overflow) {
    const int64_t nseg = *d_nseg;
    const int64_t total = *d_total;
    const uint32_t Rp = 1u << pend_log2;
    const uint32_t Pm = Rp - 1;
    const float ident = (comb == 1) ? INFINITY : (comb == 2 ? -INFINITY : 0.f);
    const int lane = threadIdx.x & 63;
    const int64_t wid = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const int64_t nw = ((int64_t)gridDim.x * blockDim.x) >> 6;
#define TPCOMB(a, b) ((comb == 1) ? fminf(a, b) : (comb == 2 ? fmaxf(a, b) : (a) + (b)))
    for (int64_t g = wid; g < total; g += nw) {
        int64_t lo = 0, hi = nseg - 1;
        while (lo < hi) {
            int64_t mid = (lo + hi + 1) >> 1;
            if ((int64_t)span_base[mid] <= g) lo = mid;
            else hi = mid - 1;
        }
        const int64_t j = lo;
        const int64_t p = pfirst[j] + (g - (int64_t)span_base[j]);
        const uint32_t slot = seg_slot[j];
        const int64_t b = seg_start[j];
        const int64_t e = (j + 1 < nseg) ? seg_start[j + 1] : n;
        // row range of pane p: [lower_bound(p*L), lower_bound((p+1)*L))
        int64_t s0 = b, s1 = e;
        const int64_t tlo = p * pane_len;
        while (s0 < s1) {
            int64_t mid = (s0 + s1) >> 1;
            if (ts_orig[idx_sorted[mid]] < tlo) s0 = mid + 1;
            else s1 = mid;
        }
        int64_t e0 = s0, e1 = e;
        const int64_t thi = (p + 1) * pane_len;
        while (e0 < e1) {
            int64_t mid = (e0 + e1) >> 1;
            if (ts_orig[idx_sorted[mid]] < thi) e0 = mid + 1;
            else e1 = mid;
        }
        const int64_t pe = e0;
        if (pe == s0) continue;   // empty pane: no write
        const int64_t base = pend_base[slot];
        if (p < base) {
            if (lane == 0) atomicAdd(ignored, (uint32_t)(pe - s0));
            continue;
        }
        if (p - base >= (int64_t)Rp) {
            if (lane == 0) atomicAdd(overflow, 1u);
            continue;
        }
        float part = ident;
        for (int64_t q = s0 + lane; q < pe; q += 64) {
            float x = (comb == 3)
                          ? 1.0f
                          : wfa_val_at(v_f32, vdt, vdt == 6 ? q : idx_sorted[q]);
            part = TPCOMB(part, x);
        }
        for (int o = 32; o; o >>= 1) part = TPCOMB(part, __shfl_xor(part, o, 64));
        if (lane == 0) {
            float* cell = &pend[(size_t)slot * Rp + ((uint64_t)p & Pm)];
            *cell = TPCOMB(*cell, part);
        }
    }
#undef TPCOMB
}

// stage-split entries (hipEvent harness; production uses wfa_ffat_tb_round)
extern "C" void wfa_tb_lift_only(
    wfa_stream_t s, const uint32_t* seg_start, const uint32_t* seg_slot,
    const int64_t* d_nseg, int64_t n, const void* v_f32, int vdt,
    const uint32_t* idx_sorted, const int64_t* ts_orig, int64_t pane_len,
    int64_t P, int64_t S, int comb, int pend_log2, float* pend,
    int64_t* pend_base, int64_t* last_pane, uint32_t* ignored,
    uint32_t* overflow, int ts_mono) {
    hipStream_t st = (hipStream_t)s;
    if (ts_mono)
        hipLaunchKernelGGL(k_tb_lift_wave, dim3(WFA_MAX_BLOCKS / 2),
                           dim3(WFA_THREADS), 0, st, seg_start, seg_slot,
                           d_nseg, n, v_f32, vdt, idx_sorted, ts_orig, pane_len,
                           P, S, comb, pend_log2, pend, pend_base, last_pane,
                           ignored, overflow);
    else
        hipLaunchKernelGGL(k_tb_lift, dim3(WFA_MAX_BLOCKS / 8),
                           dim3(WFA_THREADS), 0, st, seg_start, seg_slot,
                           d_nseg, n, v_f32, vdt, idx_sorted, ts_orig, pane_len,
                           P, S, comb, pend_log2, pend, pend_base, last_pane,
                           ignored, overflow);
}

extern "C" void wfa_tb_countscan_only(wfa_stream_t s, const uint32_t* n_slots,
                                      int64_t limit_pane, int64_t* pend_base,
                                      int64_t* last_pane, uint32_t* st_head,
                                      int64_t P, int64_t S, uint32_t* nf,
                                      int64_t* d_out_n) {
    hipStream_t st = (hipStream_t)s;
    hipLaunchKernelGGL(k_tb_count, dim3(WFA_MAX_BLOCKS / 8), dim3(WFA_THREADS),
                       0, st, n_slots, limit_pane, pend_base, last_pane,
                       st_head, P, S, nf);
    hipLaunchKernelGGL(k_tb_scan, dim3(1), dim3(1024), 0, st, nf, n_slots,
                       d_out_n);
}

extern "C" void wfa_tb_advance_only(
    wfa_stream_t s, const uint32_t* n_slots, int64_t limit_pane,
    int64_t pane_len, int64_t P, int64_t S, int comb, int ring_log2,
    int pend_log2, float* pend, int64_t* pend_base, int64_t* last_pane,
    uint32_t* st_head, float* st_wsum, float* ring,
    const uint64_t* slot_to_key, uint32_t* nf, uint64_t* out_key,
    float* out_val, int64_t* out_ts, int64_t out_cap) {
    hipStream_t st = (hipStream_t)s;
    hipLaunchKernelGGL(k_tb_advance, dim3(WFA_MAX_BLOCKS / 8),
                       dim3(WFA_THREADS), 0, st, n_slots, limit_pane, pane_len,
                       P, S, comb, ring_log2, pend_log2, pend, pend_base,
                       last_pane, st_head, st_wsum, ring, slot_to_key, nf,
                       out_key, out_val, out_ts, out_cap);
}

extern "C" void wfa_ffat_tb_round(
    wfa_stream_t s, const uint32_t* seg_start, const uint32_t* seg_slot,
    const int64_t* d_nseg, int64_t n, const void* v_f32, int vdt,
    const uint32_t* idx_sorted, const int64_t* ts_orig, int64_t pane_len,
    int64_t P, int64_t S, int comb, int ring_log2, int pend_log2,
    int64_t limit_pane, float* pend, int64_t* pend_base, int64_t* last_pane,
    uint32_t* st_head, float* st_wsum, float* ring, const uint32_t* n_slots,
    const uint64_t* slot_to_key, uint32_t* nf, uint32_t* ignored,
    uint32_t* overflow, uint64_t* out_key, float* out_val, int64_t* out_ts,
    int64_t out_cap, int64_t* d_out_n, int ts_mono, uint32_t* span_nf,
    int64_t* pfirst, int64_t* d_total_spans) {
    hipStream_t st = (hipStream_t)s;
    if (n > 0) {
        if (ts_mono && span_nf) {
            // low-segment-count shape: pane-parallel lift (wave per pane)
            hipLaunchKernelGGL(k_tb_span, dim3(WFA_MAX_BLOCKS / 8),
                               dim3(WFA_THREADS), 0, st, seg_start, seg_slot,
                               d_nseg, n, idx_sorted, ts_orig, pane_len, P, S,
                               pend_base, last_pane, span_nf, pfirst);
            hipLaunchKernelGGL(k_fire_scan, dim3(1), dim3(1024), 0, st,
                               span_nf, d_nseg, d_total_spans);
            hipLaunchKernelGGL(k_tb_lift_pw, dim3(WFA_MAX_BLOCKS),
                               dim3(WFA_THREADS), 0, st, seg_start, seg_slot,
                               d_nseg, d_total_spans, n, v_f32, vdt,
                               idx_sorted, ts_orig, pane_len, comb, pend_log2,
                               pend, pend_base, span_nf, pfirst, ignored,
                               overflow);
        } else if (ts_mono)
            hipLaunchKernelGGL(k_tb_lift_wave, dim3(WFA_MAX_BLOCKS / 2),
                               dim3(WFA_THREADS), 0, st, seg_start, seg_slot,
                               d_nseg, n, v_f32, vdt, idx_sorted, ts_orig,
                               pane_len, P, S, comb, pend_log2, pend, pend_base,
                               last_pane, ignored, overflow);
        else
            hipLaunchKernelGGL(k_tb_lift, dim3(WFA_MAX_BLOCKS / 8),
                               dim3(WFA_THREADS), 0, st, seg_start, seg_slot,
                               d_nseg, n, v_f32, vdt, idx_sorted, ts_orig,
                               pane_len, P, S, comb, pend_log2, pend, pend_base,
                               last_pane, ignored, overflow);
    }
    hipLaunchKernelGGL(k_tb_count, dim3(WFA_MAX_BLOCKS / 8), dim3(WFA_THREADS), 0, st,
                       n_slots, limit_pane, pend_base, last_pane, st_head, P, S, nf);
    hipLaunchKernelGGL(k_tb_scan, dim3(1), dim3(1024), 0, st, nf, n_slots, d_out_n);
    hipLaunchKernelGGL(k_tb_advance, dim3(WFA_MAX_BLOCKS / 8), dim3(WFA_THREADS), 0,
                       st, n_slots, limit_pane, pane_len, P, S, comb, ring_log2,
                       pend_log2, pend, pend_base, last_pane, st_head, st_wsum, ring,
                       slot_to_key, nf, out_key, out_val, out_ts, out_cap);
}

// ===== FlatFAT-tree fold: O(log R) window query for large P =====
// Per-slot complete binary tree over R = 2^ring_log2 circular pane leaves
// (tree[slot*2R + node], node 1 = root, leaves at R..2R-1).  On each pane:
// write leaf (head & Rm), bubble up log R parents; on fire: circular range
// query of the last P leaves via the classic two-pointer FlatFAT walk
// (Tangwongsan VLDB'15 — reference flatfat.hpp:311-337 getResult).
__global__ void k_ffat_tree(const uint32_t* seg_start, const uint32_t* seg_slot,
                            const int64_t* d_nseg, int64_t n, const void* v_f32,
                            int vdt, const uint32_t* idx_sorted, const int64_t* ts_orig,
                            int64_t pane_len, int64_t P,
                            int64_t S, int comb, int ring_log2, int64_t* st_count,
                            uint32_t* st_fill, float* st_acc, float* tree,
                            uint32_t* st_head, const uint64_t* slot_to_key,
                            const uint32_t* fire_base,
                            uint64_t* out_key, float* out_val, int64_t* out_ts,
                            int64_t out_cap) {
    const int64_t nseg = *d_nseg;
    const uint32_t R = 1u << ring_log2;
    const uint32_t Rm = R - 1;
    const float ident = (comb == 1) ? INFINITY : (comb == 2 ? -INFINITY : 0.f);
#define TCOMB(a, b) ((comb == 1) ? fminf(a, b) : (comb == 2 ? fmaxf(a, b) : (a) + (b)))
    for (int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; j < nseg;
         j += gridDim.x * (int64_t)blockDim.x) {
        const uint32_t slot = seg_slot[j];
        int64_t i = seg_start[j];
        const int64_t e = (j + 1 < nseg) ? seg_start[j + 1] : n;
        uint32_t fill = st_fill[slot];
        float acc = st_acc[slot];
        uint32_t head = st_head[slot];
        int64_t w = fire_base[j];
        float* tr = tree + (size_t)slot * 2 * R;
        for (; i < e; ++i) {
            float x = (comb == 3)
                          ? 1.0f
                          : wfa_val_at(v_f32, vdt, vdt == 6 ? i : idx_sorted[i]);
            acc = TCOMB(acc, x);
            if (++fill == (uint32_t)pane_len) {
                uint32_t leaf = (head & Rm) + R;
                tr[leaf] = acc;
                for (uint32_t nd = leaf >> 1; nd >= 1; nd >>= 1)
                    tr[nd] = TCOMB(tr[2 * nd], tr[2 * nd + 1]);
                acc = ident;
                fill = 0;
                ++head;
                if (head >= (uint32_t)P && ((head - (uint32_t)P) % (uint32_t)S) == 0) {
                    // window = circular leaves [head-P, head)
                    uint32_t lo = (head - (uint32_t)P) & Rm;
                    uint32_t hi = (head - 1) & Rm;  // inclusive
                    float res = ident;
                    // classic FlatFAT inclusive range query [l, r] on one
                    // contiguous range; split circular into <=2 ranges
                    auto range_q = [&](uint32_t l, uint32_t r) {
                        uint32_t a = l + R, b2 = r + R;
                        float left = ident, right = ident;
                        while (a <= b2) {
                            if (a & 1) left = TCOMB(left, tr[a++]);
                            if (!(b2 & 1)) right = TCOMB(tr[b2--], right);
                            if (a > b2) break;
                            a >>= 1;
                            b2 >>= 1;
                        }
                        return TCOMB(left, right);
                    };
                    if (lo <= hi)
                        res = range_q(lo, hi);
                    else
                        res = TCOMB(range_q(lo, Rm), range_q(0, hi));
                    if (w < out_cap) {
                        out_key[w] = slot_to_key[slot];
                        out_val[w] = res;
                        out_ts[w] = ts_orig ? ts_orig[idx_sorted[i]] : 0;
                    }
                    ++w;
                }
            }
        }
        st_fill[slot] = fill;
        st_acc[slot] = acc;
        st_head[slot] = head;
        st_count[slot] += e - seg_start[j];
    }
#undef TCOMB
}

extern "C" void wfa_ffat_tree_fold(
    wfa_stream_t s, const uint32_t* seg_start, const uint32_t* seg_slot,
    const int64_t* d_nseg, int64_t n, const void* v_f32, int vdt,
    const uint32_t* idx_sorted, const int64_t* ts_orig,
    int64_t pane_len, int64_t P, int64_t S, int comb, int ring_log2, int64_t* st_count,
    uint32_t* st_fill, float* st_acc, float* tree, uint32_t* st_head,
    const uint64_t* slot_to_key, const uint32_t* fire_base, uint64_t* out_key,
    float* out_val, int64_t* out_ts, int64_t out_cap) {
    hipLaunchKernelGGL(k_ffat_tree, dim3(WFA_MAX_BLOCKS / 4), dim3(WFA_THREADS), 0,
                       (hipStream_t)s, seg_start, seg_slot, d_nseg, n, v_f32, vdt,
                       idx_sorted, ts_orig, pane_len, P, S, comb, ring_log2, st_count,
                       st_fill, st_acc, tree, st_head, slot_to_key, fire_base,
                       out_key, out_val, out_ts, out_cap);
}

// ===== per-key last-arrival timestamp (EOS flush emit ts) =====
// One thread per segment; a slot appears in at most one segment per batch,
// so the max-fold store is race-free.  Mirrors the CPU engine's
// ks.last_ts = max(last_ts, ts) (windows.cpp Keyed/FfatCpu state).
__global__ void k_seg_last_ts(const uint32_t* seg_start, const uint32_t* seg_slot,
                              const int64_t* d_nseg, int64_t n,
                              const uint32_t* idx_sorted, const int64_t* ts_orig,
                              int64_t* st_last) {
    const int64_t nseg = *d_nseg;
    for (int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; j < nseg;
         j += gridDim.x * (int64_t)blockDim.x) {
        const int64_t e = (j + 1 < nseg) ? seg_start[j + 1] : n;
        const int64_t t = ts_orig[idx_sorted[e - 1]];
        int64_t* p = &st_last[seg_slot[j]];
        if (t > *p) *p = t;
    }
}

extern "C" void wfa_seg_last_ts(wfa_stream_t s, const uint32_t* seg_start,
                                const uint32_t* seg_slot, const int64_t* d_nseg,
                                int64_t n, const uint32_t* idx_sorted,
                                const int64_t* ts_orig, int64_t* st_last) {
    hipLaunchKernelGGL(k_seg_last_ts, dim3(WFA_MAX_BLOCKS / 8), dim3(WFA_THREADS),
                       0, (hipStream_t)s, seg_start, seg_slot, d_nseg, n,
                       idx_sorted, ts_orig, st_last);
}

// ===== CB/TB EOS partial-window flush =====
// Mirrors the CPU FfatCpu EOS (windows.cpp on_eos): close the open partial
// pane (CB only: st_fill/st_acc non-null), then fire every remaining open
// window — starts q*S for q in [F(head), ...) with q*S < H — combining the
// ring cells [q*S, head) plus the partial pane.  H = head (+1 if a partial
// pane exists).  `cells` covers both layouts: ring (stride R, off 0) and
// FlatFAT-tree leaves (stride 2R, off R).
__global__ void k_cb_flush_count(const uint32_t* n_slots, int64_t P, int64_t S,
                                 const uint32_t* st_fill, const uint32_t* st_head,
                                 uint32_t* nf) {
    const int64_t ns = *n_slots;
    for (int64_t s = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; s < ns;
         s += gridDim.x * (int64_t)blockDim.x) {
        const uint64_t head = st_head[s];
        const uint64_t H = head + ((st_fill && st_fill[s]) ? 1 : 0);
        auto F = [&](uint64_t x) {
            return x < (uint64_t)P ? 0ull : (x - (uint64_t)P) / (uint64_t)S + 1ull;
        };
        const uint64_t q0 = F(head);
        nf[s] = (H > q0 * (uint64_t)S)
                    ? (uint32_t)((H - 1 - q0 * (uint64_t)S) / (uint64_t)S + 1)
                    : 0;
    }
}

__global__ void k_cb_flush(const uint32_t* n_slots, int64_t P, int64_t S,
                           int comb, int ring_log2, const uint32_t* st_fill,
                           const float* st_acc, const float* cells,
                           int64_t slot_stride, int64_t cell_off,
                           const uint32_t* st_head, const int64_t* st_last,
                           const uint64_t* slot_to_key, const uint32_t* nf,
                           int64_t out_base, uint64_t* out_key, float* out_val,
                           int64_t* out_ts, int64_t out_cap) {
    const int64_t ns = *n_slots;
    const uint32_t Rm = (1u << ring_log2) - 1;
    const float ident = (comb == 1) ? INFINITY : (comb == 2 ? -INFINITY : 0.f);
#define XCOMB(a, b) ((comb == 1) ? fminf(a, b) : (comb == 2 ? fmaxf(a, b) : (a) + (b)))
    for (int64_t s = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; s < ns;
         s += gridDim.x * (int64_t)blockDim.x) {
        const uint64_t head = st_head[s];
        const bool part = st_fill && st_fill[s];
        const uint64_t H = head + (part ? 1 : 0);
        auto F = [&](uint64_t x) {
            return x < (uint64_t)P ? 0ull : (x - (uint64_t)P) / (uint64_t)S + 1ull;
        };
        const float* rg = cells + (size_t)s * slot_stride + cell_off;
        int64_t w = nf[s];
        for (uint64_t q = F(head); q * (uint64_t)S < H; ++q) {
            float res = ident;
            for (uint64_t p = q * (uint64_t)S; p < head; ++p)
                res = XCOMB(res, rg[(uint32_t)p & Rm]);
            if (part) res = XCOMB(res, st_acc[s]);
            // window slice [out_base, out_base + out_cap): totals beyond one
            // output batch are flushed in host-driven pages
            if (w >= out_base && w - out_base < out_cap) {
                out_key[w - out_base] = slot_to_key[s];
                out_val[w - out_base] = res;
                out_ts[w - out_base] = st_last ? st_last[s] : 0;
            }
            ++w;
        }
    }
#undef XCOMB
}

// fire one page [out_base, out_base+out_cap) of the flush windows; run
// wfa_cb_flush_count + wfa_slot_scan first (nf = per-slot offsets)
extern "C" void wfa_ffat_cb_flush_fire(
    wfa_stream_t s, const uint32_t* n_slots, int64_t P, int64_t S, int comb,
    int ring_log2, const uint32_t* st_fill, const float* st_acc,
    const float* cells, int64_t slot_stride, int64_t cell_off,
    const uint32_t* st_head, const int64_t* st_last,
    const uint64_t* slot_to_key, const uint32_t* nf, int64_t out_base,
    uint64_t* out_key, float* out_val, int64_t* out_ts, int64_t out_cap) {
    hipLaunchKernelGGL(k_cb_flush, dim3(WFA_MAX_BLOCKS / 8), dim3(WFA_THREADS), 0,
                       (hipStream_t)s, n_slots, P, S, comb, ring_log2, st_fill,
                       st_acc, cells, slot_stride, cell_off, st_head, st_last,
                       slot_to_key, nf, out_base, out_key, out_val, out_ts,
                       out_cap);
}

// single-block exclusive scan over per-slot counts (shared with the JIT
// fold modules, which generate their own count/fire kernels)
extern "C" void wfa_slot_scan(wfa_stream_t s, uint32_t* nf,
                              const uint32_t* n_slots, int64_t* d_out_n) {
    hipLaunchKernelGGL(k_tb_scan, dim3(1), dim3(1024), 0, (hipStream_t)s, nf,
                       n_slots, d_out_n);
}

// value-independent pieces exported for the JIT fold path (whose lift/
// advance/fire kernels are hiprtc-generated but share these counters)
extern "C" void wfa_cb_flush_count(wfa_stream_t s, const uint32_t* n_slots,
                                   int64_t P, int64_t S, const uint32_t* st_fill,
                                   const uint32_t* st_head, uint32_t* nf) {
    hipLaunchKernelGGL(k_cb_flush_count, dim3(WFA_MAX_BLOCKS / 8),
                       dim3(WFA_THREADS), 0, (hipStream_t)s, n_slots, P, S,
                       st_fill, st_head, nf);
}

extern "C" void wfa_tb_count(wfa_stream_t s, const uint32_t* n_slots,
                             int64_t limit_pane, const int64_t* pend_base,
                             const int64_t* last_pane, const uint32_t* st_head,
                             int64_t P, int64_t S, uint32_t* nf) {
    hipLaunchKernelGGL(k_tb_count, dim3(WFA_MAX_BLOCKS / 8), dim3(WFA_THREADS), 0,
                       (hipStream_t)s, n_slots, limit_pane, pend_base, last_pane,
                       st_head, P, S, nf);
}

// ===== unkeyed full-batch reduce (reference reduce_gpu.hpp:269
// thrust::reduce path): two deterministic stages, no atomics =====
#define RA_BLOCKS 512

__global__ void k_reduce_all_p1(const void* v, int vdt, const int64_t* ts,
                                int64_t n, int comb, float* part,
                                int64_t* part_ts) {
    __shared__ float ls[WFA_THREADS];
    __shared__ int64_t lt[WFA_THREADS];
    float acc = (comb == 1) ? INFINITY : (comb == 2 ? -INFINITY : 0.0f);
    int64_t tmax = INT64_MIN;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (int64_t)blockDim.x) {
        float x = wfa_val_at(v, vdt, i);
        acc = (comb == 0) ? acc + x : (comb == 1 ? fminf(acc, x) : fmaxf(acc, x));
        if (ts) tmax = max(tmax, ts[i]);
    }
    ls[threadIdx.x] = acc;
    lt[threadIdx.x] = tmax;
    __syncthreads();
    for (int off = WFA_THREADS / 2; off > 0; off >>= 1) {
        if (threadIdx.x < off) {
            float o = ls[threadIdx.x + off];
            ls[threadIdx.x] = (comb == 0) ? ls[threadIdx.x] + o
                              : (comb == 1 ? fminf(ls[threadIdx.x], o)
                                           : fmaxf(ls[threadIdx.x], o));
            lt[threadIdx.x] = max(lt[threadIdx.x], lt[threadIdx.x + off]);
        }
        __syncthreads();
    }
    if (threadIdx.x == 0) {
        part[blockIdx.x] = ls[0];
        part_ts[blockIdx.x] = lt[0];
    }
}

__global__ void k_reduce_all_p2(const float* part, const int64_t* part_ts,
                                int64_t n_tuples, int comb, float* out,
                                int64_t* out_ts) {
    __shared__ float ls[RA_BLOCKS];
    __shared__ int64_t lt[RA_BLOCKS];
    ls[threadIdx.x] = part[threadIdx.x];
    lt[threadIdx.x] = part_ts[threadIdx.x];
    __syncthreads();
    for (int off = RA_BLOCKS / 2; off > 0; off >>= 1) {
        if (threadIdx.x < off) {
            float o = ls[threadIdx.x + off];
            ls[threadIdx.x] = (comb == 0) ? ls[threadIdx.x] + o
                              : (comb == 1 ? fminf(ls[threadIdx.x], o)
                                           : fmaxf(ls[threadIdx.x], o));
            lt[threadIdx.x] = max(lt[threadIdx.x], lt[threadIdx.x + off]);
        }
        __syncthreads();
    }
    if (threadIdx.x == 0) {
        out[0] = (comb == 3) ? (float)n_tuples : ls[0];
        out_ts[0] = lt[0];
    }
}

extern "C" void wfa_reduce_all(wfa_stream_t s, const void* v, int vdt,
                               const int64_t* ts, int64_t n, int comb,
                               float* scratch /* >= RA_BLOCKS */,
                               int64_t* scratch_ts /* >= RA_BLOCKS */,
                               float* out, int64_t* out_ts) {
    hipLaunchKernelGGL(k_reduce_all_p1, dim3(RA_BLOCKS), dim3(WFA_THREADS), 0,
                       (hipStream_t)s, v, vdt, ts, n, comb, scratch, scratch_ts);
    hipLaunchKernelGGL(k_reduce_all_p2, dim3(1), dim3(RA_BLOCKS), 0,
                       (hipStream_t)s, scratch, scratch_ts, n, comb, out, out_ts);
}

// ===== two-stage CB fold (auto-selected; see gpu_ops.cpp) =====
// Fused variant: one kernel, block per segment.  Waves cooperatively
// compute pane partials into LDS in chunks; wave 0 runs the window machine
// over each chunk.  No global pane_vals traffic, no extra launches.
#define P2_CHUNK 512

__global__ void k_pane2_fused(const uint32_t* seg_start, const uint32_t* seg_slot,
                              const int64_t* d_nseg, int64_t n,
                              const void* v_f32, int vdt,
                              const uint32_t* idx_sorted, const int64_t* ts_orig,
                              int64_t pane_len, int64_t P, int64_t S, int comb,
                              int ring_log2, int64_t* st_count, uint32_t* st_fill,
                              float* st_acc, float* ring, uint32_t* st_head,
                              float* st_wsum, const uint64_t* slot_to_key,
                              const uint32_t* fire_base,
                              uint64_t* out_key, float* out_val, int64_t* out_ts,
                              int64_t out_cap) {
    const int64_t nseg = *d_nseg;
    const uint32_t R = 1u << ring_log2;
    const uint32_t Rm = R - 1;
    const float ident = (comb == 1) ? INFINITY : (comb == 2 ? -INFINITY : 0.f);
    const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    const int nwaves = blockDim.x >> 6;
    __shared__ float pv[P2_CHUNK];
#define FCOMB(a, b) ((comb == 1) ? fminf(a, b) : (comb == 2 ? fmaxf(a, b) : (a) + (b)))
    for (int64_t j = blockIdx.x; j < nseg; j += gridDim.x) {
        const uint32_t slot = seg_slot[j];
        const int64_t b = seg_start[j];
        const int64_t e = (j + 1 < nseg) ? seg_start[j + 1] : n;
        const int64_t len = e - b;
        const uint32_t fill0 = st_fill[slot];
        const int64_t ncomplete = ((int64_t)fill0 + len) / pane_len;
        // wave-0 machine state (lane-uniform registers)
        float acc = st_acc[slot];
        uint32_t head = st_head[slot];
        float wsum = st_wsum[slot];
        int64_t w = fire_base[j];
        float* rg = ring + (size_t)slot * R;
        for (int64_t c0 = 0; c0 < ncomplete; c0 += P2_CHUNK) {
            const int64_t nc = min((int64_t)P2_CHUNK, ncomplete - c0);
            for (int64_t i = wave; i < nc; i += nwaves) {
                const int64_t gi = c0 + i;
                const int64_t lo = (gi == 0) ? b : b + gi * pane_len - (int64_t)fill0;
                const int64_t hi = b + (gi + 1) * pane_len - (int64_t)fill0;
                float a = ident;
                for (int64_t p = lo + lane; p < hi; p += 64) {
                    float x = (comb == 3)
                                  ? 1.0f
                                  : wfa_val_at(v_f32, vdt,
                                               vdt == 6 ? p : idx_sorted[p]);
                    a = FCOMB(a, x);
                }
                for (int o = 32; o; o >>= 1) a = FCOMB(a, __shfl_xor(a, o, 64));
                if (lane == 0) pv[i] = a;
            }
            __syncthreads();
            if (wave == 0) {
                for (int64_t i = 0; i < nc; ++i) {
                    const int64_t gi = c0 + i;
                    float pane = (gi == 0) ? FCOMB(acc, pv[i]) : pv[i];
                    rg[head & Rm] = pane;
                    if (comb == 0 || comb == 3) {
                        wsum += pane;
                        if (head >= (uint32_t)P)
                            wsum -= rg[(head - (uint32_t)P) & Rm];
                    }
                    ++head;
                    if (head >= (uint32_t)P &&
                        ((head - (uint32_t)P) % (uint32_t)S) == 0) {
                        float res;
                        if (comb == 0 || comb == 3) {
                            res = wsum;
                        } else {
                            float part = ident;
                            for (uint32_t q = lane; q < (uint32_t)P; q += 64)
                                part = FCOMB(part, rg[(head - 1 - q) & Rm]);
                            for (int o = 32; o; o >>= 1)
                                part = FCOMB(part, __shfl_xor(part, o, 64));
                            res = part;
                        }
                        if (lane == 0 && w < out_cap) {
                            out_key[w] = slot_to_key[slot];
                            out_val[w] = res;
                            const int64_t last =
                                b + (gi + 1) * pane_len - (int64_t)fill0 - 1;
                            out_ts[w] = ts_orig ? ts_orig[idx_sorted[last]] : 0;
                        }
                        ++w;
                    }
                }
            }
            __syncthreads();  // pv reused next chunk
        }
        // tail remainder -> new open acc
        {
            const int64_t lo =
                (ncomplete == 0) ? b : b + ncomplete * pane_len - (int64_t)fill0;
            float a = ident;
            if (wave == 0) {
                for (int64_t p = lo + lane; p < e; p += 64) {
                    float x = (comb == 3)
                                  ? 1.0f
                                  : wfa_val_at(v_f32, vdt,
                                               vdt == 6 ? p : idx_sorted[p]);
                    a = FCOMB(a, x);
                }
                for (int o = 32; o; o >>= 1) a = FCOMB(a, __shfl_xor(a, o, 64));
                const uint32_t newfill =
                    (uint32_t)(((int64_t)fill0 + len) - ncomplete * pane_len);
                if (lane == 0) {
                    st_fill[slot] = newfill;
                    st_acc[slot] = (ncomplete == 0) ? FCOMB(acc, a)
                                                    : (newfill ? a : ident);
                    st_head[slot] = head;
                    st_wsum[slot] = wsum;
                    st_count[slot] += len;
                }
            }
        }
        __syncthreads();  // state published before the block takes segment j+grid
    }
#undef FCOMB
}

extern "C" void wfa_ffat_cb_fold_fused(
    wfa_stream_t s, const uint32_t* seg_start, const uint32_t* seg_slot,
    const int64_t* d_nseg, int64_t n, const void* v_f32, int vdt,
    const uint32_t* idx_sorted, const int64_t* ts_orig, int64_t pane_len,
    int64_t P, int64_t S, int comb, int ring_log2, int64_t* st_count,
    uint32_t* st_fill, float* st_acc, float* ring, uint32_t* st_head,
    float* st_wsum, const uint64_t* slot_to_key, const uint32_t* fire_base,
    uint64_t* out_key, float* out_val, int64_t* out_ts, int64_t out_cap) {
    hipLaunchKernelGGL(k_pane2_fused, dim3(WFA_MAX_BLOCKS), dim3(256), 0,
                       (hipStream_t)s, seg_start, seg_slot, d_nseg, n, v_f32, vdt,
                       idx_sorted, ts_orig, pane_len, P, S, comb, ring_log2,
                       st_count, st_fill, st_acc, ring, st_head, st_wsum,
                       slot_to_key, fire_base, out_key, out_val, out_ts, out_cap);
}
