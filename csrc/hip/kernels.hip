// gfx950 device kernels: generator, elementwise map/filter, stream
// compaction, gather, fills.  CDNA4-native: wave64, 256-thread blocks,
// grid-stride loops capped at 2048 blocks (guide G11), coalesced SoA
// access; no Thrust, no CUDA-compat shims.
//
// Replaces (MI355X-native redesign, not a translation):
//   Stateless_MAPGPU_Kernel     (reference map_gpu.hpp:61-76)
//   Stateless_FILTERGPU_Kernel  (reference filter_gpu.hpp:72-88)
//   thrust::copy_if compaction  (reference filter_gpu.hpp:335-346)
#include <hip/hip_runtime.h>

#include "wfa_kernels.h"

#define WFA_THREADS 256
#define WFA_MAX_BLOCKS 2048

static inline int64_t nblk(int64_t n, int64_t per_thread = 1) {
    int64_t b = (n + WFA_THREADS * per_thread - 1) / (WFA_THREADS * per_thread);
    return b < 1 ? 1 : (b > WFA_MAX_BLOCKS ? WFA_MAX_BLOCKS : b);
}

__device__ __forceinline__ uint64_t mix64(uint64_t k) {
    k += 0x9e3779b97f4a7c15ULL;
    k = (k ^ (k >> 30)) * 0xbf58476d1ce4e5b9ULL;
    k = (k ^ (k >> 27)) * 0x94d049bb133111ebULL;
    return k ^ (k >> 31);
}

__device__ __forceinline__ float bf16_to_f32(uint16_t h) {
    union { uint32_t u; float f; } c;
    c.u = ((uint32_t)h) << 16;
    return c.f;
}
__device__ __forceinline__ uint16_t f32_to_bf16(float f) {
    union { uint32_t u; float f; } c;
    c.f = f;
    // round-to-nearest-even
    uint32_t lsb = (c.u >> 16) & 1;
    c.u += 0x7fff + lsb;
    return (uint16_t)(c.u >> 16);
}

// ===== generator =====
__global__ void k_gen(int64_t* ts, uint64_t* key, void* val, int vdt, int64_t n,
                      int64_t start, uint64_t seed, uint64_t n_keys) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (int64_t)blockDim.x) {
        uint64_t h = mix64(seed ^ (uint64_t)(start + i));
        ts[i] = start + i;
        key[i] = h % n_keys;
        uint64_t h2 = mix64(h);
        switch (vdt) {
            case 0: ((int64_t*)val)[i] = (int64_t)(h2 % 1000); break;           // I64
            case 2: ((float*)val)[i] = (float)(h2 >> 40) * (1.0f / 16777216.0f); break;  // F32
            case 5: ((uint16_t*)val)[i] = f32_to_bf16((float)(h2 >> 40) * (1.0f / 16777216.0f)); break;  // U16/bf16
            default: ((int64_t*)val)[i] = (int64_t)(h2 % 1000);
        }
    }
}

extern "C" void wfa_gen_batch(wfa_stream_t s, int64_t* ts, uint64_t* key, void* val,
                              int vdt, int64_t n, int64_t start, uint64_t seed,
                              uint64_t n_keys) {
    hipLaunchKernelGGL(k_gen, dim3(nblk(n)), dim3(WFA_THREADS), 0, (hipStream_t)s,
                       ts, key, val, vdt, n, start, seed, n_keys);
}

// ===== map =====
// spec 1: affine i64  (x = a*x + b)
// spec 2: affine f32
// spec 3: affine bf16 (computed in f32)
// spec 4: square i64
// spec 5: y = x*a + b with exp on f32 (decay demo)
__global__ void k_map_i64(int spec, int64_t* x, int64_t n, int64_t a, int64_t b) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (int64_t)blockDim.x) {
        int64_t v = x[i];
        x[i] = (spec == 4) ? v * v : a * v + b;
    }
}
__global__ void k_map_f32(int spec, float* x, int64_t n, float a, float b) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (int64_t)blockDim.x) {
        float v = x[i];
        x[i] = (spec == 5) ? __expf(v * a) + b : a * v + b;
    }
}
// bf16: vectorized 8-wide (guide G13 — scalar bf16 loads are 2-2.5x slower)
__global__ void k_map_bf16(uint16_t* x, int64_t n, float a, float b) {
    int64_t n8 = n / 8;
    using u16x8 = __attribute__((ext_vector_type(8))) uint16_t;
    u16x8* p = (u16x8*)x;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n8;
         i += gridDim.x * (int64_t)blockDim.x) {
        u16x8 v = p[i];
#pragma unroll
        for (int j = 0; j < 8; ++j) v[j] = f32_to_bf16(a * bf16_to_f32(v[j]) + b);
        p[i] = v;
    }
    // tail
    if (blockIdx.x == 0 && threadIdx.x < (n - n8 * 8)) {
        int64_t i = n8 * 8 + threadIdx.x;
        x[i] = f32_to_bf16(a * bf16_to_f32(x[i]) + b);
    }
}

extern "C" void wfa_map_apply(wfa_stream_t s, int spec, void* col, int dt, int64_t n,
                              double a, double b) {
    hipStream_t st = (hipStream_t)s;
    if (spec == 1 || spec == 4)
        hipLaunchKernelGGL(k_map_i64, dim3(nblk(n)), dim3(WFA_THREADS), 0, st, spec,
                           (int64_t*)col, n, (int64_t)a, (int64_t)b);
    else if (spec == 2 || spec == 5)
        hipLaunchKernelGGL(k_map_f32, dim3(nblk(n)), dim3(WFA_THREADS), 0, st, spec,
                           (float*)col, n, (float)a, (float)b);
    else if (spec == 3)
        hipLaunchKernelGGL(k_map_bf16, dim3(nblk(n, 8)), dim3(WFA_THREADS), 0, st,
                           (uint16_t*)col, n, (float)a, (float)b);
}

// ===== filter flags =====
__global__ void k_flags(int spec, const void* col, int dt, int64_t n, double a,
                        double b, uint32_t* flags) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (int64_t)blockDim.x) {
        bool keep = true;
        if (spec == 1) {
            int64_t v = ((const int64_t*)col)[i];
            keep = (v % (int64_t)a) != (int64_t)b;
        } else if (spec == 2) {
            keep = ((const float*)col)[i] > (float)a;
        } else if (spec == 3) {
            keep = bf16_to_f32(((const uint16_t*)col)[i]) >= (float)a;
        }
        flags[i] = keep ? 1u : 0u;
    }
}

extern "C" void wfa_filter_flags(wfa_stream_t s, int spec, const void* col, int dt,
                                 int64_t n, double a, double b, uint32_t* flags) {
    hipLaunchKernelGGL(k_flags, dim3(nblk(n)), dim3(WFA_THREADS), 0, (hipStream_t)s,
                       spec, col, dt, n, a, b, flags);
}

// ===== compaction (two-pass ordered: block counts -> scan -> scatter) =====
// ITEMS=8 contiguous per thread keeps within-thread order; block scan of
// per-thread counts keeps cross-thread order; global scan keeps cross-block
// order — fully stable.
#define CP_IPT 8
#define CP_PER_BLOCK (WFA_THREADS * CP_IPT)

__global__ void k_cp_count(const uint32_t* flags, int64_t n, uint32_t* blk_cnt) {
    int64_t base = (int64_t)blockIdx.x * CP_PER_BLOCK + threadIdx.x * CP_IPT;
    uint32_t c = 0;
#pragma unroll
    for (int j = 0; j < CP_IPT; ++j) {
        int64_t i = base + j;
        if (i < n) c += flags[i];
    }
    __shared__ uint32_t red[WFA_THREADS / 64];
    // wave reduce
    for (int off = 32; off; off >>= 1) c += __shfl_down(c, off, 64);
    if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = c;
    __syncthreads();
    if (threadIdx.x == 0) {
        uint32_t t = 0;
        for (int w = 0; w < WFA_THREADS / 64; ++w) t += red[w];
        blk_cnt[blockIdx.x] = t;
    }
}

// single-block exclusive scan (chunk-per-thread; see k_rs_scan note)
__global__ void k_scan_exclusive(uint32_t* a, int64_t m, int64_t* d_total) {
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
    uint32_t run = tot[threadIdx.x] - s;
    for (int64_t i = b0; i < b1; ++i) {
        uint32_t v = a[i];
        a[i] = run;
        run += v;
    }
    if (threadIdx.x == 1023 && d_total) *d_total = tot[1023];
}

__global__ void k_cp_scatter(const uint32_t* flags, int64_t n, const uint32_t* blk_off,
                             const int64_t* ts_in, int64_t* ts_out,
                             const uint64_t* key_in, uint64_t* key_out,
                             const void* const* cols_in, void* const* cols_out,
                             const int* col_esize, int n_cols) {
    int64_t base = (int64_t)blockIdx.x * CP_PER_BLOCK + threadIdx.x * CP_IPT;
    // per-thread count
    uint32_t c = 0;
#pragma unroll
    for (int j = 0; j < CP_IPT; ++j) {
        int64_t i = base + j;
        if (i < n) c += flags[i];
    }
    // block-exclusive scan of per-thread counts
    __shared__ uint32_t tc[WFA_THREADS];
    tc[threadIdx.x] = c;
    __syncthreads();
    for (int off = 1; off < WFA_THREADS; off <<= 1) {
        uint32_t t = (threadIdx.x >= off) ? tc[threadIdx.x - off] : 0;
        __syncthreads();
        tc[threadIdx.x] += t;
        __syncthreads();
    }
    uint32_t my_off = blk_off[blockIdx.x] + tc[threadIdx.x] - c;
    for (int j = 0; j < CP_IPT; ++j) {
        int64_t i = base + j;
        if (i >= n || !flags[i]) continue;
        int64_t w = my_off++;
        if (ts_out) ts_out[w] = ts_in[i];
        if (key_out) key_out[w] = key_in[i];
        for (int cc = 0; cc < n_cols; ++cc) {
            int es = col_esize[cc];
            const char* src = (const char*)cols_in[cc] + i * es;
            char* dst = (char*)cols_out[cc] + w * es;
            switch (es) {
                case 8: *(uint64_t*)dst = *(const uint64_t*)src; break;
                case 4: *(uint32_t*)dst = *(const uint32_t*)src; break;
                case 2: *(uint16_t*)dst = *(const uint16_t*)src; break;
                default: for (int k = 0; k < es; ++k) dst[k] = src[k];
            }
        }
    }
}

extern "C" void wfa_compact(wfa_stream_t s, int64_t n, const uint32_t* flags,
                            uint32_t* scan_tmp, const int64_t* ts_in, int64_t* ts_out,
                            const uint64_t* key_in, uint64_t* key_out,
                            const void* const* cols_in, void* const* cols_out,
                            const int* col_esize, int n_cols, int64_t* d_count) {
    hipStream_t st = (hipStream_t)s;
    int64_t nb = (n + CP_PER_BLOCK - 1) / CP_PER_BLOCK;
    hipLaunchKernelGGL(k_cp_count, dim3(nb), dim3(WFA_THREADS), 0, st, flags, n,
                       scan_tmp);
    hipLaunchKernelGGL(k_scan_exclusive, dim3(1), dim3(1024), 0, st, scan_tmp, nb,
                       d_count);
    hipLaunchKernelGGL(k_cp_scatter, dim3(nb), dim3(WFA_THREADS), 0, st, flags, n,
                       scan_tmp, ts_in, ts_out, key_in, key_out, cols_in, cols_out,
                       col_esize, n_cols);
}

// ===== gather by permutation =====
__global__ void k_gather(const uint32_t* idx, int64_t n, const void* v_in, void* v_out,
                         int esize, const int64_t* ts_in, int64_t* ts_out) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (int64_t)blockDim.x) {
        uint32_t j = idx[i];
        if (v_out) {
            if (esize == 4)
                ((uint32_t*)v_out)[i] = ((const uint32_t*)v_in)[j];
            else if (esize == 8)
                ((uint64_t*)v_out)[i] = ((const uint64_t*)v_in)[j];
            else
                ((uint16_t*)v_out)[i] = ((const uint16_t*)v_in)[j];
        }
        if (ts_out) ts_out[i] = ts_in[j];
    }
}

extern "C" void wfa_gather(wfa_stream_t s, const uint32_t* idx, int64_t n,
                           const void* v_in, void* v_out, int esize,
                           const int64_t* ts_in, int64_t* ts_out) {
    hipLaunchKernelGGL(k_gather, dim3(nblk(n)), dim3(WFA_THREADS), 0, (hipStream_t)s,
                       idx, n, v_in, v_out, esize, ts_in, ts_out);
}

// ===== fills / casts / bucketing =====
__global__ void k_fill_u64(uint64_t* p, uint64_t v, int64_t n) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (int64_t)blockDim.x)
        p[i] = v;
}
__global__ void k_fill_u32(uint32_t* p, uint32_t v, int64_t n) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (int64_t)blockDim.x)
        p[i] = v;
}
__global__ void k_fill_f32(float* p, float v, int64_t n) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (int64_t)blockDim.x)
        p[i] = v;
}
__global__ void k_iota_u32(uint32_t* p, int64_t n) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (int64_t)blockDim.x)
        p[i] = (uint32_t)i;
}
__global__ void k_cast(const void* in, int di, void* out, int d_o, int64_t n) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (int64_t)blockDim.x) {
        float v = 0;
        switch (di) {
            case 0: v = (float)((const int64_t*)in)[i]; break;
            case 2: v = ((const float*)in)[i]; break;
            case 5: v = bf16_to_f32(((const uint16_t*)in)[i]); break;
        }
        switch (d_o) {
            case 0: ((int64_t*)out)[i] = (int64_t)v; break;
            case 2: ((float*)out)[i] = v; break;
            case 5: ((uint16_t*)out)[i] = f32_to_bf16(v); break;
        }
    }
}
__global__ void k_flags_eq_u32(const uint32_t* v, int64_t n, uint32_t b,
                               uint32_t* flags) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (int64_t)blockDim.x)
        flags[i] = v[i] == b ? 1u : 0u;
}
__global__ void k_fill_f32_strided(float* p, float v, int64_t n, int64_t stride) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (int64_t)blockDim.x)
        p[i * stride] = v;
}
__global__ void k_bucket(const uint64_t* key, int64_t n, int world, uint32_t* dest) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (int64_t)blockDim.x)
        dest[i] = (uint32_t)(mix64(key[i]) % (uint64_t)world);
}

extern "C" {
void wfa_fill_u64(wfa_stream_t s, uint64_t* p, uint64_t v, int64_t n) {
    hipLaunchKernelGGL(k_fill_u64, dim3(nblk(n)), dim3(WFA_THREADS), 0, (hipStream_t)s, p, v, n);
}
void wfa_fill_u32(wfa_stream_t s, uint32_t* p, uint32_t v, int64_t n) {
    hipLaunchKernelGGL(k_fill_u32, dim3(nblk(n)), dim3(WFA_THREADS), 0, (hipStream_t)s, p, v, n);
}
void wfa_fill_f32(wfa_stream_t s, float* p, float v, int64_t n) {
    hipLaunchKernelGGL(k_fill_f32, dim3(nblk(n)), dim3(WFA_THREADS), 0, (hipStream_t)s, p, v, n);
}
void wfa_flags_eq_u32(wfa_stream_t s, const uint32_t* v, int64_t n, uint32_t b, uint32_t* flags) {
    hipLaunchKernelGGL(k_flags_eq_u32, dim3(nblk(n)), dim3(WFA_THREADS), 0, (hipStream_t)s, v, n, b, flags);
}
void wfa_fill_f32_strided(wfa_stream_t s, float* p, float v, int64_t n, int64_t stride) {
    hipLaunchKernelGGL(k_fill_f32_strided, dim3(nblk(n)), dim3(WFA_THREADS), 0, (hipStream_t)s, p, v, n, stride);
}
void wfa_iota_u32(wfa_stream_t s, uint32_t* p, int64_t n) {
    hipLaunchKernelGGL(k_iota_u32, dim3(nblk(n)), dim3(WFA_THREADS), 0, (hipStream_t)s, p, n);
}
void wfa_cast(wfa_stream_t s, const void* in, int dt_in, void* out, int dt_out, int64_t n) {
    hipLaunchKernelGGL(k_cast, dim3(nblk(n)), dim3(WFA_THREADS), 0, (hipStream_t)s, in, dt_in, out, dt_out, n);
}
void wfa_bucket_by_key(wfa_stream_t s, const uint64_t* key, int64_t n, int world, uint32_t* dest_out) {
    hipLaunchKernelGGL(k_bucket, dim3(nblk(n)), dim3(WFA_THREADS), 0, (hipStream_t)s, key, n, world, dest_out);
}
}

// ===== small-value histogram (all-to-all dest counts) =====
// nb <= 16: thread-private register counts -> LDS block reduce -> one
// global atomic per (block, bin).  A naive per-element global atomicAdd
// serializes on the hot bin (measured 45 ms for 4M rows at world=1).
__global__ void k_count_u32(const uint32_t* v, int64_t n, uint32_t* counts,
                            int nb) {
    uint32_t loc[16];
#pragma unroll
    for (int b = 0; b < 16; ++b) loc[b] = 0;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (int64_t)blockDim.x) {
        uint32_t d = v[i];
        if (d < 16u) loc[d]++;
        else atomicAdd(&counts[d], 1u);  // generic fallback
    }
    __shared__ uint32_t h[16];
    if (threadIdx.x < 16) h[threadIdx.x] = 0;
    __syncthreads();
#pragma unroll
    for (int b = 0; b < 16; ++b) {
        uint32_t c = loc[b];
        for (int off = 32; off; off >>= 1) c += __shfl_down(c, off, 64);
        if ((threadIdx.x & 63) == 0 && c) atomicAdd(&h[b], c);
    }
    __syncthreads();
    if (threadIdx.x < (uint32_t)min(nb, 16) && h[threadIdx.x])
        atomicAdd(&counts[threadIdx.x], h[threadIdx.x]);
}

// ===== full-row gather by permutation (ts + key + payload columns) =====
// cols table layout: [in_0..in_{nc-1}, out_0..out_{nc-1}] device pointers.
__global__ void k_gather_rows(const uint32_t* idx, int64_t n, const int64_t* ts_in,
                              int64_t* ts_out, const uint64_t* key_in,
                              uint64_t* key_out, const void* const* cols_in,
                              void* const* cols_out, const int* esize, int nc) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (int64_t)blockDim.x) {
        uint32_t j = idx[i];
        ts_out[i] = ts_in[j];
        key_out[i] = key_in[j];
        for (int c = 0; c < nc; ++c) {
            int es = esize[c];
            const char* src = (const char*)cols_in[c] + (int64_t)j * es;
            char* dst = (char*)cols_out[c] + i * es;
            switch (es) {
                case 8: *(uint64_t*)dst = *(const uint64_t*)src; break;
                case 4: *(uint32_t*)dst = *(const uint32_t*)src; break;
                case 2: *(uint16_t*)dst = *(const uint16_t*)src; break;
                default: for (int k = 0; k < es; ++k) dst[k] = src[k];
            }
        }
    }
}

// pack [counts..., wm, done] as i64 for the exchange metadata allgather
__global__ void k_pack_meta(const uint32_t* counts, int world, int64_t wm,
                            int64_t done, int64_t* meta) {
    int i = threadIdx.x;
    if (i < world) meta[i] = (int64_t)counts[i];
    if (i == world) meta[world] = wm;
    if (i == world + 1) meta[world + 1] = done;
}

extern "C" {
void wfa_pack_meta(wfa_stream_t s, const uint32_t* counts, int world, int64_t wm,
                   int64_t done, int64_t* meta) {
    hipLaunchKernelGGL(k_pack_meta, dim3(1), dim3(64), 0, (hipStream_t)s, counts,
                       world, wm, done, meta);
}
void wfa_count_u32(wfa_stream_t s, const uint32_t* v, int64_t n, uint32_t* counts,
                   int n_bins) {
    (void)hipMemsetAsync(counts, 0, 4 * n_bins, (hipStream_t)s);
    hipLaunchKernelGGL(k_count_u32, dim3(nblk(n, 8)), dim3(WFA_THREADS), 0,
                       (hipStream_t)s, v, n, counts, n_bins);
}
void wfa_gather_rows(wfa_stream_t s, const uint32_t* idx, int64_t n,
                     const int64_t* ts_in, int64_t* ts_out, const uint64_t* key_in,
                     uint64_t* key_out, const void* const* cols_in,
                     void* const* cols_out, const int* esize, int nc) {
    hipLaunchKernelGGL(k_gather_rows, dim3(nblk(n)), dim3(WFA_THREADS), 0,
                       (hipStream_t)s, idx, n, ts_in, ts_out, key_in, key_out,
                       cols_in, cols_out, esize, nc);
}
}
