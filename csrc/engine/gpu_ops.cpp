// GPU operator logic — host side.  Each GPU operator replica owns a HIP
// stream on its device and drives the gfx950 kernels in csrc/hip/ over
// device-resident SoA batches.  Chained GPU stages (map->filter->ffat)
// share one replica thread and one stream; queue edges between GPU ops
// synchronize through per-batch hipEvents.
//
// MI355X-native replacements (redesigns, not translations):
//   recycling_gpu.hpp pool      -> DeviceArena (size-bucketed HBM slabs,
//                                  one allocation per batch, 288 GB budget)
//   batch_gpu_t.hpp             -> Batch{loc=DEVICE} single-slab layout
//   map_gpu/filter_gpu/reduce_gpu/ffat_windows_gpu operator classes
//                               -> Gpu{Map,Filter,Reduce,Ffat}Logic below
#include <cstring>
#include <deque>
#include <map>
#include <tuple>

#include "engine.hpp"

#ifdef WFA_WITH_HIP
#include <rccl/rccl.h>

#include "../hip/wfa_kernels.h"
#include "gpu_common.hpp"

namespace wfa {

// hipEventCreate + hipHostMalloc cost ~0.1-1 ms each; recycle them across
// batches AND engines (a fresh PipeGraph would otherwise re-pin per batch
// inside the timed region).
struct AuxPool {
    std::mutex mu;
    std::vector<std::pair<void*, int64_t*>> free_list;  // (event, pinned slot)
    std::pair<void*, int64_t*> get(int device) {
        {
            std::lock_guard<std::mutex> g(mu);
            if (!free_list.empty()) {
                auto p = free_list.back();
                free_list.pop_back();
                return p;
            }
        }
        hipEvent_t ev;
        int64_t* slot;
        HIPCHK(hipSetDevice(device));
        HIPCHK(hipEventCreateWithFlags(&ev, hipEventDisableTiming));
        HIPCHK(hipHostMalloc((void**)&slot, 8, hipHostMallocDefault));
        return {ev, slot};
    }
    void put(void* ev, int64_t* slot) {
        std::lock_guard<std::mutex> g(mu);
        free_list.push_back({ev, slot});
    }
};
static AuxPool g_aux;

// ===== device batch alloc (core.cpp hooks) =====
static size_t dev_batch_bytes(const Schema& s, int64_t cap) {
    size_t b = 16 * cap;  // ts + key
    for (auto d : s.payload) b += ((dsize(d) * cap + 255) & ~size_t(255));
    return b;
}

Batch* gpu_alloc_batch(Pool& pool) {
    Batch* b = new Batch();
    b->capacity = pool.capacity;
    b->schema = pool.schema;
    b->loc = Loc::DEVICE;
    b->device = pool.device;
    size_t total = dev_batch_bytes(pool.schema, pool.capacity);
    char* base = (char*)arena(pool.device).get(total);
    b->ts = (int64_t*)base;
    b->key = (uint64_t*)(base + 8 * pool.capacity);
    char* p = base + 16 * pool.capacity;
    b->cols.resize(pool.schema.payload.size());
    for (size_t i = 0; i < b->cols.size(); ++i) {
        b->cols[i] = p;
        p += (dsize(pool.schema.payload[i]) * pool.capacity + 255) & ~size_t(255);
    }
    auto aux = g_aux.get(pool.device);
    b->ready_event = aux.first;
    b->lazy_count = aux.second;
    return b;
}

void gpu_resolve_count(Batch* b) {
    if (b->count >= 0) return;
    HIPCHK(hipEventSynchronize((hipEvent_t)b->ready_event));
    if (*b->lazy_count > b->capacity)
        throw std::runtime_error(
            "GPU output batch overflow (" + std::to_string(*b->lazy_count) +
            " rows > capacity " + std::to_string(b->capacity) +
            ") — raise the operator's out_batch");
    b->count = *b->lazy_count;
}

void gpu_free_batch(Batch* b) {
    arena(b->device).put(b->ts, dev_batch_bytes(b->schema, b->capacity));
    if (b->ready_event) g_aux.put(b->ready_event, b->lazy_count);
    delete b;
}

// D2D clone on the producing stream (broadcast fan-out of device batches —
// reference broadcast_emitter_gpu.hpp / splitting_emitter_gpu.hpp:186-200
// replicate per consumer; per-batch events are single-consumer here, so a
// clone per extra destination keeps the recycling protocol intact).
Batch* gpu_clone_batch(Batch* b) {
    gpu_resolve_count(b);  // count on host; contents valid once event fired
    Pool* p = b->pool;
    if (!p) throw std::runtime_error("gpu_clone_batch: batch has no pool");
    Batch* c = p->get();
    hipStream_t s = (hipStream_t)b->stream;
    if (c->ready_event)
        HIPCHK(hipStreamWaitEvent(s, (hipEvent_t)c->ready_event, 0));
    const int64_t n = b->count;
    HIPCHK(hipMemcpyAsync(c->ts, b->ts, 8 * n, hipMemcpyDeviceToDevice, s));
    HIPCHK(hipMemcpyAsync(c->key, b->key, 8 * n, hipMemcpyDeviceToDevice, s));
    for (size_t i = 0; i < b->cols.size(); ++i) {
        size_t es = dsize(b->schema.payload[i]);
        HIPCHK(hipMemcpyAsync(c->cols[i], b->cols[i], es * n,
                              hipMemcpyDeviceToDevice, s));
    }
    c->count = n;
    c->watermark = b->watermark;
    c->stream_tag = b->stream_tag;
    c->born_us = b->born_us;
    c->ts_mono = b->ts_mono;
    c->stream = s;
    HIPCHK(hipEventRecord((hipEvent_t)c->ready_event, s));
    return c;
}

// ===== device source (GPU-resident synthetic generator) =====
struct GpuSourceLogic : GpuLogicBase {
    int64_t len, n_keys, bsz;
    int vdt;
    uint64_t seed;
    int64_t pos = 0;
    GpuSourceLogic(int64_t l, int64_t k, int64_t b, int v, uint64_t sd, int dev,
                   Schema os) {
        len = l; n_keys = k; bsz = b; vdt = v; seed = sd;
        device = dev;
        out_schema = os;
        out_cap = b;
    }
    bool is_source() const override { return true; }
    bool source_step(EmitCtx& out, RuntimeCtx& ctx) override {
        int64_t t0 = now_us();
        ensure_init();
        if (pos >= len) return false;
        Batch* db = get_dev();
        int64_t tg = now_us() - t0;
        if (wfa_prof() && tg > 50)
            fprintf(stderr, "[prof] source get_dev wait %ld us\n", (long)tg);
        int64_t n = std::min<int64_t>(bsz, len - pos);
        wfa_gen_batch(stream, db->ts, db->key, db->cols[0], vdt, n, pos, seed, n_keys);
        dbg_sync(stream, "gen");
        db->count = n;
        db->ts_mono = true;  // ts = start + i
        db->born_us = now_us();
        pos += n;
        db->watermark = pos - 1;
        record_ready(db);
        if (ctx.stats) ctx.stats->num_kernels++;
        out.emit(db);
        return pos < len;
    }
};

// ===== stateless Map_GPU =====
struct GpuMapLogic : GpuLogicBase {
    int spec, col;
    double a, b;
    GpuMapLogic(int sp, int c, double a_, double b_, int dev, Schema os) {
        spec = sp; col = c; a = a_; b = b_;
        device = dev;
        out_schema = os;
    }
    void process(Batch* in, EmitCtx& out, RuntimeCtx& ctx) override {
        ensure_init();
        Batch* db = input_on_device(in, ctx);
        wfa_map_apply(stream, spec, db->cols[col], (int)db->schema.payload[col],
                      db->count, a, b);
        record_ready(db);
        if (ctx.stats) ctx.stats->num_kernels++;
        out.emit(db);
    }
};

// ===== stateless Filter_GPU with on-device compaction =====
struct GpuFilterLogic : GpuLogicBase {
    int spec, col;
    double a, b;
    uint32_t* d_flags = nullptr;
    uint32_t* d_scan = nullptr;
    int64_t* d_cnt = nullptr;
    void** d_colptrs = nullptr;  // [2*ncols] in,out
    int* d_esize = nullptr;
    GpuFilterLogic(int sp, int c, double a_, double b_, int dev, Schema os,
                   int64_t cap) {
        spec = sp; col = c; a = a_; b = b_;
        device = dev;
        out_schema = os;
        out_cap = cap;
    }
    void init_device() override {
        d_flags = (uint32_t*)arena(device).get(4 * out_cap);
        d_scan = (uint32_t*)arena(device).get(4 * (out_cap / 2048 + 2));
        d_cnt = (int64_t*)arena(device).get(64);
        size_t nc = out_schema.payload.size();
        d_colptrs = (void**)arena(device).get(16 * (nc + 1));
        d_esize = (int*)arena(device).get(4 * (nc + 1));
        std::vector<int> es;
        for (auto d : out_schema.payload) es.push_back((int)dsize(d));
        HIPCHK(hipMemcpy(d_esize, es.data(), 4 * nc, hipMemcpyHostToDevice));
    }
    void process(Batch* in, EmitCtx& out, RuntimeCtx& ctx) override {
        ensure_init();
        Batch* db = input_on_device(in, ctx);
        int64_t n = db->count;
        if (n > out_cap)
            throw std::runtime_error("filter input batch > out_batch capacity");
        Batch* ob = get_dev();
        wfa_filter_flags(stream, spec, db->cols[col], (int)db->schema.payload[col], n,
                         a, b, d_flags);
        // column pointer table (in then out)
        size_t nc = db->cols.size();
        std::vector<void*> ptrs(2 * nc);
        for (size_t c = 0; c < nc; ++c) {
            ptrs[c] = db->cols[c];
            ptrs[nc + c] = ob->cols[c];
        }
        HIPCHK(hipMemcpyAsync(d_colptrs, ptrs.data(), 8 * 2 * nc,
                              hipMemcpyHostToDevice, stream));
        wfa_compact(stream, n, d_flags, d_scan, db->ts, ob->ts, db->key, ob->key,
                    (const void* const*)d_colptrs, (void* const*)(d_colptrs + nc),
                    d_esize, (int)nc, d_cnt);
        HIPCHK(hipMemcpyAsync(ob->lazy_count, d_cnt, 8, hipMemcpyDeviceToHost, stream));
        ob->count = -1;
        ob->watermark = db->watermark;
        ob->stream_tag = db->stream_tag;
        ob->born_us = db->born_us;
        ob->ts_mono = db->ts_mono;  // compaction preserves row order
        if (ctx.stats) ctx.stats->num_kernels += 3;
        release_after_use(db);
        record_ready(ob);
        out.emit(ob);
    }
};

// ===== shared keyed front half: slot -> sort -> gather -> segments =====
// (struct declared in gpu_common.hpp; shared with gpu_jit.cpp)
void KeyedScratch::alloc(int dev, int64_t cap_, int64_t mk, hipStream_t s) {
    cap = cap_;
    max_keys = mk;
    table_cap = 1;
    while (table_cap < 2 * mk) table_cap <<= 1;
    bits = 1;
    while ((1ll << bits) < mk + 1) ++bits;
    // floor 5: the 4-bit sort path ignores base_shift (VIK packs the value
    // in the low 16 bits) — always take the 8-bit machinery
    if (bits < 5) bits = 5;
    auto& A = arena(dev);
    tab = (uint64_t*)A.get(16 * table_cap);
    d_nslots = (uint32_t*)A.get(64);
    slot_to_key = (uint64_t*)A.get(8 * mk);
    slot = (uint32_t*)A.get(4 * cap);
    idx = (uint32_t*)A.get(4 * cap);
    slot_t = (uint32_t*)A.get(4 * cap);
    idx_t = (uint32_t*)A.get(4 * cap);
    hist = (uint32_t*)A.get(4 * wfa_sort_hist_u32(cap));
    seg_start = (uint32_t*)A.get(4 * cap);
    seg_slot = (uint32_t*)A.get(4 * cap);
    d_nseg = (int64_t*)A.get(64);
    v_sorted = (float*)A.get(4 * cap);   // reused as fire-offset scratch
    v_f32 = (float*)A.get(4 * cap);
    wfa_fill_u64(s, tab, ~0ULL, 2 * table_cap);
    wfa_fill_u32(s, d_nslots, 0, 1);
    if (dense) {
        d_overflow = (uint32_t*)A.get(64);
        wfa_fill_u32(s, d_overflow, 0, 1);
        wfa_iota_u64(s, slot_to_key, mk);  // identity mapping
    }
    // table_cap (>= 2*mk) entries: covers every legal slot id with margin
    if (max_keys <= 65536) by_slot = (uint32_t*)A.get(4 * table_cap);
}

void KeyedScratch::segs(hipStream_t s, const uint32_t* slot_sorted, int64_t n,
                        int shr) {
    if (by_slot)
        wfa_segments_dense(s, slot_sorted, n, by_slot, table_cap, hist,
                           seg_start, seg_slot, d_nseg, shr);
    else
        wfa_segments(s, slot_sorted, n, hist, seg_start, seg_slot, d_nseg, shr);
}

void KeyedScratch::check_dense_overflow() {
    if (!dense) return;
    uint32_t ovf = 0;
    HIPCHK(hipMemcpy(&ovf, d_overflow, 4, hipMemcpyDeviceToHost));
    if (ovf)
        throw std::runtime_error(
            "withDenseKeys: a key >= max_keys arrived — dense mode requires "
            "integer keys in [0, max_keys); use the default hashed mode");
}

// sorts (slot, idx) pairs and fills segments; values stay unsorted and
// are read through idx_sorted (saves the gather round trip) — EXCEPT in
// value-in-key (VIK) mode: for bf16 value columns with <= 65535 keys the
// bf16 bits ride in the low 16 bits of the sort key (radix passes sort
// on bits 16.. only, stability keeps per-key order), so the fold reads
// values SEQUENTIALLY from the sorted key array instead of gathering
// one cache line per tuple.  Numerics are identical (same bf16 bits).
// vcol < 0: no value cast (stateful map/filter operate in place)
void KeyedScratch::group(hipStream_t s, Batch* db, int vcol, RuntimeCtx& ctx,
                         bool want_vik, bool want_carry) {
        int64_t n = db->count;
        if (n > cap)
            throw std::runtime_error("batch larger than keyed scratch capacity — "
                                     "set the GPU op's out_batch >= upstream batch");
        bool vik = want_vik && vcol >= 0 && max_keys <= 65535 &&
                   (int)db->schema.payload[vcol] == 5;
        if (vik) {
            uint32_t *os, *oi;
            if (dense) {
                // fused: slot write + sort pass-0 histogram in one kernel
                wfa_key_dense_h(s, db->key, n, max_keys, slot, d_nslots,
                                d_overflow, (const uint16_t*)db->cols[vcol],
                                hist, /*shift=*/16);
                wfa_sort_pairs2_ph(s, slot, idx, slot_t, idx_t, nullptr,
                                   nullptr, hist, n, bits, &os, &oi, nullptr,
                                   /*implicit_iota=*/1, /*base_shift=*/16);
            } else {
                wfa_key_to_slot_h(s, db->key, n, tab, d_nslots, table_cap, slot,
                                  slot_to_key, (const uint16_t*)db->cols[vcol],
                                  hist, /*shift=*/16);
                wfa_sort_pairs2_ph(s, slot, idx, slot_t, idx_t, nullptr,
                                   nullptr, hist, n, bits, &os, &oi, nullptr,
                                   /*implicit_iota=*/1, /*base_shift=*/16);
            }
            idx_sorted = oi;
            v_as_f32 = os;
            v_dt = 6;
            segs(s, os, n, /*shr=*/16);
            if (ctx.stats) ctx.stats->num_kernels += 5 + 3 * ((bits + 3) / 4);
            return;
        }
        if (dense)
            wfa_key_dense_h(s, db->key, n, max_keys, slot, d_nslots, d_overflow,
                            nullptr, hist, /*shift=*/0);
        else
            wfa_key_to_slot_h(s, db->key, n, tab, d_nslots, table_cap, slot,
                              slot_to_key, nullptr, hist, /*shift=*/0);
        dbg_sync(s, "key_to_slot");
        if (want_carry && vcol >= 0 && (int)db->schema.payload[vcol] == 2) {
            // carry the f32 value bits as a second sort payload: the fold
            // then reads values SEQUENTIALLY at the sorted position instead
            // of one random cache line per tuple (A/B: wins for the
            // latency-bound segmented reduce; the FFAT folds hide the
            // gather and keep the cheaper LDS scatter instead)
            wfa_cast(s, db->cols[vcol], 2, v_f32, 2, n);  // sort ping buffer
            uint32_t *os, *oi, *ov;
            wfa_sort_pairs2_ph(s, slot, idx, slot_t, idx_t, (uint32_t*)v_f32,
                               (uint32_t*)v_sorted, hist, n, bits, &os, &oi,
                               &ov, /*implicit_iota=*/1, /*base_shift=*/0);
            idx_sorted = oi;
            v_as_f32 = ov;
            v_dt = 7;
            segs(s, os, n, 0);
            if (ctx.stats) ctx.stats->num_kernels += 6 + 3 * ((bits + 3) / 4);
            return;
        }
        uint32_t *os, *oi;
        wfa_sort_pairs2_ph(s, slot, idx, slot_t, idx_t, nullptr, nullptr,
                           hist, n, bits, &os, &oi, nullptr,
                           /*implicit_iota=*/1, /*base_shift=*/0);
        idx_sorted = oi;
        if (vcol >= 0) {
            // A/B-measured: reading values THROUGH idx_sorted in the folds
            // beats carrying them as a sort payload (the wave fold hides the
            // gather; the extra scatter writes did not pay — 11.5 vs 13.0
            // B tuples/s).  f32/bf16 columns are read in place (bf16
            // converts inline in the fold: half the gather bytes, no cast
            // kernel); other dtypes cast to f32 once.
            int vdt = (int)db->schema.payload[vcol];
            if (vdt == 2 || vdt == 5) {
                v_as_f32 = db->cols[vcol];
                v_dt = vdt;
            } else {
                wfa_cast(s, db->cols[vcol], vdt, v_f32, 2, n);
                v_as_f32 = v_f32;
                v_dt = 2;
            }
        }
        dbg_sync(s, "sort");
        segs(s, os, n, /*shr=*/0);
        dbg_sync(s, "segments");
        if (ctx.stats) ctx.stats->num_kernels += 5 + 3 * ((bits + 3) / 4);
}

// ===== hiprtc JIT: user device logic (reference MAP_GPU/FILTER_GPU
// accept arbitrary __device__ lambdas; here a C expression over
// (v, ts, key) is runtime-compiled for gfx950 and cached).  The general
// multi-field fold JIT (Reduce_GPU / FFAT combines) lives in gpu_jit.cpp;
// this file keeps only the scalar map/filter expression path.
hipModule_t jit_module(const std::string& src, int device) {
    static std::mutex mu;
    static std::map<std::string, hipModule_t> cache;
    std::lock_guard<std::mutex> g(mu);
    auto it = cache.find(src);
    if (it != cache.end()) return it->second;
    hiprtcProgram prog;
    RTCCHK(hiprtcCreateProgram(&prog, src.c_str(), "wfa_jit.hip", 0, nullptr,
                               nullptr));
    hipDeviceProp_t props;
    HIPCHK(hipGetDeviceProperties(&props, device));
    std::string arch = std::string("--offload-arch=") + props.gcnArchName;
    const char* opts[] = {arch.c_str(), "-O3"};
    hiprtcResult cr = hiprtcCompileProgram(prog, 2, opts);
    if (cr != HIPRTC_SUCCESS) {
        size_t lsz = 0;
        hiprtcGetProgramLogSize(prog, &lsz);
        std::string log(lsz, '\0');
        hiprtcGetProgramLog(prog, log.data());
        hiprtcDestroyProgram(&prog);
        throw std::runtime_error("JIT device logic failed to compile:\n" + log);
    }
    size_t csz = 0;
    RTCCHK(hiprtcGetCodeSize(prog, &csz));
    std::vector<char> code(csz);
    RTCCHK(hiprtcGetCode(prog, code.data()));
    hiprtcDestroyProgram(&prog);
    hipModule_t mod;
    HIPCHK(hipModuleLoadData(&mod, code.data()));
    cache[src] = mod;
    return mod;
}

hipFunction_t jit_fn(hipModule_t mod, const char* name) {
    hipFunction_t fn;
    HIPCHK(hipModuleGetFunction(&fn, mod, name));
    return fn;
}

std::string jit_type(DType d) {
    switch (d) {
        case DType::I64: return "long long";
        case DType::F64: return "double";
        case DType::F32: return "float";
        case DType::U64: return "unsigned long long";
        case DType::I32: return "int";
        default: throw std::runtime_error("jit: unsupported column dtype");
    }
}

struct JitKernel {
    hipModule_t mod = nullptr;
    hipFunction_t fn = nullptr;
};

static JitKernel jit_compile(const std::string& src, int device) {
    JitKernel k;
    k.mod = jit_module(src, device);
    k.fn = jit_fn(k.mod, "wfa_jit");
    return k;
}

// expr sees: v (value column, mutable type T), ts (long long), key (u64)
static std::string jit_source(const std::string& expr, DType dt, bool filter) {
    std::string T = jit_type(dt);
    std::string body = filter
        ? "        flags[i] = (" + expr + ") ? 1u : 0u;\n"
        : "        v_[i] = (" + expr + ");\n";
    return "typedef long long i64; typedef unsigned long long u64;\n"
           "extern \"C\" __global__ void wfa_jit(" + T + "* v_, const i64* ts_,\n"
           "        const u64* key_, unsigned int* flags, i64 n) {\n"
           "    for (i64 i = blockIdx.x * (i64)blockDim.x + threadIdx.x; i < n;\n"
           "         i += gridDim.x * (i64)blockDim.x) {\n"
           "        " + T + " v = v_[i]; i64 ts = ts_[i]; u64 key = key_[i];\n"
           "        (void)v; (void)ts; (void)key;\n" + body +
           "    }\n}\n";
}

static void jit_launch(const JitKernel& k, hipStream_t stream, void* col,
                       const int64_t* ts, const uint64_t* key, uint32_t* flags,
                       int64_t n) {
    void* args[] = {&col, (void*)&ts, (void*)&key, &flags, &n};
    unsigned nb = (unsigned)std::min<int64_t>((n + 255) / 256, 2048);
    HIPCHK(hipModuleLaunchKernel(k.fn, nb, 1, 1, 256, 1, 1, 0, stream, args,
                                 nullptr));
}

struct GpuJitMapLogic : GpuLogicBase {
    std::string expr;
    int col;
    JitKernel k;
    GpuJitMapLogic(std::string e, int c, int dev, Schema os) : expr(std::move(e)), col(c) {
        device = dev;
        out_schema = os;
    }
    void init_device() override {
        k = jit_compile(jit_source(expr, out_schema.payload[col], false), device);
    }
    void process(Batch* in, EmitCtx& out, RuntimeCtx& ctx) override {
        ensure_init();
        Batch* db = input_on_device(in, ctx);
        jit_launch(k, stream, db->cols[col], db->ts, db->key, nullptr, db->count);
        record_ready(db);
        if (ctx.stats) ctx.stats->num_kernels++;
        out.emit(db);
    }
};

struct GpuJitFilterLogic : GpuLogicBase {
    std::string expr;
    int col;
    JitKernel k;
    uint32_t* d_flags = nullptr;
    uint32_t* d_scan = nullptr;
    int64_t* d_cnt = nullptr;
    void** d_colptrs = nullptr;
    int* d_esize = nullptr;
    GpuJitFilterLogic(std::string e, int c, int dev, Schema os, int64_t cap)
        : expr(std::move(e)), col(c) {
        device = dev;
        out_schema = os;
        out_cap = cap;
    }
    void init_device() override {
        k = jit_compile(jit_source(expr, out_schema.payload[col], true), device);
        auto& A = arena(device);
        d_flags = (uint32_t*)A.get(4 * out_cap);
        d_scan = (uint32_t*)A.get(4 * (out_cap / 2048 + 2));
        d_cnt = (int64_t*)A.get(64);
        size_t nc = out_schema.payload.size();
        d_colptrs = (void**)A.get(16 * (nc + 1));
        d_esize = (int*)A.get(4 * (nc + 1));
        std::vector<int> es;
        for (auto d : out_schema.payload) es.push_back((int)dsize(d));
        HIPCHK(hipMemcpy(d_esize, es.data(), 4 * nc, hipMemcpyHostToDevice));
    }
    void process(Batch* in, EmitCtx& out, RuntimeCtx& ctx) override {
        ensure_init();
        Batch* db = input_on_device(in, ctx);
        int64_t n = db->count;
        jit_launch(k, stream, db->cols[col], db->ts, db->key, d_flags, n);
        Batch* ob = get_dev();
        size_t nc = db->cols.size();
        std::vector<void*> ptrs(2 * nc);
        for (size_t c = 0; c < nc; ++c) {
            ptrs[c] = db->cols[c];
            ptrs[nc + c] = ob->cols[c];
        }
        HIPCHK(hipMemcpyAsync(d_colptrs, ptrs.data(), 8 * 2 * nc,
                              hipMemcpyHostToDevice, stream));
        wfa_compact(stream, n, d_flags, d_scan, db->ts, ob->ts, db->key, ob->key,
                    (const void* const*)d_colptrs, (void* const*)(d_colptrs + nc),
                    d_esize, (int)nc, d_cnt);
        HIPCHK(hipMemcpyAsync(ob->lazy_count, d_cnt, 8, hipMemcpyDeviceToHost, stream));
        ob->count = -1;
        ob->watermark = db->watermark;
        ob->stream_tag = db->stream_tag;
        ob->born_us = db->born_us;
        ob->ts_mono = db->ts_mono;  // compaction preserves row order
        if (ctx.stats) ctx.stats->num_kernels += 4;
        release_after_use(db);
        record_ready(ob);
        out.emit(ob);
    }
};

// ===== stateful Map_GPU / Filter_GPU: keyed device state =====
// Reference map_gpu.hpp:106-295 keeps per-key state objects in a shared
// tbb map with a spinlock serializing replicas; here state is a dense
// per-slot arena owned by the replica, advanced in key order by
// wfa_stateful_apply over the batch's segments.
struct GpuStatefulMapLogic : GpuLogicBase {
    int spec, col;
    double a, b;
    int64_t max_keys;
    KeyedScratch ks;
    double* d_state = nullptr;
    GpuStatefulMapLogic(int sp, int c, double a_, double b_, int64_t mk, int dev,
                        Schema os, int64_t cap) {
        spec = sp; col = c; a = a_; b = b_; max_keys = mk;
        device = dev;
        out_schema = os;
        out_cap = cap;
    }
    void init_device() override {
        ks.alloc(device, out_cap, max_keys, stream);
        d_state = (double*)arena(device).get(8 * max_keys);
        HIPCHK(hipMemsetAsync(d_state, 0, 8 * max_keys, stream));
    }
    void process(Batch* in, EmitCtx& out, RuntimeCtx& ctx) override {
        ensure_init();
        Batch* db = input_on_device(in, ctx);
        ks.group(stream, db, -1, ctx);
        wfa_stateful_apply(stream, ks.seg_start, ks.seg_slot, ks.d_nseg, db->count,
                           ks.idx_sorted, db->cols[col],
                           (int)db->schema.payload[col], spec, 0, a, b, d_state,
                           nullptr);
        dbg_sync(stream, "stateful_apply");
        record_ready(db);
        if (ctx.stats) ctx.stats->num_kernels += 1;
        out.emit(db);
    }
};

struct GpuStatefulFilterLogic : GpuLogicBase {
    int spec, col;
    double a, b;
    int64_t max_keys;
    KeyedScratch ks;
    double* d_state = nullptr;
    uint32_t* d_flags = nullptr;
    uint32_t* d_scan = nullptr;
    int64_t* d_cnt = nullptr;
    void** d_colptrs = nullptr;
    int* d_esize = nullptr;
    GpuStatefulFilterLogic(int sp, int c, double a_, double b_, int64_t mk, int dev,
                           Schema os, int64_t cap) {
        spec = sp; col = c; a = a_; b = b_; max_keys = mk;
        device = dev;
        out_schema = os;
        out_cap = cap;
    }
    void init_device() override {
        ks.alloc(device, out_cap, max_keys, stream);
        auto& A = arena(device);
        d_state = (double*)A.get(8 * max_keys);
        // dedup compares x != state: NaN init keeps every key's first tuple
        HIPCHK(hipMemsetAsync(d_state, spec == 1 ? 0xFF : 0, 8 * max_keys, stream));
        d_flags = (uint32_t*)A.get(4 * out_cap);
        d_scan = (uint32_t*)A.get(4 * (out_cap / 2048 + 2));
        d_cnt = (int64_t*)A.get(64);
        size_t nc = out_schema.payload.size();
        d_colptrs = (void**)A.get(16 * (nc + 1));
        d_esize = (int*)A.get(4 * (nc + 1));
        std::vector<int> es;
        for (auto d : out_schema.payload) es.push_back((int)dsize(d));
        HIPCHK(hipMemcpy(d_esize, es.data(), 4 * nc, hipMemcpyHostToDevice));
    }
    void process(Batch* in, EmitCtx& out, RuntimeCtx& ctx) override {
        ensure_init();
        Batch* db = input_on_device(in, ctx);
        int64_t n = db->count;
        ks.group(stream, db, -1, ctx);
        wfa_stateful_apply(stream, ks.seg_start, ks.seg_slot, ks.d_nseg, n,
                           ks.idx_sorted, db->cols[col],
                           (int)db->schema.payload[col], spec, 1, a, b, d_state,
                           d_flags);
        Batch* ob = get_dev();
        size_t nc = db->cols.size();
        std::vector<void*> ptrs(2 * nc);
        for (size_t c = 0; c < nc; ++c) {
            ptrs[c] = db->cols[c];
            ptrs[nc + c] = ob->cols[c];
        }
        HIPCHK(hipMemcpyAsync(d_colptrs, ptrs.data(), 8 * 2 * nc,
                              hipMemcpyHostToDevice, stream));
        wfa_compact(stream, n, d_flags, d_scan, db->ts, ob->ts, db->key, ob->key,
                    (const void* const*)d_colptrs, (void* const*)(d_colptrs + nc),
                    d_esize, (int)nc, d_cnt);
        HIPCHK(hipMemcpyAsync(ob->lazy_count, d_cnt, 8, hipMemcpyDeviceToHost, stream));
        ob->count = -1;
        ob->watermark = db->watermark;
        ob->stream_tag = db->stream_tag;
        ob->born_us = db->born_us;
        if (ctx.stats) ctx.stats->num_kernels += 4;
        release_after_use(db);
        record_ready(ob);
        out.emit(ob);
    }
};

// ===== Reduce_GPU: per-batch keyed reduction =====
struct GpuReduceLogic : GpuLogicBase {
    int comb, vcol;
    int64_t max_keys;
    KeyedScratch ks;
    uint64_t* d_okey = nullptr;
    float* d_oval = nullptr;
    int64_t* d_ots = nullptr;
    int64_t* d_on = nullptr;
    GpuReduceLogic(int comb_, int vc, int64_t mk, int dev, Schema os,
                   int64_t cap, bool dense = false) {
        ks.dense = dense;
        comb = comb_; vcol = vc; max_keys = mk;
        device = dev;
        out_schema = os;  // payload [F32]
        out_cap = cap;
    }
    void init_device() override {
        ks.alloc(device, out_cap, max_keys, stream);
        auto& A = arena(device);
        d_okey = (uint64_t*)A.get(8 * out_cap);
        d_oval = (float*)A.get(4 * out_cap);
        d_ots = (int64_t*)A.get(8 * out_cap);
        d_on = (int64_t*)A.get(64);
    }
    void process(Batch* in, EmitCtx& out, RuntimeCtx& ctx) override {
        ensure_init();
        Batch* db = input_on_device(in, ctx);
        int64_t n = db->count;
        ks.group(stream, db, vcol, ctx, /*want_vik=*/false, /*want_carry=*/true);
        Batch* ob = get_dev();
        wfa_segment_reduce_wave(stream, ks.seg_start, ks.seg_slot, ks.d_nseg, n,
                                ks.v_as_f32, ks.v_dt, ks.idx_sorted, db->ts, comb,
                                db->ts_mono ? 1 : 0, ks.slot_to_key, ob->key,
                                ob->cols[0], ob->ts, d_on);
        HIPCHK(hipMemcpyAsync(ob->lazy_count, d_on, 8, hipMemcpyDeviceToHost, stream));
        ob->count = -1;
        ob->watermark = db->watermark;
        ob->born_us = db->born_us;
        if (ctx.stats) ctx.stats->num_kernels += 1;
        release_after_use(db);
        record_ready(ob);
        out.emit(ob);
    }

    void on_eos(EmitCtx&, RuntimeCtx&) override {
        if (inited) ks.check_dense_overflow();
    }
};

// ----- unkeyed full-batch reduce (reference reduce_gpu.hpp:269
// thrust::reduce): one (value, max_ts) tuple per input batch -----
struct GpuReduceAllLogic : GpuLogicBase {
    int comb, vcol;
    float* d_part = nullptr;
    int64_t* d_part_ts = nullptr;
    float* d_cast = nullptr;
    int64_t cast_cap = 0;
    GpuReduceAllLogic(int comb_, int vc, int dev, Schema os, int64_t cap) {
        comb = comb_; vcol = vc;
        device = dev;
        out_schema = os;  // payload [F32]
        out_cap = cap;
    }
    void init_device() override {
        auto& A = arena(device);
        d_part = (float*)A.get(4 * 512);
        d_part_ts = (int64_t*)A.get(8 * 512);
    }
    void process(Batch* in, EmitCtx& out, RuntimeCtx& ctx) override {
        ensure_init();
        Batch* db = input_on_device(in, ctx);
        int64_t n = db->count;
        const void* v = db->cols[vcol];
        int vdt = (int)db->schema.payload[vcol];
        if (vdt != 2 && vdt != 5) {
            if (cast_cap < n) {
                d_cast = (float*)arena(device).get(4 * n);
                cast_cap = n;
            }
            wfa_cast(stream, v, vdt, d_cast, 2, n);
            v = d_cast;
            vdt = 2;
        }
        Batch* ob = get_dev();
        wfa_reduce_all(stream, v, vdt, db->ts, n, comb, d_part, d_part_ts,
                       (float*)ob->cols[0], ob->ts);
        HIPCHK(hipMemsetAsync(ob->key, 0, 8, stream));
        *ob->lazy_count = 1;
        ob->count = 1;
        ob->watermark = db->watermark;
        ob->born_us = db->born_us;
        if (ctx.stats) ctx.stats->num_kernels += 2;
        release_after_use(db);
        record_ready(ob);
        out.emit(ob);
    }
};

// ===== Ffat_Windows_GPU: keyed CB/TB sliding window =====
struct GpuFfatLogic : GpuLogicBase {
    int comb, vcol;
    int64_t win, slide, max_keys;
    bool use_tree;
    bool tb = false;          // time-based (event-time) windows
    int64_t lateness = 0;
    int pend_log2 = 16;       // TB pending-pane ring size
    int64_t pane_len, P, S;
    int ring_log2;
    KeyedScratch ks;
    // state arenas
    int64_t* st_count = nullptr;
    uint32_t* st_fill = nullptr;
    float* st_acc = nullptr;
    float* ring_or_tree = nullptr;
    uint32_t* st_head = nullptr;
    float* st_wsum = nullptr;
    int64_t* st_last = nullptr;  // per-slot last-arrival ts (EOS flush)
    int64_t* d_on = nullptr;
    uint32_t* cb_nf = nullptr;  // CB fire-offset scratch
    // pane-wave fold scratch (pane_len >= 32): completed-pane scan, pane
    // partials, next open-pane accumulators
    uint32_t* pw_base = nullptr;
    float* pw_temp = nullptr;
    float* st_acc_new = nullptr;
    // TB state
    float* tb_pend = nullptr;
    int64_t* tb_base = nullptr;
    int64_t* tb_last = nullptr;
    uint32_t* tb_flags = nullptr;  // [ignored, overflow]
    uint32_t* tb_nf = nullptr;
    // pane-parallel lift scratch (allocated when max_keys <= 4096: span
    // scan + first-pane per segment; null disables the path)
    uint32_t* tb_span_nf = nullptr;
    int64_t* tb_pfirst = nullptr;
    uint32_t* h_flags = nullptr;   // pinned readback
    Engine* eng_ = nullptr;        // central dropped-tuple accounting
    int64_t batches = 0;
    bool use_pane2 = false;
    // A/B on MI355X (8M-tuple batches): the two-stage fold wins when
    // segments are few (1024 keys: 16.9 vs 13.9 B t/s — the wave fold is
    // occupancy-bound at one wave per key) and loses slightly when the
    // chip is already full (8192 keys: 18.6 vs 19.4).  Auto-select by key
    // count; WFA_PANE2=0/1 forces either path.
    bool pane2_enabled() const {
        static int v = -1;
        if (v < 0) {
            const char* e = getenv("WFA_PANE2");
            v = e ? ((e[0] == '1') ? 1 : 0) : 2;  // 2 = auto
        }
        if (v == 2) return max_keys <= 4096;
        return v == 1;
    }

    GpuFfatLogic(int comb_, int vc, int64_t w, int64_t sl, int64_t mk, bool tree,
                 int dev, Schema os, int64_t cap, bool tb_, int64_t lat, int plog2,
                 bool dense = false) {
        ks.dense = dense;
        comb = comb_; vcol = vc; win = w; slide = sl; max_keys = mk; use_tree = tree;
        tb = tb_; lateness = lat;
        if (plog2 > 0) pend_log2 = plog2;
        device = dev;
        out_schema = os;  // payload [F32]
        out_cap = cap;
        pane_len = std::__gcd(win, slide);
        P = win / pane_len;
        S = slide / pane_len;
        ring_log2 = 1;
        while ((1ll << ring_log2) < P + 2) ++ring_log2;
        // TB windows always use the pane ring (the pending-pane machine IS
        // the TB path); a requested tree layout silently falls back — same
        // results, different CB-only perf trade (round 1 threw here)
        if (tb && use_tree) use_tree = false;
    }
    void init_device() override {
        ks.alloc(device, out_cap, max_keys, stream);
        auto& A = arena(device);
        int64_t R = 1ll << ring_log2;
        st_count = (int64_t*)A.get(8 * max_keys);
        st_fill = (uint32_t*)A.get(4 * max_keys);
        st_acc = (float*)A.get(4 * max_keys);
        ring_or_tree = (float*)A.get(4 * max_keys * (use_tree ? 2 * R : R));
        st_head = (uint32_t*)A.get(4 * max_keys);
        st_wsum = (float*)A.get(4 * max_keys);
        st_last = (int64_t*)A.get(8 * max_keys);
        d_on = (int64_t*)A.get(64);
        HIPCHK(hipMemsetAsync(st_count, 0, 8 * max_keys, stream));
        HIPCHK(hipMemsetAsync(st_fill, 0, 4 * max_keys, stream));
        HIPCHK(hipMemsetAsync(st_head, 0, 4 * max_keys, stream));
        HIPCHK(hipMemsetAsync(st_wsum, 0, 4 * max_keys, stream));
        HIPCHK(hipMemsetAsync(st_last, 0, 8 * max_keys, stream));
        float ident = comb == 1 ? INFINITY : (comb == 2 ? -INFINITY : 0.f);
        wfa_fill_f32(stream, st_acc, ident, max_keys);
        wfa_fill_f32(stream, ring_or_tree, ident,
                     max_keys * (use_tree ? 2 * (1ll << ring_log2) : (1ll << ring_log2)));
        // fire-offset scratch doubles as the EOS-flush per-slot counter
        cb_nf = (uint32_t*)A.get(4 * (std::max(out_cap, max_keys) + 1));
        if (!tb && !use_tree && pane_len >= 32 && pane_wave_enabled()) {
            int64_t np = out_cap / pane_len + max_keys + 64;
            pw_base = (uint32_t*)A.get(4 * (np + 1));
            pw_temp = (float*)A.get(4 * (np + 1));
            st_acc_new = (float*)A.get(4 * max_keys);
        }
        use_pane2 = !tb && !use_tree && pane2_enabled() && pane_len >= 32;
        if (tb) {
            int64_t Rp = 1ll << pend_log2;
            tb_pend = (float*)A.get(4 * max_keys * Rp);
            tb_base = (int64_t*)A.get(8 * max_keys);
            tb_last = (int64_t*)A.get(8 * max_keys);
            tb_flags = (uint32_t*)A.get(64);
            tb_nf = (uint32_t*)A.get(4 * (max_keys + 1));
            if (max_keys <= 4096) {  // pane-parallel lift (few segments)
                tb_span_nf = (uint32_t*)A.get(4 * (max_keys + 1));
                tb_pfirst = (int64_t*)A.get(8 * (max_keys + 1));
            }
            wfa_fill_f32(stream, tb_pend, ident, max_keys * Rp);
            wfa_fill_u64(stream, (uint64_t*)tb_base, (uint64_t)-1ll, max_keys);
            wfa_fill_u64(stream, (uint64_t*)tb_last, (uint64_t)-1ll, max_keys);
            HIPCHK(hipMemsetAsync(tb_flags, 0, 64, stream));
            HIPCHK(hipHostMalloc((void**)&h_flags, 64, hipHostMallocDefault));
        }
    }
    ~GpuFfatLogic() override {
        if (h_flags) (void)hipHostFree(h_flags);
        destroy_graphs();
    }

    uint32_t dropped_seen = 0;
    void check_tb_flags() {
        HIPCHK(hipMemcpyAsync(h_flags, tb_flags, 8, hipMemcpyDeviceToHost, stream));
        HIPCHK(hipStreamSynchronize(stream));
        if (h_flags[1])
            throw std::runtime_error(
                "gpu_ffat TB pending-pane ring overflow: raise pend_ring_log2 "
                "or lateness is too large for the configured ring");
        // surface device-side late-tuple drops in the engine's central
        // counter (reference PipeGraph::getNumDroppedTuples)
        if (eng_ && h_flags[0] > dropped_seen) {
            eng_->dropped_tuples.fetch_add(h_flags[0] - dropped_seen,
                                           std::memory_order_relaxed);
            dropped_seen = h_flags[0];
        }
    }

    // hipGraph capture of the steady-state per-batch chain (slot -> sort ->
    // segments -> offsets -> fold -> count copy): the pool rotates a small
    // fixed set of in/out batches, so the (in, out, n) tuple recurs and one
    // instantiated graph per tuple replays the whole chain with a single
    // launch.  A/B on MI355X (4M-tuple batches, value-in-key chain): graphs
    // 267 us/step vs direct launches 248 us/step — the VIK chain's kernels
    // are few and large enough that per-launch gaps are cheaper than the
    // graph launch itself, so graphs are OPT-IN via WFA_HIPGRAPH=1.
    std::map<std::tuple<const void*, const void*, int64_t>, hipGraphExec_t> graphs;
    static bool graphs_enabled() {
        static int v = -1;
        if (v < 0) {
            const char* e = getenv("WFA_HIPGRAPH");
            v = (e && e[0] == '1') ? 1 : (getenv("WFA_NO_HIPGRAPH") ? 0 : 0);
        }
        return v;
    }

    // pane-wave fold (see wfa_ffat_cb_fold_pw) — measured SLOWER than the
    // wave-per-segment fold at the flagship shape (23.7 vs 25.7 B t/s,
    // 16M batch, 8192 keys: the extra pane-scan + temp round-trip +
    // low-occupancy advance outweigh the shorter serial walk), so it is
    // OPT-IN via WFA_PANE_WAVE=1 (kept: correct per full suite + fuzz,
    // and the decomposition is the right shape for much longer panes)
    static bool pane_wave_enabled() {
        static int v = -1;
        if (v < 0) {
            const char* e = getenv("WFA_PANE_WAVE");
            v = (e && e[0] == '1') ? 1 : 0;
        }
        return v;
    }

    void chain_cb(Batch* db, Batch* ob, int64_t n, RuntimeCtx& ctx) {
        ks.group(stream, db, vcol, ctx, /*want_vik=*/!use_tree);
        uint32_t* nf = cb_nf;
        if (pw_base && !use_pane2) {
            wfa_ffat_fire_offsets_pw(stream, ks.seg_start, ks.seg_slot,
                                     ks.d_nseg, n, pane_len, P, S, st_fill,
                                     st_head, nf, d_on, ks.idx_sorted, db->ts,
                                     st_last, pw_base, d_on + 1);
            wfa_ffat_cb_fold_pw(stream, ks.seg_start, ks.seg_slot, ks.d_nseg,
                                d_on + 1, n, ks.v_as_f32, ks.v_dt,
                                ks.idx_sorted, db->ts, pane_len, P, S, comb,
                                ring_log2, st_count, st_fill, st_acc,
                                st_acc_new, ring_or_tree, st_head, st_wsum,
                                ks.slot_to_key, nf, pw_base, pw_temp, ob->key,
                                (float*)ob->cols[0], ob->ts, ob->capacity);
            HIPCHK(hipMemcpyAsync(ob->lazy_count, d_on, 8,
                                  hipMemcpyDeviceToHost, stream));
            return;
        }
        wfa_ffat_fire_offsets(stream, ks.seg_start, ks.seg_slot, ks.d_nseg, n,
                              pane_len, P, S, st_fill, st_head, nf, d_on,
                              ks.idx_sorted, db->ts, st_last);
        if (use_tree)
            wfa_ffat_tree_fold(stream, ks.seg_start, ks.seg_slot, ks.d_nseg, n,
                               ks.v_as_f32, ks.v_dt, ks.idx_sorted, db->ts, pane_len, P, S,
                               comb, ring_log2, st_count, st_fill, st_acc,
                               ring_or_tree, st_head, ks.slot_to_key, nf, ob->key,
                               (float*)ob->cols[0], ob->ts, ob->capacity);
        else if (use_pane2)
            wfa_ffat_cb_fold_fused(stream, ks.seg_start, ks.seg_slot, ks.d_nseg, n,
                                   ks.v_as_f32, ks.v_dt, ks.idx_sorted, db->ts,
                                   pane_len, P, S, comb, ring_log2, st_count,
                                   st_fill, st_acc, ring_or_tree, st_head, st_wsum,
                                   ks.slot_to_key, nf, ob->key,
                                   (float*)ob->cols[0], ob->ts, ob->capacity);
        else
            wfa_ffat_cb_fold(stream, ks.seg_start, ks.seg_slot, ks.d_nseg, n,
                             ks.v_as_f32, ks.v_dt, ks.idx_sorted, db->ts, pane_len, P, S,
                             comb, ring_log2, st_count, st_fill, st_acc,
                             ring_or_tree, st_head, st_wsum, ks.slot_to_key, nf,
                             ob->key, (float*)ob->cols[0], ob->ts, ob->capacity);
        HIPCHK(hipMemcpyAsync(ob->lazy_count, d_on, 8, hipMemcpyDeviceToHost, stream));
    }

    void process(Batch* in, EmitCtx& out, RuntimeCtx& ctx) override {
        ensure_init();
        Batch* db = input_on_device(in, ctx);
        int64_t n = db->count;
        if (tb) {
            ks.group(stream, db, vcol, ctx, /*want_vik=*/true);
            wfa_seg_last_ts(stream, ks.seg_start, ks.seg_slot, ks.d_nseg, n,
                            ks.idx_sorted, db->ts, st_last);
            tb_round_with_count(db, n, db->watermark, out, ctx);
            release_after_use(db);
            return;
        }
        Batch* ob = get_dev();
        auto key = std::make_tuple((const void*)db->ts, (const void*)ob->ts, n);
        auto it = graphs.find(key);
        if (it != graphs.end()) {
            HIPCHK(hipGraphLaunch(it->second, stream));
            if (ctx.stats) ctx.stats->num_kernels += 1;
        } else if (graphs_enabled() && graphs.size() < 64) {
            HIPCHK(hipStreamBeginCapture(stream, hipStreamCaptureModeThreadLocal));
            chain_cb(db, ob, n, ctx);
            hipGraph_t graph;
            HIPCHK(hipStreamEndCapture(stream, &graph));
            hipGraphExec_t exec;
            HIPCHK(hipGraphInstantiate(&exec, graph, nullptr, nullptr, 0));
            HIPCHK(hipGraphDestroy(graph));
            graphs.emplace(key, exec);
            HIPCHK(hipGraphLaunch(exec, stream));
        } else {
            chain_cb(db, ob, n, ctx);
        }
        ob->count = -1;  // resolved by the consumer via gpu_resolve_count
        ob->watermark = db->watermark;
        ob->born_us = db->born_us;
        // async kernels still read db: record its reuse event on our stream
        // (a plain release lets the producer refill it mid-fold)
        release_after_use(db);
        record_ready(ob);
        out.emit(ob);
        if ((++batches & 63) == 0) ks.check_dense_overflow();
    }
    void destroy_graphs() {
        for (auto& [k, e] : graphs) (void)hipGraphExecDestroy(e);
        graphs.clear();
    }

    void tb_round_with_count(Batch* db, int64_t n, int64_t wm, EmitCtx& out,
                             RuntimeCtx& ctx) {
        Batch* ob = get_dev();
        int64_t limit = (wm - lateness) / pane_len - 1;
        wfa_ffat_tb_round(stream, ks.seg_start, ks.seg_slot, ks.d_nseg, n,
                          ks.v_as_f32, ks.v_dt, ks.idx_sorted, db ? db->ts : nullptr,
                          pane_len, P, S, comb, ring_log2, pend_log2, limit,
                          tb_pend, tb_base, tb_last, st_head, st_wsum,
                          ring_or_tree, ks.d_nslots, ks.slot_to_key, tb_nf,
                          tb_flags, tb_flags + 1, ob->key, (float*)ob->cols[0],
                          ob->ts, ob->capacity, d_on,
                          db && db->ts_mono ? 1 : 0, tb_span_nf, tb_pfirst,
                          d_on + 1);
        HIPCHK(hipMemcpyAsync(ob->lazy_count, d_on, 8, hipMemcpyDeviceToHost, stream));
        ob->count = -1;
        ob->watermark = wm - lateness;
        if (ctx.stats) ctx.stats->num_kernels += 4;
        record_ready(ob);
        out.emit(ob);
        if ((++batches & 63) == 0) {
            check_tb_flags();
            ks.check_dense_overflow();
        }
    }

    bool on_punct(int64_t wm, EmitCtx& out, RuntimeCtx& ctx) override {
        if (!tb) return false;
        ensure_init();
        tb_round_with_count(nullptr, 0, wm, out, ctx);
        return false;
    }

    void on_eos(EmitCtx& out, RuntimeCtx& ctx) override {
        ensure_init();
        ks.check_dense_overflow();
        if (tb) {
            // complete every remaining data pane (wm -> +inf) ...
            tb_round_with_count(nullptr, 0, INT64_MAX / 4, out, ctx);
            check_tb_flags();
            // ... then fire the partial pane-windows (< P panes) the CPU
            // twin flushes (windows.cpp FfatCpu on_eos TB branch)
            flush_open(nullptr, nullptr, out, ctx);
            return;
        }
        // CB: flush every open (partial) window — round 1 fired complete
        // windows only (mirrors windows.cpp FfatCpu on_eos CB branch)
        flush_open(st_fill, st_acc, out, ctx);
    }

    void flush_open(const uint32_t* fill, const float* acc, EmitCtx& out,
                    RuntimeCtx& ctx) {
        int64_t R = 1ll << ring_log2;
        wfa_cb_flush_count(stream, ks.d_nslots, P, S, fill, st_head, cb_nf);
        wfa_slot_scan(stream, cb_nf, ks.d_nslots, d_on);
        // EOS only: one sync to learn the total, then page the fires so any
        // window count fits the output batch capacity
        int64_t total = 0;
        HIPCHK(hipStreamSynchronize(stream));
        HIPCHK(hipMemcpy(&total, d_on, 8, hipMemcpyDeviceToHost));
        for (int64_t base = 0; base < total || base == 0; ) {
            Batch* ob = get_dev();
            int64_t page = std::min(total - base, ob->capacity);
            wfa_ffat_cb_flush_fire(stream, ks.d_nslots, P, S, comb, ring_log2,
                                   fill, acc, ring_or_tree,
                                   use_tree ? 2 * R : R, use_tree ? R : 0,
                                   st_head, st_last, ks.slot_to_key, cb_nf,
                                   base, ob->key, (float*)ob->cols[0], ob->ts,
                                   ob->capacity);
            ob->count = page < 0 ? 0 : page;
            ob->watermark = WM_MAX / 4;
            if (ctx.stats) ctx.stats->num_kernels += 1;
            record_ready(ob);
            out.emit(ob);
            base += ob->capacity;
            if (total == 0) break;
        }
        if (ctx.stats) ctx.stats->num_kernels += 2;
    }
};

// ===== MFMA windowed Gram aggregator =====
// Per-key tumbling windows over 16-dim f32 vectors; the GEMM-shaped
// combine sum(v * v^T) runs on the matrix cores (see k_gram in
// sortwin.hip and the guide's MFMA section).  Output: 16 rows per fired
// window — (key, ts=last tuple, c0=gwid, c1..c16 = one Gram row).
struct GpuGramLogic : GpuLogicBase {
    int64_t win, max_keys;
    KeyedScratch ks;
    uint32_t* st_fill = nullptr;
    uint32_t* st_head = nullptr;
    float* st_acc = nullptr;
    uint32_t* nf = nullptr;
    int64_t* d_on = nullptr;
    const float** d_incols = nullptr;
    float** d_outcols = nullptr;
    uint32_t* d_inv = nullptr;
    float* d_staged = nullptr;
    int64_t staged_cap = 0;

    GpuGramLogic(int64_t w, int64_t mk, int dev, Schema os, int64_t cap) {
        win = w;
        max_keys = mk;
        device = dev;
        out_schema = os;  // [I64 gwid, F32 x16]
        out_cap = cap;
        if (os.payload.size() != 17)
            throw std::runtime_error("gpu_gram: out schema must be [I64, F32 x16]");
    }
    void init_device() override {
        ks.alloc(device, out_cap, max_keys, stream);
        auto& A = arena(device);
        st_fill = (uint32_t*)A.get(4 * max_keys);
        st_head = (uint32_t*)A.get(4 * max_keys);
        st_acc = (float*)A.get(4 * max_keys * 256);
        nf = (uint32_t*)A.get(4 * (out_cap + 1));
        d_on = (int64_t*)A.get(64);
        d_incols = (const float**)A.get(16 * 16);
        d_outcols = (float**)A.get(16 * 16);
        HIPCHK(hipMemsetAsync(st_fill, 0, 4 * max_keys, stream));
        HIPCHK(hipMemsetAsync(st_head, 0, 4 * max_keys, stream));
        HIPCHK(hipMemsetAsync(st_acc, 0, 4 * max_keys * 256, stream));
    }
    void process(Batch* in, EmitCtx& out, RuntimeCtx& ctx) override {
        ensure_init();
        Batch* db = input_on_device(in, ctx);
        if (db->schema.payload.size() < 16)
            throw std::runtime_error("gpu_gram: input needs 16 F32 columns");
        int64_t n = db->count;
        ks.group(stream, db, -1, ctx);
        Batch* ob = get_dev();
        const void* inp[16];
        void* outp[16];
        for (int c = 0; c < 16; ++c) {
            inp[c] = db->cols[c];
            outp[c] = ob->cols[1 + c];
        }
        HIPCHK(hipMemcpyAsync(d_incols, inp, 16 * 8, hipMemcpyHostToDevice, stream));
        HIPCHK(hipMemcpyAsync(d_outcols, outp, 16 * 8, hipMemcpyHostToDevice, stream));
        // tumbling windows = panes of `win`, fire every pane (P=1, S=1)
        wfa_ffat_fire_offsets(stream, ks.seg_start, ks.seg_slot, ks.d_nseg, n,
                              win, 1, 1, st_fill, st_head, nf, d_on, nullptr,
                              nullptr, nullptr);
        if (n > staged_cap) {  // sorted row-major staging (see k_gram_stage)
            auto& A = arena(device);
            if (d_staged) {
                A.put(d_staged, 64 * staged_cap);
                A.put(d_inv, 4 * staged_cap);
            }
            staged_cap = n;
            d_staged = (float*)A.get(64 * staged_cap);
            d_inv = (uint32_t*)A.get(4 * staged_cap);
        }
        wfa_gram_prep(stream, d_incols, ks.idx_sorted, n, d_inv, d_staged);
        wfa_gram_fold(stream, ks.seg_start, ks.seg_slot, ks.d_nseg, n, d_incols,
                      ks.idx_sorted, db->ts, win, st_fill, st_acc, st_head,
                      ks.slot_to_key, nf, ob->key, ob->col<int64_t>(0), d_outcols,
                      ob->ts, ob->capacity, d_on, d_inv, d_staged);
        HIPCHK(hipMemcpyAsync(ob->lazy_count, d_on, 8, hipMemcpyDeviceToHost, stream));
        ob->count = -1;
        ob->watermark = db->watermark;
        ob->born_us = db->born_us;
        if (ctx.stats) ctx.stats->num_kernels += 3;
        release_after_use(db);
        record_ready(ob);
        out.emit(ob);
    }
    void on_eos(EmitCtx&, RuntimeCtx&) override {
        if (inited) ks.check_dense_overflow();
    }
};

// ===== GPU -> host stage (D2H) =====
struct GpuToHostLogic : GpuLogicBase {
    explicit GpuToHostLogic(int dev) { device = dev; }
    void process(Batch* in, EmitCtx& out, RuntimeCtx& ctx) override {
        if (in->loc == Loc::HOST) {
            out.emit(in);
            return;
        }
        ensure_init();
        gpu_resolve_count(in);
        wait_ready(in);
        Batch* hb = out.new_batch();
        int64_t n = std::min(in->count, hb->capacity);
        HIPCHK(hipMemcpyAsync(hb->ts, in->ts, 8 * n, hipMemcpyDeviceToHost, stream));
        HIPCHK(hipMemcpyAsync(hb->key, in->key, 8 * n, hipMemcpyDeviceToHost, stream));
        size_t bytes = 16 * n;
        for (size_t c = 0; c < hb->cols.size() && c < in->cols.size(); ++c) {
            size_t es = dsize(in->schema.payload[c]);
            HIPCHK(hipMemcpyAsync(hb->cols[c], in->cols[c], es * n,
                                  hipMemcpyDeviceToHost, stream));
            bytes += es * n;
        }
        HIPCHK(hipStreamSynchronize(stream));
        hb->count = n;
        hb->watermark = in->watermark;
        hb->stream_tag = in->stream_tag;
        if (ctx.stats) ctx.stats->bytes_d2h += bytes;
        release(in);
        out.emit(hb);
    }
};

// ===== GPU sink: consume device batches, count tuples (bench path) =====
struct GpuCountSink : GpuLogicBase {
    Engine* eng;
    int op_id;
    int64_t tuples = 0;
    std::deque<Batch*> pending;
    GpuCountSink(Engine* e, int id, int dev) : eng(e), op_id(id) { device = dev; }
    int64_t wait_us = 0;
    std::vector<int64_t> lat_us;  // per-batch source->sink latency
    void drain_one() {
        Batch* b = pending.front();
        pending.pop_front();
        int64_t t0 = now_us();
        gpu_resolve_count(b);
        wait_us += now_us() - t0;
        if (b->born_us) lat_us.push_back(now_us() - b->born_us);
        tuples += b->count;
        release(b);
    }
    void process(Batch* b, EmitCtx&, RuntimeCtx& ctx) override {
        pending.push_back(b);
        while (pending.size() > 4) drain_one();
    }
    void on_eos(EmitCtx&, RuntimeCtx&) override {
        while (!pending.empty()) drain_one();
        if (stream) (void)hipStreamSynchronize(stream);
        if (wfa_prof())
            fprintf(stderr, "[prof] sink event-wait total: %ld us\n", (long)wait_us);
        eng->sink_tuples[op_id].fetch_add(tuples, std::memory_order_relaxed);
        {
            std::lock_guard<std::mutex> g(eng->sink_f64_mu);
            auto& v = eng->sink_latencies[op_id];
            v.insert(v.end(), lat_us.begin(), lat_us.end());
        }
    }
};

// ===== RCCL keyby exchange: the inter-GPU shuffle =====
// Replaces the reference's KeyBy_Emitter_GPU host-staged re-batching
// (keyby_emitter_gpu.hpp:399-638) with an MI355X-native design: rows are
// bucketed on-device by hash(key) % world, gathered into a contiguous
// per-destination send layout, and exchanged with a grouped RCCL
// send/recv all-to-allv over xGMI.  Watermarks ride in the metadata
// allgather and min-fold across ranks (the collector rule).
// One replica per rank; every rank executes the same collective sequence
// (EOS rounds with zero payload keep lagging ranks in lockstep).
#define NCCLCHK(x)                                                              \
    do {                                                                        \
        ncclResult_t r_ = (x);                                                  \
        if (r_ != ncclSuccess)                                                  \
            throw std::runtime_error(std::string("RCCL error: ") +              \
                                     ncclGetErrorString(r_) + " at " #x);       \
    } while (0)

static ncclComm_t get_rccl_comm(const std::string& id, int rank, int world,
                                int device) {
    static std::mutex mu;
    static std::map<std::string, ncclComm_t> cache;
    std::lock_guard<std::mutex> g(mu);
    auto key = id + "/" + std::to_string(rank);
    auto it = cache.find(key);
    if (it != cache.end()) return it->second;
    if (id.size() != sizeof(ncclUniqueId))
        throw std::runtime_error("rccl_id not set (PipeGraph.set_dist) or wrong size");
    ncclUniqueId uid;
    memcpy(&uid, id.data(), sizeof(uid));
    HIPCHK(hipSetDevice(device));
    ncclComm_t comm;
    NCCLCHK(ncclCommInitRank(&comm, world, uid, rank));
    cache[key] = comm;
    return comm;
}

std::string wfa_rccl_unique_id() {
    ncclUniqueId id;
    NCCLCHK(ncclGetUniqueId(&id));
    return std::string((char*)&id, sizeof(id));
}

struct GpuExchangeLogic : GpuLogicBase {
    Engine* eng;
    int rank = 0, world = 1;
    ncclComm_t comm = nullptr;
    int bits = 1;
    // scratch (dest ping-pongs: batch N's bucket runs on meta_stream while
    // batch N-1's sort still READS its dest array on the main stream)
    uint32_t* dest2[2] = {nullptr, nullptr};
    uint32_t *idx = nullptr, *dest_t = nullptr, *idx_t = nullptr;
    int dest_flip = 0;
    uint32_t* hist = nullptr;
    uint32_t* hist_b = nullptr;  // ping-pongs with dest2 (meta vs main stream)
    uint32_t* d_counts = nullptr;
    int64_t* d_meta = nullptr;      // [world+2] send metadata
    int64_t* d_meta_all = nullptr;  // [world*(world+2)] allgathered
    int64_t* h_meta = nullptr;      // pinned mirror of d_meta_all
    uint32_t* h_counts = nullptr;   // pinned send counts
    void** d_colptrs = nullptr;
    int* d_esize = nullptr;
    std::unique_ptr<Pool> send_pool;
    int64_t cur_wm = 0;
    // metadata rendezvous runs on its own stream, OVERLAPPED with the
    // sort/gather of the same batch (round 1 synced the whole chain per
    // batch: the ~800 us "round" was really the pipeline drain — the
    // k_count_u32 anomaly in profiles/a2a_kernel_stats_r01_rs8.csv was this
    // serialization attributed to the first kernel after the sync)
    hipStream_t meta_stream = nullptr;
    hipEvent_t ev_counts = nullptr;
    // per-ping-pong-buffer liveness: bucket(N+2) on the meta stream must
    // not overwrite dest2[i]/hist[i] before the MAIN-stream sort that
    // reads them has executed (a slow peer's RCCL group can let the host
    // run several batches ahead)
    hipEvent_t ev_dest[2] = {nullptr, nullptr};

    GpuExchangeLogic(Engine* e, int dev, Schema os, int64_t cap) : eng(e) {
        device = dev;
        out_schema = std::move(os);
        out_cap = cap;
        rank = e->dist_rank;
        world = e->dist_world;
        if (world > 8)
            throw std::runtime_error(
                "gpu_keyby_exchange supports up to 8 ranks (one xGMI node)");
        bits = 1;
        while ((1 << bits) < world) ++bits;
        if (bits < 5) bits = 5;  // 8-bit sort machinery (1 pass, fused hist)
    }
    void init_device() override {
        comm = get_rccl_comm(eng->rccl_id, rank, world, device);
        auto& A = arena(device);
        dest2[0] = (uint32_t*)A.get(4 * out_cap);
        dest2[1] = (uint32_t*)A.get(4 * out_cap);
        idx = (uint32_t*)A.get(4 * out_cap);
        dest_t = (uint32_t*)A.get(4 * out_cap);
        idx_t = (uint32_t*)A.get(4 * out_cap);
        hist = (uint32_t*)A.get(4 * wfa_sort_hist_u32(out_cap));
        hist_b = (uint32_t*)A.get(4 * wfa_sort_hist_u32(out_cap));
        d_counts = (uint32_t*)A.get(4 * world + 64);
        d_meta = (int64_t*)A.get(8 * (world + 2));
        d_meta_all = (int64_t*)A.get(8 * world * (world + 2));
        size_t nc = out_schema.payload.size();
        d_colptrs = (void**)A.get(16 * (nc + 1));
        d_esize = (int*)A.get(4 * (nc + 1));
        std::vector<int> es;
        for (auto d : out_schema.payload) es.push_back((int)dsize(d));
        HIPCHK(hipMemcpy(d_esize, es.data(), 4 * nc, hipMemcpyHostToDevice));
        HIPCHK(hipHostMalloc((void**)&h_meta, 8 * world * (world + 2),
                             hipHostMallocDefault));
        HIPCHK(hipHostMalloc((void**)&h_counts, 4 * world, hipHostMallocDefault));
        send_pool = std::make_unique<Pool>(out_schema, out_cap, false);
        send_pool->loc = Loc::DEVICE;
        send_pool->device = device;
        HIPCHK(hipStreamCreateWithFlags(&meta_stream, hipStreamNonBlocking));
        HIPCHK(hipEventCreateWithFlags(&ev_counts, hipEventDisableTiming));
        for (int i = 0; i < 2; ++i) {
            HIPCHK(hipEventCreateWithFlags(&ev_dest[i], hipEventDisableTiming));
            HIPCHK(hipEventRecord(ev_dest[i], stream));  // initially free
        }
    }
    ~GpuExchangeLogic() override {
        if (h_meta) (void)hipHostFree(h_meta);
        if (h_counts) (void)hipHostFree(h_counts);
        if (meta_stream) (void)hipStreamDestroy(meta_stream);
        if (ev_counts) (void)hipEventDestroy(ev_counts);
        for (int i = 0; i < 2; ++i)
            if (ev_dest[i]) (void)hipEventDestroy(ev_dest[i]);
    }

    // kick the metadata rendezvous on meta_stream: pack counts -> allgather
    // -> D2H.  d_counts must be ready on `stream` (ev_counts recorded);
    // meta_stream waits on the event, so sort/gather on `stream` overlap
    // the collective instead of serializing behind it.
    void start_meta(bool have_counts, int64_t wm, bool done) {
        if (!have_counts) wfa_fill_u32(meta_stream, d_counts, 0, world);
        wfa_pack_meta(meta_stream, d_counts, world, done ? WM_MAX : wm,
                      done ? 1 : 0, d_meta);
        if (world > 1)
            NCCLCHK(ncclAllGather(d_meta, d_meta_all, world + 2, ncclInt64,
                                  comm, meta_stream));
        else
            HIPCHK(hipMemcpyAsync(d_meta_all, d_meta, 8 * (world + 2),
                                  hipMemcpyDeviceToDevice, meta_stream));
        HIPCHK(hipMemcpyAsync(h_meta, d_meta_all, 8 * world * (world + 2),
                              hipMemcpyDeviceToHost, meta_stream));
    }

    // analyzed metadata of the current round (valid after sync_meta)
    struct Meta {
        int64_t roff[9] = {0};
        int64_t total = 0;
        int64_t remote_in = 0;   // rows arriving from other ranks
        int64_t remote_out = 0;  // rows leaving to other ranks
        bool all_done = true;
        int64_t min_wm = WM_MAX;
    };

    Meta sync_meta(bool sending) {
        HIPCHK(hipStreamSynchronize(meta_stream));
        Meta m;
        const int64_t* scnt = sending ? h_meta + (int64_t)rank * (world + 2)
                                      : nullptr;
        for (int p = 0; p < world; ++p) {
            const int64_t* mp = h_meta + (int64_t)p * (world + 2);
            m.roff[p] = m.total;
            m.total += mp[rank];
            if (p != rank) m.remote_in += mp[rank];
            if (scnt && p != rank) m.remote_out += scnt[p];
            if (mp[world + 1] == 0) m.all_done = false;
            m.min_wm = std::min(m.min_wm, mp[world]);
        }
        return m;
    }

    // one collective round: metadata already in flight on meta_stream
    // (start_meta); sync META ONLY, lay out recvs, then move rows.
    // Returns true when every rank reported done.
    bool round(Batch* send, EmitCtx& out, RuntimeCtx& ctx) {
        const size_t nc = out_schema.payload.size();
        HIPCHK(hipStreamSynchronize(meta_stream));
        const int64_t* scnt = send ? h_meta + (int64_t)rank * (world + 2) : nullptr;
        // recv layout: rows from rank p land at roff[p]
        int64_t roff[9] = {0};
        int64_t total = 0;
        bool all_done = true;
        int64_t min_wm = WM_MAX;
        for (int p = 0; p < world; ++p) {
            const int64_t* mp = h_meta + (int64_t)p * (world + 2);
            roff[p] = total;
            total += mp[rank];
            if (mp[world + 1] == 0) all_done = false;
            min_wm = std::min(min_wm, mp[world]);
        }
        if (total > out_cap)
            throw std::runtime_error("exchange recv overflow: raise out_batch");
        Batch* rb = nullptr;
        if (total > 0 || send) {
            rb = get_dev();
            // send offsets: prefix of scnt
            int64_t soff[9] = {0};
            for (int p = 1; p < world; ++p)
                soff[p] = soff[p - 1] + (scnt ? scnt[p - 1] : 0);
            // self rows never touch RCCL: direct D2D on our stream (at
            // world=1 the whole exchange is one device copy)
            int64_t selfc = scnt ? scnt[rank] : 0;
            if (selfc) {
                HIPCHK(hipMemcpyAsync(rb->ts + roff[rank], send->ts + soff[rank],
                                      8 * selfc, hipMemcpyDeviceToDevice, stream));
                HIPCHK(hipMemcpyAsync(rb->key + roff[rank], send->key + soff[rank],
                                      8 * selfc, hipMemcpyDeviceToDevice, stream));
                for (size_t c = 0; c < nc; ++c) {
                    size_t es = dsize(out_schema.payload[c]);
                    HIPCHK(hipMemcpyAsync((char*)rb->cols[c] + es * roff[rank],
                                          (char*)send->cols[c] + es * soff[rank],
                                          es * selfc, hipMemcpyDeviceToDevice,
                                          stream));
                }
            }
            bool any_p2p = false;
            for (int p = 0; p < world; ++p) {
                if (p == rank) continue;
                if ((scnt && scnt[p]) || h_meta[(int64_t)p * (world + 2) + rank])
                    any_p2p = true;
            }
            if (any_p2p) {
                NCCLCHK(ncclGroupStart());
                for (int p = 0; p < world; ++p) {
                    if (p == rank) continue;
                    int64_t sc = scnt ? scnt[p] : 0;
                    int64_t rc = h_meta[(int64_t)p * (world + 2) + rank];
                    if (sc) {
                        NCCLCHK(ncclSend(send->ts + soff[p], 8 * sc, ncclChar, p, comm, stream));
                        NCCLCHK(ncclSend(send->key + soff[p], 8 * sc, ncclChar, p, comm, stream));
                        for (size_t c = 0; c < nc; ++c) {
                            size_t es = dsize(out_schema.payload[c]);
                            NCCLCHK(ncclSend((char*)send->cols[c] + es * soff[p], es * sc,
                                             ncclChar, p, comm, stream));
                        }
                    }
                    if (rc) {
                        NCCLCHK(ncclRecv(rb->ts + roff[p], 8 * rc, ncclChar, p, comm, stream));
                        NCCLCHK(ncclRecv(rb->key + roff[p], 8 * rc, ncclChar, p, comm, stream));
                        for (size_t c = 0; c < nc; ++c) {
                            size_t es = dsize(out_schema.payload[c]);
                            NCCLCHK(ncclRecv((char*)rb->cols[c] + es * roff[p], es * rc,
                                             ncclChar, p, comm, stream));
                        }
                    }
                }
                NCCLCHK(ncclGroupEnd());
            }
            // hand off via events (no host sync): the consumer waits on
            // rb's event; sb's recycler waits on the event its releaser
            // records below
            if (send && send->ready_event)
                HIPCHK(hipEventRecord((hipEvent_t)send->ready_event, stream));
        }
        if (ctx.stats) ctx.stats->num_kernels++;
        int64_t out_wm = min_wm == WM_MAX ? cur_wm : min_wm;
        if (rb) {
            if (total > 0) {
                rb->count = total;
                rb->watermark = out_wm;
                record_ready(rb);
                out.emit(rb);
            } else {
                release(rb);
                for (auto* e : out.emitters) e->punct(out_wm);
            }
        } else if (out_wm > cur_wm) {
            for (auto* e : out.emitters) e->punct(out_wm);
        }
        cur_wm = std::max(cur_wm, out_wm);
        return all_done;
    }

    void process(Batch* in, EmitCtx& out, RuntimeCtx& ctx) override {
        int64_t t0 = now_us();
        ensure_init();
        Batch* db = input_on_device(in, ctx);
        int64_t n = db->count;
        int64_t wm = db->watermark;
        if (n > out_cap) throw std::runtime_error("exchange input > out_batch");
        // world=1: the rendezvous is the identity (counts/watermark/done
        // are all local) — forward without the per-batch meta D2H + host
        // sync.  Besides the cycles, this makes the world=1 exchange
        // immune to the pool-side host<->device stalls that intermittently
        // inflated a2a p99 to ~200 ms (BASELINE.md straggler note).
        // WFA_XCHG_NO_SELFPASS keeps the full machinery testable.
        if (world == 1 && !getenv("WFA_XCHG_NO_SELFPASS")) {
            cur_wm = std::max(cur_wm, wm);
            out.emit(db);
            return;
        }
        Batch* sb = send_pool->get();
        int64_t t1 = now_us();
        if (sb->ready_event)
            HIPCHK(hipStreamWaitEvent(stream, (hipEvent_t)sb->ready_event, 0));
        // bucket + count + metadata rendezvous all run on meta_stream,
        // fully decoupled from the main stream's backlog (the previous
        // batch's sort/gather/sends): the per-batch host sync then waits
        // only for THIS batch's input + tiny kernels, not the pipeline
        uint32_t* dest = dest2[dest_flip];
        uint32_t* hi = dest_flip ? hist_b : hist;
        hipEvent_t ev_free = ev_dest[dest_flip];
        dest_flip ^= 1;
        if (db->ready_event)
            HIPCHK(hipStreamWaitEvent(meta_stream, (hipEvent_t)db->ready_event, 0));
        HIPCHK(hipStreamWaitEvent(meta_stream, ev_free, 0));
        wfa_bucket_by_key_h(meta_stream, db->key, n, world, dest, hi);
        wfa_count_u32(meta_stream, dest, n, d_counts, world);
        HIPCHK(hipEventRecord(ev_counts, meta_stream));
        start_meta(true, wm, false);
        Meta m = sync_meta(true);
        // WFA_XCHG_NO_SELFPASS=1 forces the full sort/gather/transfer path
        // (keeps the world=1 device tests exercising the real machinery)
        if (m.remote_in == 0 && m.remote_out == 0 &&
            !getenv("WFA_XCHG_NO_SELFPASS")) {
            // pure-self round: every row of every rank stays home this
            // round — forward the batch untouched (no dest sort, no
            // gather, no self copy).  At world=1 EVERY round is pure-self
            // (the round-1 path burned ~280 us/batch re-materializing an
            // identity permutation); at world>1 this engages whenever the
            // allgathered count matrix is diagonal.
            release(sb);  // staged send batch unused
            int64_t out_wm = m.min_wm == WM_MAX ? cur_wm : m.min_wm;
            cur_wm = std::max(cur_wm, out_wm);
            db->watermark = out_wm;
            if (ctx.stats) ctx.stats->num_kernels += 3;
            out.emit(db);
            if (wfa_prof())
                fprintf(stderr, "[prof] xchg n=%ld self-forward meta=%ld us\n",
                        (long)n, (long)(now_us() - t1));
            return;
        }
        // stable sort by dest -> contiguous per-dest send layout (main
        // stream; waits for this batch's bucket output)
        HIPCHK(hipStreamWaitEvent(stream, ev_counts, 0));
        uint32_t *od, *oi;
        wfa_sort_pairs2_ph(stream, dest, idx, dest_t, idx_t, nullptr, nullptr,
                           hi, n, bits, &od, &oi, nullptr, /*implicit_iota=*/1,
                           /*base_shift=*/0);
        HIPCHK(hipEventRecord(ev_free, stream));  // dest/hist consumed
        size_t nc = db->cols.size();
        std::vector<void*> ptrs(2 * nc);
        for (size_t c = 0; c < nc; ++c) {
            ptrs[c] = db->cols[c];
            ptrs[nc + c] = sb->cols[c];
        }
        HIPCHK(hipMemcpyAsync(d_colptrs, ptrs.data(), 8 * 2 * nc,
                              hipMemcpyHostToDevice, stream));
        wfa_gather_rows(stream, oi, n, db->ts, sb->ts, db->key, sb->key,
                        (const void* const*)d_colptrs, (void* const*)(d_colptrs + nc),
                        d_esize, (int)nc);
        release_after_use(db);
        if (ctx.stats) ctx.stats->num_kernels += 4;
        int64_t t2 = now_us();
        round(sb, out, ctx);
        // round() recorded sb's event after the collective consumed it; the
        // pool's next get() waits on that event before reuse
        release(sb);
        if (wfa_prof())
            fprintf(stderr, "[prof] xchg n=%ld in+get=%ld us bucket=%ld us round=%ld us\n",
                    (long)n, (long)(t1 - t0), (long)(t2 - t1), (long)(now_us() - t2));
    }

    // Puncts are NOT collective-safe (their count differs per rank and
    // would desynchronize the round sequence): swallow them — watermarks
    // cross ranks inside the data/EOS rounds' metadata allgather.
    bool on_punct(int64_t, EmitCtx&, RuntimeCtx&) override { return true; }

    void on_eos(EmitCtx& out, RuntimeCtx& ctx) override {
        ensure_init();
        if (world == 1 && !getenv("WFA_XCHG_NO_SELFPASS")) return;
        // EOS rounds: keep matching other ranks' collectives until all done
        for (;;) {
            start_meta(false, cur_wm, true);
            if (round(nullptr, out, ctx)) break;
        }
    }
};

// ===== debug round trips (isolation tests on a real device) =====
std::pair<std::vector<uint32_t>, std::vector<uint32_t>> debug_sort_pairs_host(
    const uint32_t* keys, int64_t n, int bits) {
    HIPCHK(hipSetDevice(0));
    auto& A = arena(0);
    uint32_t* d_k = (uint32_t*)A.get(4 * n);
    uint32_t* d_v = (uint32_t*)A.get(4 * n);
    uint32_t* d_kt = (uint32_t*)A.get(4 * n);
    uint32_t* d_vt = (uint32_t*)A.get(4 * n);
    uint32_t* d_h = (uint32_t*)A.get(4 * wfa_sort_hist_u32(n));
    HIPCHK(hipMemcpy(d_k, keys, 4 * n, hipMemcpyHostToDevice));
    wfa_iota_u32(nullptr, d_v, n);
    uint32_t *ok, *ov;
    wfa_sort_pairs(nullptr, d_k, d_v, d_kt, d_vt, d_h, n, bits, &ok, &ov);
    std::vector<uint32_t> hk(n), hv(n);
    HIPCHK(hipMemcpy(hk.data(), ok, 4 * n, hipMemcpyDeviceToHost));
    HIPCHK(hipMemcpy(hv.data(), ov, 4 * n, hipMemcpyDeviceToHost));
    A.put(d_k, 4 * n); A.put(d_v, 4 * n); A.put(d_kt, 4 * n); A.put(d_vt, 4 * n);
    A.put(d_h, 4 * wfa_sort_hist_u32(n));
    return {std::move(hk), std::move(hv)};
}

std::vector<uint32_t> debug_key_slots_host(const uint64_t* keys, int64_t n,
                                           int64_t max_keys) {
    HIPCHK(hipSetDevice(0));
    auto& A = arena(0);
    int64_t cap = 1;
    while (cap < 2 * max_keys) cap <<= 1;
    uint64_t* d_key = (uint64_t*)A.get(8 * n);
    uint64_t* d_tab = (uint64_t*)A.get(16 * cap);
    uint32_t* d_ns = (uint32_t*)A.get(64);
    uint32_t* d_out = (uint32_t*)A.get(4 * n);
    uint64_t* d_s2k = (uint64_t*)A.get(8 * max_keys);
    HIPCHK(hipMemcpy(d_key, keys, 8 * n, hipMemcpyHostToDevice));
    wfa_fill_u64(nullptr, d_tab, ~0ULL, 2 * cap);
    wfa_fill_u32(nullptr, d_ns, 0, 1);
    wfa_key_to_slot(nullptr, d_key, n, d_tab, d_ns, cap, d_out, d_s2k);
    std::vector<uint32_t> out(n);
    HIPCHK(hipMemcpy(out.data(), d_out, 4 * n, hipMemcpyDeviceToHost));
    A.put(d_key, 8 * n); A.put(d_tab, 16 * cap);
    A.put(d_ns, 64); A.put(d_out, 4 * n); A.put(d_s2k, 8 * max_keys);
    return out;
}

std::shared_ptr<OpLogic> make_gpu_logic(const std::string& kind, const std::string& spec,
                                        const std::vector<double>& fp,
                                        const std::vector<int64_t>& ip, Engine* eng,
                                        int op_id, int device, const Schema& os,
                                        int64_t out_batch) {
    if (kind == "gpu_source")
        // ip: [len, n_keys, batch, vdt, seed]
        return std::make_shared<GpuSourceLogic>(ip[0], ip[1], ip[2], (int)ip[3],
                                                (uint64_t)ip[4], device, os);
    if (kind == "gpu_map")
        // ip: [spec, col]; fp: [a, b]
        return std::make_shared<GpuMapLogic>((int)ip[0], (int)ip[1], fp[0], fp[1],
                                             device, os);
    if (kind == "gpu_filter")
        return std::make_shared<GpuFilterLogic>((int)ip[0], (int)ip[1], fp[0], fp[1],
                                                device, os, out_batch);
    if (kind == "gpu_jit_map")
        // spec = C expression over (v, ts, key); ip: [col]
        return std::make_shared<GpuJitMapLogic>(spec, (int)(ip.empty() ? 0 : ip[0]),
                                                device, os);
    if (kind == "gpu_jit_filter")
        return std::make_shared<GpuJitFilterLogic>(spec, (int)(ip.empty() ? 0 : ip[0]),
                                                   device, os, out_batch);
    if (kind == "gpu_jit_reduce" || kind == "gpu_jit_ffat" ||
        kind == "gpu_jit_stateful" || kind == "gpu_split")
        // generalized user folds (multi-field accumulators) — gpu_jit.cpp
        return make_gpu_jit_logic(kind, spec, fp, ip, eng, op_id, device, os,
                                  out_batch);
    if (kind == "gpu_map_keyed")
        // ip: [spec, col, max_keys]; fp: [a, b]
        return std::make_shared<GpuStatefulMapLogic>((int)ip[0], (int)ip[1], fp[0],
                                                     fp[1], ip[2], device, os,
                                                     out_batch);
    if (kind == "gpu_filter_keyed")
        return std::make_shared<GpuStatefulFilterLogic>((int)ip[0], (int)ip[1], fp[0],
                                                        fp[1], ip[2], device, os,
                                                        out_batch);
    if (kind == "gpu_reduce_all")
        return std::make_shared<GpuReduceAllLogic>((int)ip[0], (int)ip[1], device,
                                                   os, out_batch);
    if (kind == "gpu_reduce")
        // ip: [comb, vcol, max_keys, dense]
        return std::make_shared<GpuReduceLogic>((int)ip[0], (int)ip[1], ip[2], device,
                                                os, out_batch,
                                                ip.size() > 3 && ip[3] != 0);
    if (kind == "gpu_ffat") {
        // ip: [comb, vcol, win, slide, max_keys, use_tree,
        //      wintype(0 CB/1 TB), lateness, pend_ring_log2]
        auto l = std::make_shared<GpuFfatLogic>(
            (int)ip[0], (int)ip[1], ip[2], ip[3], ip[4], ip[5] != 0, device, os,
            out_batch, ip.size() > 6 && ip[6] != 0, ip.size() > 7 ? ip[7] : 0,
            ip.size() > 8 ? (int)ip[8] : 0, ip.size() > 9 && ip[9] != 0);
        l->eng_ = eng;
        return l;
    }
    if (kind == "gpu_gram") {
        // ip: [win, max_keys, dense]
        auto l = std::make_shared<GpuGramLogic>(ip[0], ip[1], device, os,
                                                out_batch);
        l->ks.dense = ip.size() > 2 && ip[2] != 0;
        return l;
    }
    if (kind == "gpu_exchange")
        return std::make_shared<GpuExchangeLogic>(eng, device, os, out_batch);
    if (kind == "gpu_to_host")
        return std::make_shared<GpuToHostLogic>(device);
    if (kind == "gpu_count_sink") {
        eng->sink_acc_i64[op_id].store(0);
        eng->sink_tuples[op_id].store(0);
        return std::make_shared<GpuCountSink>(eng, op_id, device);
    }
    throw std::runtime_error("unknown gpu logic: " + kind);
}

// Per-stage hipEvent timing of the a2a exchange + reduce chain on
// synthetic data (explain the a2a cost).  Stages:
// gen, map, bucket, count, sort(dest), gather_rows, selfD2D, meta(D2H),
// then the downstream reduce chain: key_to_slot, sort(slot), segments,
// seg_reduce.  All on one stream, world=1 semantics.
std::vector<std::pair<std::string, double>> debug_a2a_stage_times(
    int64_t n, int64_t n_keys, int iters) {
    HIPCHK(hipSetDevice(0));
    hipStream_t s = nullptr;
    auto& A = arena(0);
    int world = 1;
    Schema sch;
    sch.payload.push_back(DType::F32);
    int64_t* d_ts = (int64_t*)A.get(8 * n);
    uint64_t* d_key = (uint64_t*)A.get(8 * n);
    float* d_val = (float*)A.get(4 * n);
    int64_t* o_ts = (int64_t*)A.get(8 * n);
    uint64_t* o_key = (uint64_t*)A.get(8 * n);
    float* o_val = (float*)A.get(4 * n);
    int64_t* r_ts = (int64_t*)A.get(8 * n);
    uint64_t* r_key = (uint64_t*)A.get(8 * n);
    float* r_val = (float*)A.get(4 * n);
    uint32_t* dest = (uint32_t*)A.get(4 * n);
    uint32_t* idx = (uint32_t*)A.get(4 * n);
    uint32_t* dest_t = (uint32_t*)A.get(4 * n);
    uint32_t* idx_t = (uint32_t*)A.get(4 * n);
    uint32_t* hist = (uint32_t*)A.get(4 * wfa_sort_hist_u32(n));
    uint32_t* d_counts = (uint32_t*)A.get(64);
    int64_t* d_meta = (int64_t*)A.get(8 * 3);
    int64_t* h_meta = nullptr;
    HIPCHK(hipHostMalloc((void**)&h_meta, 64, hipHostMallocDefault));
    void** d_colptrs = (void**)A.get(64);
    int* d_esize = (int*)A.get(64);
    {
        int es = 4;
        HIPCHK(hipMemcpy(d_esize, &es, 4, hipMemcpyHostToDevice));
        void* ptrs[2] = {d_val, o_val};
        HIPCHK(hipMemcpy(d_colptrs, ptrs, 16, hipMemcpyHostToDevice));
    }
    KeyedScratch ks;
    ks.alloc(0, n, 4 * n_keys, s);
    uint64_t* rk = (uint64_t*)A.get(8 * n);
    float* rv = (float*)A.get(4 * n);
    int64_t* rt = (int64_t*)A.get(8 * n);
    int64_t* d_on = (int64_t*)A.get(64);
    wfa_gen_batch(s, d_ts, d_key, d_val, 2, n, 0, 42, n_keys);

    constexpr int NS = 11;
    const char* names[NS] = {"gen",        "map",    "bucket", "count",
                             "sort_dest",  "gather", "selfD2D", "meta_d2h",
                             "key_to_slot", "sort_slot+seg", "seg_reduce"};
    hipEvent_t ev[NS + 1];
    for (auto& e : ev) HIPCHK(hipEventCreate(&e));
    double acc[NS] = {0};
    for (int it = -2; it < iters; ++it) {
        HIPCHK(hipEventRecord(ev[0], s));
        wfa_gen_batch(s, d_ts, d_key, d_val, 2, n, (int64_t)it * n, 42, n_keys);
        HIPCHK(hipEventRecord(ev[1], s));
        wfa_map_apply(s, 2, d_val, 2, n, 2.0, 0.5);
        HIPCHK(hipEventRecord(ev[2], s));
        wfa_bucket_by_key(s, d_key, n, world, dest);
        HIPCHK(hipEventRecord(ev[3], s));
        wfa_count_u32(s, dest, n, d_counts, world);
        HIPCHK(hipEventRecord(ev[4], s));
        uint32_t *od, *oi;
        wfa_sort_pairs2(s, dest, idx, dest_t, idx_t, nullptr, nullptr, hist, n,
                        1, &od, &oi, nullptr, 1, 0);
        HIPCHK(hipEventRecord(ev[5], s));
        const void* ci[1] = {d_val};
        void* co[1] = {o_val};
        (void)ci; (void)co;
        wfa_gather_rows(s, oi, n, d_ts, o_ts, d_key, o_key,
                        (const void* const*)d_colptrs,
                        (void* const*)(d_colptrs + 1), d_esize, 1);
        HIPCHK(hipEventRecord(ev[6], s));
        HIPCHK(hipMemcpyAsync(r_ts, o_ts, 8 * n, hipMemcpyDeviceToDevice, s));
        HIPCHK(hipMemcpyAsync(r_key, o_key, 8 * n, hipMemcpyDeviceToDevice, s));
        HIPCHK(hipMemcpyAsync(r_val, o_val, 4 * n, hipMemcpyDeviceToDevice, s));
        HIPCHK(hipEventRecord(ev[7], s));
        wfa_pack_meta(s, d_counts, world, 123, 0, d_meta);
        HIPCHK(hipMemcpyAsync(h_meta, d_meta, 24, hipMemcpyDeviceToHost, s));
        HIPCHK(hipEventRecord(ev[8], s));
        // downstream reduce chain over the received batch
        wfa_key_to_slot(s, r_key, n, ks.tab, ks.d_nslots, ks.table_cap, ks.slot,
                        ks.slot_to_key);
        HIPCHK(hipEventRecord(ev[9], s));
        uint32_t *os2, *oi2, *ov2;
        wfa_cast(s, r_val, 2, ks.v_f32, 2, n);
        wfa_sort_pairs2(s, ks.slot, ks.idx, ks.slot_t, ks.idx_t,
                        (uint32_t*)ks.v_f32, (uint32_t*)ks.v_sorted, ks.hist, n,
                        ks.bits, &os2, &oi2, &ov2, 1, 0);
        ks.segs(s, os2, n, 0);
        HIPCHK(hipEventRecord(ev[10], s));
        wfa_segment_reduce_wave(s, ks.seg_start, ks.seg_slot, ks.d_nseg, n,
                                ov2, 7, oi2, r_ts, 0, /*ts_last=*/1,
                                ks.slot_to_key, rk, rv, rt, d_on);
        HIPCHK(hipEventRecord(ev[11], s));
        HIPCHK(hipStreamSynchronize(s));
        if (it >= 0)
            for (int k = 0; k < NS; ++k) {
                float ms = 0;
                HIPCHK(hipEventElapsedTime(&ms, ev[k], ev[k + 1]));
                acc[k] += ms * 1000.0;
            }
    }
    for (auto& e : ev) (void)hipEventDestroy(e);
    (void)hipHostFree(h_meta);
    std::vector<std::pair<std::string, double>> out;
    for (int k = 0; k < NS; ++k) out.push_back({names[k], acc[k] / iters});
    return out;
}

// Per-stage hipEvent timing of the keyed FFAT CB chain (slot -> sort ->
// segments -> fire offsets -> fold) on synthetic data.  The safe substitute
// for rocprofv3 kernel stats on this pool (profiling runs hang boxes);
// drives perf work from real measurements, not guesses.
std::vector<std::pair<std::string, double>> debug_ffat_stage_times(
    int64_t n, int64_t n_keys, int64_t win, int64_t slide, int iters, int vik) {
    HIPCHK(hipSetDevice(0));
    hipStream_t s = nullptr;
    KeyedScratch ks;
    ks.alloc(0, n, n_keys, s);
    auto& A = arena(0);
    uint64_t* d_key = (uint64_t*)A.get(8 * n);
    int64_t* d_ts = (int64_t*)A.get(8 * n);
    uint16_t* d_val = (uint16_t*)A.get(2 * n);
    {
        std::vector<uint64_t> hk(n);
        std::vector<int64_t> ht(n);
        std::vector<uint16_t> hv(n, 0x3f80);  // bf16 1.0
        uint64_t x = 0x12345;
        for (int64_t i = 0; i < n; ++i) {
            x = x * 6364136223846793005ULL + 1442695040888963407ULL;
            hk[i] = (x >> 33) % (uint64_t)n_keys;
            ht[i] = i;
        }
        HIPCHK(hipMemcpy(d_key, hk.data(), 8 * n, hipMemcpyHostToDevice));
        HIPCHK(hipMemcpy(d_ts, ht.data(), 8 * n, hipMemcpyHostToDevice));
        HIPCHK(hipMemcpy(d_val, hv.data(), 2 * n, hipMemcpyHostToDevice));
    }
    int64_t pane = std::__gcd(win, slide), P = win / pane, S = slide / pane;
    int ring_log2 = 1;
    while ((1ll << ring_log2) < P + 2) ++ring_log2;
    int64_t R = 1ll << ring_log2;
    int64_t fires_cap = n / slide + n_keys + 64;
    int64_t* st_count = (int64_t*)A.get(8 * n_keys);
    uint32_t* st_fill = (uint32_t*)A.get(4 * n_keys);
    float* st_acc = (float*)A.get(4 * n_keys);
    float* ring = (float*)A.get(4 * n_keys * R);
    uint32_t* st_head = (uint32_t*)A.get(4 * n_keys);
    float* st_wsum = (float*)A.get(4 * n_keys);
    uint32_t* nf = (uint32_t*)A.get(4 * (fires_cap + 1));
    uint64_t* o_key = (uint64_t*)A.get(8 * fires_cap);
    float* o_val = (float*)A.get(4 * fires_cap);
    int64_t* o_ts = (int64_t*)A.get(8 * fires_cap);
    int64_t* d_on = (int64_t*)A.get(64);
    HIPCHK(hipMemsetAsync(st_count, 0, 8 * n_keys, s));
    HIPCHK(hipMemsetAsync(st_fill, 0, 4 * n_keys, s));
    HIPCHK(hipMemsetAsync(st_head, 0, 4 * n_keys, s));
    HIPCHK(hipMemsetAsync(st_wsum, 0, 4 * n_keys, s));
    wfa_fill_f32(s, st_acc, 0.f, n_keys);
    wfa_fill_f32(s, ring, 0.f, n_keys * R);

    constexpr int NS = 5;
    const char* names[NS] = {"key_to_slot", "radix_sort", "segments",
                             "fire_offsets", "fold"};
    hipEvent_t ev[NS + 1];
    for (auto& e : ev) HIPCHK(hipEventCreate(&e));
    double acc[NS] = {0};
    for (int it = -2; it < iters; ++it) {  // 2 warmup rounds
        HIPCHK(hipEventRecord(ev[0], s));
        if (vik)
            wfa_key_to_slot_v(s, d_key, n, ks.tab, ks.d_nslots, ks.table_cap,
                              ks.slot, ks.slot_to_key, d_val);
        else
            wfa_key_to_slot(s, d_key, n, ks.tab, ks.d_nslots, ks.table_cap, ks.slot,
                            ks.slot_to_key);
        HIPCHK(hipEventRecord(ev[1], s));
        uint32_t *os_, *oi;
        wfa_sort_pairs2(s, ks.slot, ks.idx, ks.slot_t, ks.idx_t, nullptr, nullptr,
                        ks.hist, n, ks.bits, &os_, &oi, nullptr, 1, vik ? 16 : 0);
        HIPCHK(hipEventRecord(ev[2], s));
        ks.segs(s, os_, n, vik ? 16 : 0);
        HIPCHK(hipEventRecord(ev[3], s));
        wfa_ffat_fire_offsets(s, ks.seg_start, ks.seg_slot, ks.d_nseg, n, pane, P,
                              S, st_fill, st_head, nf, d_on, nullptr, nullptr,
                              nullptr);
        HIPCHK(hipEventRecord(ev[4], s));
        wfa_ffat_cb_fold(s, ks.seg_start, ks.seg_slot, ks.d_nseg, n,
                         vik ? (const void*)os_ : (const void*)d_val, vik ? 6 : 5,
                         oi, d_ts, pane, P, S, 0, ring_log2, st_count, st_fill,
                         st_acc, ring, st_head, st_wsum, ks.slot_to_key, nf, o_key,
                         o_val, o_ts, fires_cap);
        HIPCHK(hipEventRecord(ev[5], s));
        HIPCHK(hipStreamSynchronize(s));
        if (it >= 0)
            for (int k = 0; k < NS; ++k) {
                float ms = 0;
                HIPCHK(hipEventElapsedTime(&ms, ev[k], ev[k + 1]));
                acc[k] += ms * 1000.0;
            }
    }
    for (auto& e : ev) (void)hipEventDestroy(e);
    std::vector<std::pair<std::string, double>> out;
    for (int k = 0; k < NS; ++k) out.push_back({names[k], acc[k] / iters});
    return out;
}

// TB (event-time) window chain harness: group / lift / count+scan /
// advance per-stage hipEvent times.  Window extents scale with key count
// (per-key ts density 1/n_keys) like tools/tb_bench.py.
std::vector<std::pair<std::string, double>> debug_tb_stage_times(
    int64_t n, int64_t n_keys, int iters, int mono) {
    HIPCHK(hipSetDevice(0));
    hipStream_t s = nullptr;
    KeyedScratch ks;
    ks.alloc(0, n, n_keys, s);
    auto& A = arena(0);
    uint64_t* d_key = (uint64_t*)A.get(8 * n);
    int64_t* d_ts = (int64_t*)A.get(8 * n);
    uint16_t* d_val = (uint16_t*)A.get(2 * n);
    wfa_gen_batch(s, d_ts, d_key, d_val, 5, n, 0, 42, n_keys);
    const int64_t pane = 100 * n_keys, P = 10, S = 1;
    int ring_log2 = 5, pend_log2 = 18;
    int64_t R = 1ll << ring_log2;
    int64_t fires_cap = n / pane + n_keys + 64;
    float* pend = (float*)A.get(4 * n_keys * (1ll << pend_log2));
    int64_t* pend_base = (int64_t*)A.get(8 * n_keys);
    int64_t* last_pane = (int64_t*)A.get(8 * n_keys);
    uint32_t* st_head = (uint32_t*)A.get(4 * n_keys);
    float* st_wsum = (float*)A.get(4 * n_keys);
    float* ring = (float*)A.get(4 * n_keys * R);
    uint32_t* nf = (uint32_t*)A.get(4 * (n_keys + 1));
    uint32_t* flags = (uint32_t*)A.get(64);
    uint64_t* o_key = (uint64_t*)A.get(8 * fires_cap);
    float* o_val = (float*)A.get(4 * fires_cap);
    int64_t* o_ts = (int64_t*)A.get(8 * fires_cap);
    int64_t* d_on = (int64_t*)A.get(64);
    wfa_fill_f32(s, pend, 0.f, n_keys * (1ll << pend_log2));
    wfa_fill_f32(s, ring, 0.f, n_keys * R);
    wfa_fill_f32(s, st_wsum, 0.f, n_keys);
    wfa_fill_u64(s, (uint64_t*)pend_base, (uint64_t)-1ll, n_keys);
    wfa_fill_u64(s, (uint64_t*)last_pane, (uint64_t)-1ll, n_keys);
    HIPCHK(hipMemsetAsync(st_head, 0, 4 * n_keys, s));
    HIPCHK(hipMemsetAsync(flags, 0, 64, s));

    constexpr int NS = 4;
    const char* names[NS] = {"group", "lift", "count_scan", "advance"};
    hipEvent_t ev[NS + 1];
    for (auto& e : ev) HIPCHK(hipEventCreate(&e));
    double acc[NS] = {0};
    RuntimeCtx rctx;
    for (int it = -2; it < iters; ++it) {
        wfa_gen_batch(s, d_ts, d_key, d_val, 5, n, (int64_t)(it + 2) * n, 42,
                      n_keys);
        HIPCHK(hipEventRecord(ev[0], s));
        wfa_key_to_slot(s, d_key, n, ks.tab, ks.d_nslots, ks.table_cap,
                        ks.slot, ks.slot_to_key);
        uint32_t *os_, *oi;
        wfa_sort_pairs2(s, ks.slot, ks.idx, ks.slot_t, ks.idx_t, nullptr,
                        nullptr, ks.hist, n, ks.bits, &os_, &oi, nullptr, 1, 0);
        ks.segs(s, os_, n, 0);
        HIPCHK(hipEventRecord(ev[1], s));
        wfa_tb_lift_only(s, ks.seg_start, ks.seg_slot, ks.d_nseg, n, d_val, 5,
                         oi, d_ts, pane, P, S, 0, pend_log2, pend, pend_base,
                         last_pane, flags, flags + 1, mono);
        HIPCHK(hipEventRecord(ev[2], s));
        int64_t limit = ((int64_t)(it + 3) * n) / pane - 1;
        wfa_tb_countscan_only(s, ks.d_nslots, limit, pend_base, last_pane,
                              st_head, P, S, nf, d_on);
        HIPCHK(hipEventRecord(ev[3], s));
        wfa_tb_advance_only(s, ks.d_nslots, limit, pane, P, S, 0, ring_log2,
                            pend_log2, pend, pend_base, last_pane, st_head,
                            st_wsum, ring, ks.slot_to_key, nf, o_key, o_val,
                            o_ts, fires_cap);
        HIPCHK(hipEventRecord(ev[4], s));
        HIPCHK(hipStreamSynchronize(s));
        if (it >= 0)
            for (int k = 0; k < NS; ++k) {
                float ms = 0;
                HIPCHK(hipEventElapsedTime(&ms, ev[k], ev[k + 1]));
                acc[k] += ms * 1000.0;
            }
    }
    for (auto& e : ev) (void)hipEventDestroy(e);
    std::vector<std::pair<std::string, double>> out;
    for (int k = 0; k < NS; ++k) out.push_back({names[k], acc[k] / iters});
    return out;
}

// MFMA Gram-window chain harness: per-stage hipEvent times for the
// matrix-core combine (k_gram).  Also reports derived MFMA throughput:
// each tuple contributes one 16-vector to C += v*v^T (512 FLOP/tuple,
// one v_mfma_f32_16x16x4_f32 per 4 tuples per key segment).
std::vector<std::pair<std::string, double>> debug_gram_stage_times(
    int64_t n, int64_t n_keys, int64_t win, int iters) {
    HIPCHK(hipSetDevice(0));
    hipStream_t s = nullptr;
    KeyedScratch ks;
    ks.alloc(0, n, n_keys, s);
    auto& A = arena(0);
    uint64_t* d_key = (uint64_t*)A.get(8 * n);
    int64_t* d_ts = (int64_t*)A.get(8 * n);
    float* d_cols[16];
    for (int c = 0; c < 16; ++c) {
        d_cols[c] = (float*)A.get(4 * n);
        wfa_gen_batch(s, d_ts, d_key, d_cols[c], 2, n, 0, 1000 + c, n_keys);
    }
    // keys/ts from the last gen call are fine (random keys, ts = i)
    int64_t fires16 = (n / std::max<int64_t>(win, 1) + n_keys + 64) * 16;
    uint32_t* st_fill = (uint32_t*)A.get(4 * n_keys);
    uint32_t* st_head = (uint32_t*)A.get(4 * n_keys);
    float* st_acc = (float*)A.get(4 * n_keys * 256);
    uint32_t* nf = (uint32_t*)A.get(4 * (fires16 + 1));
    uint64_t* o_key = (uint64_t*)A.get(8 * fires16);
    int64_t* o_gwid = (int64_t*)A.get(8 * fires16);
    int64_t* o_ts = (int64_t*)A.get(8 * fires16);
    float* o_cols[16];
    for (int c = 0; c < 16; ++c) o_cols[c] = (float*)A.get(4 * fires16);
    int64_t* d_on = (int64_t*)A.get(64);
    uint32_t* d_inv = (uint32_t*)A.get(4 * n);
    float* d_staged = (float*)A.get(64 * n);
    const float** d_in = (const float**)A.get(16 * 8);
    float** d_out = (float**)A.get(16 * 8);
    HIPCHK(hipMemcpyAsync(d_in, d_cols, 16 * 8, hipMemcpyHostToDevice, s));
    HIPCHK(hipMemcpyAsync(d_out, o_cols, 16 * 8, hipMemcpyHostToDevice, s));
    HIPCHK(hipMemsetAsync(st_fill, 0, 4 * n_keys, s));
    HIPCHK(hipMemsetAsync(st_head, 0, 4 * n_keys, s));
    HIPCHK(hipMemsetAsync(st_acc, 0, 4 * n_keys * 256, s));

    constexpr int NS = 5;
    const char* names[NS] = {"group", "fire_offsets", "stage_rows", "mfma_fold",
                             "gflops"};
    hipEvent_t ev[NS];
    for (auto& e : ev) HIPCHK(hipEventCreate(&e));
    double acc[NS] = {0};
    RuntimeCtx rctx;
    for (int it = -2; it < iters; ++it) {
        HIPCHK(hipEventRecord(ev[0], s));
        wfa_key_to_slot(s, d_key, n, ks.tab, ks.d_nslots, ks.table_cap,
                        ks.slot, ks.slot_to_key);
        uint32_t *os_, *oi;
        wfa_sort_pairs2(s, ks.slot, ks.idx, ks.slot_t, ks.idx_t, nullptr,
                        nullptr, ks.hist, n, ks.bits, &os_, &oi, nullptr, 1, 0);
        ks.segs(s, os_, n, 0);
        HIPCHK(hipEventRecord(ev[1], s));
        wfa_ffat_fire_offsets(s, ks.seg_start, ks.seg_slot, ks.d_nseg, n, win,
                              1, 1, st_fill, st_head, nf, d_on, nullptr,
                              nullptr, nullptr);
        HIPCHK(hipEventRecord(ev[2], s));
        wfa_gram_prep(s, d_in, oi, n, d_inv, d_staged);
        HIPCHK(hipEventRecord(ev[3], s));
        wfa_gram_fold(s, ks.seg_start, ks.seg_slot, ks.d_nseg, n, d_in, oi,
                      d_ts, win, st_fill, st_acc, st_head, ks.slot_to_key, nf,
                      o_key, o_gwid, d_out, o_ts, fires16, d_on, d_inv,
                      d_staged);
        HIPCHK(hipEventRecord(ev[4], s));
        HIPCHK(hipStreamSynchronize(s));
        if (it >= 0)
            for (int k = 0; k < NS - 1; ++k) {
                float ms = 0;
                HIPCHK(hipEventElapsedTime(&ms, ev[k], ev[k + 1]));
                acc[k] += ms * 1000.0;
            }
    }
    for (auto& e : ev) (void)hipEventDestroy(e);
    std::vector<std::pair<std::string, double>> out;
    for (int k = 0; k < NS - 1; ++k) out.push_back({names[k], acc[k] / iters});
    // FLOPs through the matrix cores per batch: 512 per tuple (v*v^T into
    // a 16x16 accumulator), over the fold stage time
    double fold_us = acc[3] / iters;
    out.push_back({"gflops", fold_us > 0 ? (512.0 * n) / (fold_us * 1e3) : 0});
    return out;
}

}  // namespace wfa

#else  // !WFA_WITH_HIP

namespace wfa {
Batch* gpu_alloc_batch(Pool&) { throw std::runtime_error("built without HIP"); }
void gpu_free_batch(Batch*) {}
void gpu_resolve_count(Batch*) {}
Batch* gpu_clone_batch(Batch*) { throw std::runtime_error("built without HIP"); }
std::string wfa_rccl_unique_id() { throw std::runtime_error("built without HIP"); }
std::pair<std::vector<uint32_t>, std::vector<uint32_t>> debug_sort_pairs_host(
    const uint32_t*, int64_t, int) {
    throw std::runtime_error("built without HIP");
}
std::vector<uint32_t> debug_key_slots_host(const uint64_t*, int64_t, int64_t) {
    throw std::runtime_error("built without HIP");
}
std::vector<std::pair<std::string, double>> debug_ffat_stage_times(int64_t, int64_t,
                                                                   int64_t, int64_t,
                                                                   int, int) {
    throw std::runtime_error("built without HIP");
}
std::vector<std::pair<std::string, double>> debug_a2a_stage_times(int64_t, int64_t,
                                                                  int) {
    throw std::runtime_error("built without HIP");
}
std::vector<std::pair<std::string, double>> debug_gram_stage_times(int64_t,
                                                                   int64_t,
                                                                   int64_t,
                                                                   int) {
    throw std::runtime_error("built without HIP");
}
std::vector<std::pair<std::string, double>> debug_tb_stage_times(int64_t,
                                                                 int64_t, int,
                                                                 int) {
    throw std::runtime_error("built without HIP");
}
std::shared_ptr<OpLogic> make_gpu_logic(const std::string&, const std::string&,
                                        const std::vector<double>&,
                                        const std::vector<int64_t>&, Engine*, int, int,
                                        const Schema&, int64_t) {
    throw std::runtime_error("built without HIP");
}
}  // namespace wfa

#endif
