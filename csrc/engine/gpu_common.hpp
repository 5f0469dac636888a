// Shared host-side GPU infrastructure used by gpu_ops.cpp and gpu_jit.cpp:
// the per-device arena allocator, the GPU operator-logic base class, the
// keyed grouping scratch (hash->slot + radix sort + segments), and the
// hiprtc JIT compile cache.  Only meaningful under WFA_WITH_HIP.
#pragma once
#ifdef WFA_WITH_HIP

#include <hip/hip_runtime.h>
#include <hip/hiprtc.h>

#include <map>
#include <mutex>
#include <set>
#include <string>
#include <vector>

#include "engine.hpp"

namespace wfa {

#define HIPCHK(x)                                                                  \
    do {                                                                           \
        hipError_t err_ = (x);                                                     \
        if (err_ != hipSuccess)                                                    \
            throw std::runtime_error(std::string("HIP error: ") +                  \
                                     hipGetErrorString(err_) + " at " #x);         \
    } while (0)

#define RTCCHK(x)                                                              \
    do {                                                                       \
        hiprtcResult r_ = (x);                                                 \
        if (r_ != HIPRTC_SUCCESS)                                              \
            throw std::runtime_error(std::string("hiprtc error: ") +           \
                                     hiprtcGetErrorString(r_) + " at " #x);    \
    } while (0)

// ===== per-device arena allocator =====
struct DeviceArena {
    std::mutex mu;
    std::map<size_t, std::vector<void*>> free_by_size;
    int device;
    size_t allocated = 0;

    void* get(size_t bytes) {
        bytes = (bytes + 255) & ~size_t(255);
        {
            std::lock_guard<std::mutex> g(mu);
            auto it = free_by_size.find(bytes);
            if (it != free_by_size.end() && !it->second.empty()) {
                void* p = it->second.back();
                it->second.pop_back();
                return p;
            }
        }
        void* p = nullptr;
        HIPCHK(hipSetDevice(device));
        HIPCHK(hipMalloc(&p, bytes));
        if (getenv("WFA_DEBUG_SYNC"))
            fprintf(stderr, "[alloc] %p..%p (%zu)\n", p, (char*)p + bytes,
                    bytes);
        std::lock_guard<std::mutex> g(mu);
        allocated += bytes;
        return p;
    }
    void put(void* p, size_t bytes) {
        bytes = (bytes + 255) & ~size_t(255);
        std::lock_guard<std::mutex> g(mu);
        free_by_size[bytes].push_back(p);
    }
};

inline DeviceArena g_arena[64];

inline DeviceArena& arena(int dev) {
    g_arena[dev].device = dev;
    return g_arena[dev];
}

inline bool wfa_prof() {
    static int v = -1;
    if (v < 0) v = getenv("WFA_PROF") ? 1 : 0;
    return v;
}

// WFA_DEBUG_SYNC=1: synchronize + error-check after every kernel group and
// print a marker — turns an async "Memory access fault" abort into a
// pinpointed stage (the poor man's compute-sanitizer for this pool)
inline bool wfa_debug_sync() {
    static int v = -1;
    if (v < 0) v = getenv("WFA_DEBUG_SYNC") ? 1 : 0;
    return v;
}

inline void dbg_sync(hipStream_t s, const char* where) {
    if (!wfa_debug_sync()) return;
    hipError_t e = hipStreamSynchronize(s);
    hipError_t e2 = hipGetLastError();
    fprintf(stderr, "[dbgsync] %s sync=%s last=%s\n", where,
            hipGetErrorString(e), hipGetErrorString(e2));
    fflush(stderr);
}

// ===== base for GPU logics =====
struct GpuLogicBase : OpLogic {
    int64_t prof_host_us = 0, prof_calls = 0;
    int device = 0;
    hipStream_t stream = nullptr;
    std::unique_ptr<Pool> dev_pool;  // device batches this logic emits
    Schema out_schema;
    int64_t out_cap = 1 << 20;
    bool inited = false;

    virtual void init_device() {}
    void ensure_init() {
        if (inited) return;
        HIPCHK(hipSetDevice(device));
        HIPCHK(hipStreamCreateWithFlags(&stream, hipStreamNonBlocking));
        dev_pool = std::make_unique<Pool>(out_schema, out_cap, false);
        dev_pool->loc = Loc::DEVICE;
        dev_pool->device = device;
        init_device();
        inited = true;
    }
    // held host batches go back at post_eos() — Engine::pools is destroyed
    // BEFORE replica logics, so the destructor must not release into pools
    void post_eos() override { drain_h2d(true); }
    ~GpuLogicBase() override {
        h2d_inflight.clear();  // post_eos drained; abort paths just leak
        if (stream) (void)hipStreamDestroy(stream);
    }

    void warm(RuntimeCtx&) override { ensure_init(); }
    bool accepts_device() const override { return true; }

    // make the producing stream's work visible to our stream
    void wait_ready(Batch* b) {
        if (b->loc == Loc::DEVICE && b->ready_event)
            HIPCHK(hipStreamWaitEvent(stream, (hipEvent_t)b->ready_event, 0));
    }
    void record_ready(Batch* b) {
        b->stream = stream;
        HIPCHK(hipEventRecord((hipEvent_t)b->ready_event, stream));
    }

    // Releasing a device batch whose contents our ASYNC kernels still read:
    // re-record its event on our stream ("uses done") so whoever pulls it
    // from the pool next waits before overwriting.  (Single-consumer only;
    // device-batch broadcast clones instead — see BroadcastEmitter.)
    void release_after_use(Batch* b) {
        if (b->loc == Loc::DEVICE && b->ready_event &&
            b->refcnt.load(std::memory_order_acquire) == 1)
            HIPCHK(hipEventRecord((hipEvent_t)b->ready_event, stream));
        release(b);
    }

    // pool get, preferring a batch whose reuse event already completed —
    // otherwise allocate a fresh one up to a bounded depth so producers
    // never serialize against consumers still reading a recycled batch
    // (288 GB HBM makes a deep rotation cheap; reference recycling_gpu
    // instead spin-waits under memory pressure).
    Batch* get_from(Pool& pool) {
        constexpr int MAX_DEPTH = 8;
        Batch* chosen = nullptr;
        Batch* skipped[MAX_DEPTH];
        int nskip = 0;
        while (nskip < MAX_DEPTH) {
            Batch* b = pool.try_pop();
            if (!b) break;
            if (!b->ready_event ||
                hipEventQuery((hipEvent_t)b->ready_event) == hipSuccess) {
                chosen = b;
                break;
            }
            skipped[nskip++] = b;
        }
        for (int i = 0; i < nskip; ++i) pool.put(skipped[i]);
        if (!chosen) {
            if (nskip > 0 && pool.live.load(std::memory_order_relaxed) >= MAX_DEPTH)
                chosen = pool.get();  // bounded: reuse, stream-waits below
            else
                chosen = pool.make_new();
        }
        if (chosen->ready_event)
            HIPCHK(hipStreamWaitEvent(stream, (hipEvent_t)chosen->ready_event, 0));
        return chosen;
    }

    Batch* get_dev() { return get_from(*dev_pool); }

    // Input-side device pool: H2D/peer clones must carry the SOURCE batch's
    // schema, not this logic's output schema — a logic whose in-dtype
    // differs from its out-dtype (e.g. f64 column in, f32 results out)
    // would otherwise mislabel (and under-size) the staged columns.
    std::unique_ptr<Pool> in_pool;
    std::vector<std::unique_ptr<Pool>> retired_in_pools;

    Batch* get_in(const Batch* src) {
        if (!in_pool || in_pool->schema.payload != src->schema.payload ||
            in_pool->capacity < src->count) {
            if (in_pool) retired_in_pools.push_back(std::move(in_pool));
            in_pool = std::make_unique<Pool>(
                src->schema, std::max({out_cap, src->capacity, src->count}),
                false);
            in_pool->loc = Loc::DEVICE;
            in_pool->device = device;
        }
        return get_from(*in_pool);
    }

    // Staging in-flight window (H2D and peer D2D): the source batch may
    // be recycled by its producer the moment we release it, so it is held
    // until its copies' event fires — released OPPORTUNISTICALLY on later
    // calls instead of a per-batch stream sync (round 1 synced every
    // batch; reference forward_emitter_gpu.hpp has the same 1-batch
    // overlap on its pinned staging buffers).
    std::deque<std::pair<Batch*, hipEvent_t>> h2d_inflight;

    void drain_h2d(bool block) {
        while (!h2d_inflight.empty()) {
            auto [old_hb, ev] = h2d_inflight.front();
            if (!block && hipEventQuery(ev) != hipSuccess) break;
            if (block) HIPCHK(hipEventSynchronize(ev));
            release(old_hb);
            h2d_inflight.pop_front();
        }
    }

    Batch* to_device(Batch* hb, RuntimeCtx& ctx) {
        drain_h2d(false);
        Batch* db = get_in(hb);
        int64_t n = hb->count;
        HIPCHK(hipMemcpyAsync(db->ts, hb->ts, 8 * n, hipMemcpyHostToDevice, stream));
        HIPCHK(hipMemcpyAsync(db->key, hb->key, 8 * n, hipMemcpyHostToDevice, stream));
        size_t bytes = 16 * n;
        for (size_t c = 0; c < hb->cols.size() && c < db->cols.size(); ++c) {
            size_t es = dsize(hb->schema.payload[c]);
            HIPCHK(hipMemcpyAsync(db->cols[c], hb->cols[c], es * n,
                                  hipMemcpyHostToDevice, stream));
            bytes += es * n;
        }
        db->count = n;
        db->watermark = ctx.current_wm;   // folded (hb may be a shared batch)
        db->stream_tag = ctx.current_tag;
        db->ts_mono = hb->ts_mono;
        if (ctx.stats) ctx.stats->bytes_h2d += bytes;
        HIPCHK(hipEventRecord((hipEvent_t)db->ready_event, stream));
        // db's event may be re-recorded later (recycling protocol) — that
        // only delays the host batch's release, never un-orders it
        h2d_inflight.push_back({hb, (hipEvent_t)db->ready_event});
        if (h2d_inflight.size() > 4) drain_h2d(true);  // bounded holding
        return db;
    }

    Batch* input_on_device(Batch* b, RuntimeCtx& ctx) {
        if (b->loc == Loc::HOST) return to_device(b, ctx);
        gpu_resolve_count(b);
        if (b->device != device || force_peer_copy())
            return peer_copy(b, ctx);
        wait_ready(b);
        return b;
    }

    // WFA_FORCE_PEER_COPY=1 exercises the cross-device path on a 1-GPU box
    // (device 0 -> 0 through hipMemcpyPeerAsync)
    static bool force_peer_copy() {
        static int v = -1;
        if (v < 0) {
            const char* e = getenv("WFA_FORCE_PEER_COPY");
            v = (e && e[0] == '1') ? 1 : 0;
        }
        return v;
    }

    // Cross-GPU forward: producer replica on another device handed us its
    // batch pointer (withDevice(0) -> withDevice(1) chains move
    // over xGMI).  hipMemcpyPeerAsync uses the direct link when peer access
    // is up (enabled once per pair below) and stages through the host
    // otherwise — the reference has no P2P path at all
    // (wf/forward_emitter_gpu.hpp:296-328 always stages via pinned host).
    Batch* peer_copy(Batch* b, RuntimeCtx& ctx) {
        drain_h2d(false);
        enable_peer(device, b->device);
        // producer-side contents must be valid before the engine reads them
        // from the consumer device: wait on the producing event (legal
        // cross-device), then copy on OUR stream
        wait_ready(b);
        Batch* db = get_in(b);
        const int64_t n = b->count;
        HIPCHK(hipMemcpyPeerAsync(db->ts, device, b->ts, b->device, 8 * n,
                                  stream));
        HIPCHK(hipMemcpyPeerAsync(db->key, device, b->key, b->device, 8 * n,
                                  stream));
        size_t bytes = 16 * n;
        for (size_t c = 0; c < b->cols.size() && c < db->cols.size(); ++c) {
            size_t es = dsize(b->schema.payload[c]);
            HIPCHK(hipMemcpyPeerAsync(db->cols[c], device, b->cols[c],
                                      b->device, es * n, stream));
            bytes += es * n;
        }
        db->count = n;
        db->watermark = b->watermark;
        db->stream_tag = b->stream_tag;
        db->born_us = b->born_us;
        if (ctx.stats) ctx.stats->bytes_d2h += bytes;  // inter-device traffic
        // the source batch may be recycled by its (other-device) producer
        // the moment we release: hold it on the copies' event instead of a
        // per-batch stream sync (same 1-batch-overlap protocol as H2D)
        HIPCHK(hipEventRecord((hipEvent_t)db->ready_event, stream));
        h2d_inflight.push_back({b, (hipEvent_t)db->ready_event});
        if (h2d_inflight.size() > 4) drain_h2d(true);
        return db;
    }

    static void enable_peer(int dst, int src) {
        if (dst == src) return;
        static std::mutex mu;
        static std::set<std::pair<int, int>> done;
        std::lock_guard<std::mutex> g(mu);
        if (!done.insert({dst, src}).second) return;
        int can = 0;
        if (hipDeviceCanAccessPeer(&can, dst, src) == hipSuccess && can) {
            HIPCHK(hipSetDevice(dst));
            hipError_t e = hipDeviceEnablePeerAccess(src, 0);
            if (e != hipSuccess && e != hipErrorPeerAccessAlreadyEnabled)
                (void)hipGetLastError();  // fall back to staged copies
        }
    }
};

// ===== shared keyed front half: slot -> sort -> gather -> segments =====
struct KeyedScratch {
    uint64_t* tab = nullptr;        // packed (key, slot) 16 B entries
    uint32_t* d_nslots = nullptr;
    uint64_t* slot_to_key = nullptr;
    uint32_t *slot = nullptr, *idx = nullptr, *slot_t = nullptr, *idx_t = nullptr;
    uint32_t* hist = nullptr;
    uint32_t *seg_start = nullptr, *seg_slot = nullptr;
    int64_t* d_nseg = nullptr;
    float* v_sorted = nullptr;   // second-payload ping-pong (sort_pairs2)
    float* v_f32 = nullptr;      // cast buffer / sort ping-pong
    int64_t table_cap = 0;
    int64_t max_keys = 0;
    int64_t cap = 0;
    int bits = 20;
    // user-asserted dense integer keys (< max_keys): slot = key, the hash
    // probe is skipped entirely (withDenseKeys)
    bool dense = false;
    uint32_t* d_overflow = nullptr;
    // bounded-slot segments table (allocated when max_keys <= 64 K):
    // single-pass boundary scatter replaces the two-pass count+scan+scatter
    uint32_t* by_slot = nullptr;

    void alloc(int dev, int64_t cap_, int64_t mk, hipStream_t s);
    void check_dense_overflow();
    void segs(hipStream_t s, const uint32_t* slot_sorted, int64_t n, int shr);

    uint32_t* idx_sorted = nullptr;  // valid after group()
    const void* v_as_f32 = nullptr;  // f32 or bf16, per v_dt
    int v_dt = 2;                    // effective dtype of v_as_f32 (2/5)

    void group(hipStream_t s, Batch* db, int vcol, RuntimeCtx& ctx,
               bool want_vik = false, bool want_carry = false);
};

// ===== hiprtc JIT compile cache =====
// Returns a loaded module for the given source (compiled for the local
// arch); modules are cached per source string for the process lifetime.
hipModule_t jit_module(const std::string& src, int device);
hipFunction_t jit_fn(hipModule_t mod, const char* name);
std::string jit_type(DType d);

// generalized user-fold JIT factory (gpu_jit.cpp): kinds gpu_jit_reduce /
// gpu_jit_ffat
std::shared_ptr<OpLogic> make_gpu_jit_logic(const std::string& kind,
                                            const std::string& spec,
                                            const std::vector<double>& fp,
                                            const std::vector<int64_t>& ip,
                                            Engine* eng, int op_id, int device,
                                            const Schema& os, int64_t out_batch);

}  // namespace wfa

#endif  // WFA_WITH_HIP
