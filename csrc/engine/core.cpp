#include "core.hpp"

#include <cstdio>

#ifdef WFA_WITH_HIP
#include <hip/hip_runtime.h>
#endif

namespace wfa {

void* host_alloc(size_t bytes, bool pinned) {
#ifdef WFA_WITH_HIP
    if (pinned) {
        void* p = nullptr;
        if (hipHostMalloc(&p, bytes, hipHostMallocDefault) == hipSuccess) return p;
        // fall through to plain malloc when no GPU is present
    }
#endif
    void* p = nullptr;
    if (posix_memalign(&p, 64, bytes ? bytes : 64) != 0) throw std::bad_alloc();
    return p;
}

void host_free(void* p, bool pinned) {
    if (!p) return;
#ifdef WFA_WITH_HIP
    if (pinned) {
        hipPointerAttribute_t attr;
        if (hipPointerGetAttributes(&attr, p) == hipSuccess &&
            attr.type == hipMemoryTypeHost) {
            (void)hipHostFree(p);
            return;
        }
    }
#endif
    free(p);
}

static Batch* alloc_batch(const Schema& s, int64_t cap, bool pinned) {
    (void)s;
    Batch* b = new Batch();
    b->capacity = cap;
    b->schema = s;
    b->pinned = pinned;
    b->ts = reinterpret_cast<int64_t*>(host_alloc(cap * 8, pinned));
    b->key = reinterpret_cast<uint64_t*>(host_alloc(cap * 8, pinned));
    b->cols.resize(s.payload.size());
    for (size_t i = 0; i < s.payload.size(); ++i)
        b->cols[i] = host_alloc(cap * dsize(s.payload[i]), pinned);
    return b;
}

static void free_batch(Batch* b) {
    if (b->loc == Loc::HOST) {
        host_free(b->ts, b->pinned);
        host_free(b->key, b->pinned);
        for (size_t i = 0; i < b->cols.size(); ++i) host_free(b->cols[i], b->pinned);
        delete b;
    } else {
        gpu_free_batch(b);
    }
}

Pool::~Pool() {
    for (Batch* b : free_list) free_batch(b);
}

Batch* Pool::try_pop() {
    std::lock_guard<std::mutex> g(mu);
    if (free_list.empty()) return nullptr;
    Batch* b = free_list.back();
    free_list.pop_back();
    b->count = 0;
    b->watermark = 0;
    b->punct = false;
    b->stream_tag = -1;
    b->ts_mono = false;
    b->refcnt.store(1, std::memory_order_relaxed);
    return b;
}

Batch* Pool::make_new() {
    Batch* b = (loc == Loc::DEVICE) ? gpu_alloc_batch(*this)
                                    : alloc_batch(schema, capacity, pinned);
    b->pool = this;
    b->refcnt.store(1, std::memory_order_relaxed);
    live.fetch_add(1, std::memory_order_relaxed);
    return b;
}

Batch* Pool::get() {
    if (Batch* b = try_pop()) return b;
    return make_new();
}

void Pool::put(Batch* b) {
    std::lock_guard<std::mutex> g(mu);
    free_list.push_back(b);
}

void release(Batch* b) {
    if (!b) return;
    if (b->refcnt.fetch_sub(1, std::memory_order_acq_rel) == 1) {
        if (b->pool)
            b->pool->put(b);
        else
            free_batch(b);
    }
}

Batch* clone(Batch* b, Pool& pool) {
    Batch* c = pool.get();
    c->count = b->count;
    c->watermark = b->watermark;
    c->punct = b->punct;
    c->stream_tag = b->stream_tag;
    c->ts_mono = b->ts_mono;
    memcpy(c->ts, b->ts, b->count * 8);
    memcpy(c->key, b->key, b->count * 8);
    for (size_t i = 0; i < b->cols.size(); ++i)
        memcpy(c->cols[i], b->cols[i], b->count * dsize(b->schema.payload[i]));
    return c;
}

}  // namespace wfa
