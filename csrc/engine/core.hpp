// windflow_amd native engine — core types.
//
// MI355X-native re-design of WindFlow's message layer (reference:
// wf/basic.hpp, wf/single_t.hpp, wf/batch_t.hpp, wf/batch_cpu_t.hpp,
// wf/batch_gpu_t.hpp, wf/recycling.hpp, wf/recycling_gpu.hpp).
//
// Design departure from the reference: there is no Single_t / per-tuple
// message.  The native message unit is ALWAYS a micro-batch in
// structure-of-arrays layout (one contiguous array per column), because
// (a) SoA is what coalesced HIP kernels want, (b) per-batch Python/numpy
// callbacks amortize interpreter cost, and (c) CPU SIMD loops vectorize.
// A "single" is a batch of count==1.
#pragma once

#include <atomic>
#include <cstdint>
#include <cstdlib>
#include <cstring>
#include <memory>
#include <mutex>
#include <stdexcept>
#include <string>
#include <vector>

namespace wfa {

// ----- enums (reference: wf/basic.hpp:78-97) -----
enum class ExecMode : int { DEFAULT = 0, DETERMINISTIC = 1, PROBABILISTIC = 2 };
enum class TimePolicy : int { INGRESS_TIME = 0, EVENT_TIME = 1 };
enum class WinType : int { CB = 0, TB = 1 };
enum class JoinMode : int { KP = 0, DP = 1 };
enum class Routing : int { FORWARD = 0, KEYBY = 1, BROADCAST = 2, REBALANCING = 3 };

enum class DType : int { I64 = 0, F64 = 1, F32 = 2, U64 = 3, I32 = 4, U16 = 5, U8 = 6 };

inline size_t dsize(DType d) {
    switch (d) {
        case DType::I64: case DType::F64: case DType::U64: return 8;
        case DType::F32: case DType::I32: return 4;
        case DType::U16: return 2;
        case DType::U8: return 1;
    }
    return 8;
}

// Timestamps/watermarks are int64 (microseconds or logical ids).
// WM_NONE marks "no watermark yet".
constexpr int64_t WM_MAX = INT64_MAX;

enum class Loc : int { HOST = 0, DEVICE = 1 };

// ----- schema -----
// Every batch carries: ts column (i64), key column (u64), payload columns.
struct Schema {
    std::vector<DType> payload;
    bool operator==(const Schema& o) const { return payload == o.payload; }
};

struct Pool;  // fwd

// ----- Batch -----
// One micro-batch.  Columns are raw arenas of `capacity` elements; `count`
// of them are valid.  Host batches allocate with aligned malloc (or pinned
// memory when `pinned` is set, for fast H2D).  Device batches (Loc::DEVICE)
// are allocated from the per-GPU arena allocator (see gpu_alloc.hpp) and
// carry the owning device + the hipStream the producing kernels ran on.
struct Batch {
    int64_t count = 0;
    int64_t capacity = 0;
    int64_t watermark = 0;      // min-folded watermark of the tuples within
    bool punct = false;         // watermark-only punctuation (count==0)
    int stream_tag = -1;        // join stream id (0=A,1=B), -1 otherwise
    Loc loc = Loc::HOST;
    int device = -1;
    void* stream = nullptr;     // hipStream_t of producer (device batches)
    void* ready_event = nullptr;// hipEvent_t signalled when contents valid

    int64_t born_us = 0;        // host clock when the source filled it
    // ts column is nondecreasing in row order (set by monotonic sources,
    // preserved by order-preserving ops): lets keyed reductions take the
    // per-segment LAST row as the ts max instead of a per-row gather
    bool ts_mono = false;
    int64_t* ts = nullptr;      // [capacity]
    uint64_t* key = nullptr;    // [capacity]
    // Deferred count: device producers may emit before the row count is
    // known on host (count == -1).  The count lands in *lazy_count (pinned)
    // once ready_event fires; resolve with gpu_resolve_count().
    int64_t* lazy_count = nullptr;
    std::vector<void*> cols;    // payload columns per schema
    Schema schema;
    bool pinned = false;

    std::atomic<int> refcnt{1}; // broadcast multicast refcount
    Pool* pool = nullptr;       // recycling home

    template <typename T> T* col(size_t i) {
        // bounds-checked: an out-of-range column index (user config error)
        // must be a clear exception, not a segfault.  Fetched once per
        // BATCH in every hot loop, so the branch costs nothing.
        if (i >= cols.size())
            throw std::runtime_error("column c" + std::to_string(i) +
                                     " out of range (batch has " +
                                     std::to_string(cols.size()) +
                                     " payload columns)");
        return reinterpret_cast<T*>(cols[i]);
    }
    size_t n_payload() const { return schema.payload.size(); }
};

void* host_alloc(size_t bytes, bool pinned);
void host_free(void* p, bool pinned);

// ----- recycling pool (reference: wf/recycling.hpp) -----
// A mutex freelist is sufficient: allocation is per-batch (thousands/s),
// not per-tuple.  One pool per (schema, capacity, pinned, loc/device).
struct Pool {
    Schema schema;
    int64_t capacity;
    bool pinned;
    Loc loc = Loc::HOST;
    int device = -1;
    std::mutex mu;
    std::vector<Batch*> free_list;
    std::atomic<int64_t> live{0};

    Pool(Schema s, int64_t cap, bool pin) : schema(std::move(s)), capacity(cap), pinned(pin) {}
    ~Pool();

    Batch* get();               // pop or allocate
    Batch* try_pop();           // pop only (null when empty)
    Batch* make_new();          // always allocate
    void put(Batch* b);         // return to freelist
};

// drop one reference; recycle into pool when it hits zero
void release(Batch* b);

// device-batch hooks (gpu_ops.cpp): one contiguous HBM slab per batch from
// the per-GPU arena allocator — the recycling_gpu redesign for 288 GB HBM.
Batch* gpu_alloc_batch(Pool& pool);
void gpu_free_batch(Batch* b);
void gpu_resolve_count(Batch* b);  // blocks on ready_event if count == -1

// Deep-copy b (same pool class) — used for copy-on-write under broadcast.
Batch* clone(Batch* b, Pool& pool);

// Device-batch clone for broadcast fan-out (reference
// splitting_emitter_gpu.hpp:186-200 D2D per-branch replication): D2D copy
// on the producing stream into a fresh batch from b's own pool.
Batch* gpu_clone_batch(Batch* b);

}  // namespace wfa
