// windflow_amd native engine — operators, emitters, collectors, replicas.
//
// Replaces WindFlow's FastFlow-based runtime:
//  - Basic_Emitter family   (wf/basic_emitter.hpp, forward/keyby/broadcast)
//  - collectors             (wf/watermark_collector.hpp, ordering, kslack)
//  - Basic_Operator/Replica (wf/basic_operator.hpp)
//  - PipeGraph::run threads (wf/pipegraph.hpp, multipipe.hpp)
// The graph itself is flattened by the Python layer (windflow_amd/pipegraph.py)
// into OpSpec/EdgeSpec lists; this engine spawns one pinned thread per
// replica connected by SPSC queues and runs the watermark/EOS protocol.
#pragma once

#include <chrono>
#include <deque>
#include <functional>
#include <thread>
#include <unordered_map>

#include "core.hpp"
#include "queue.hpp"

namespace wfa {

struct Engine;
struct Replica;

inline int64_t now_us() {
    return std::chrono::duration_cast<std::chrono::microseconds>(
               std::chrono::steady_clock::now().time_since_epoch())
        .count();
}

// ----- per-replica stats (reference: wf/stats_record.hpp:70-82) -----
struct StatsRecord {
    int64_t inputs_received = 0;      // batches
    int64_t tuples_received = 0;
    int64_t outputs_sent = 0;         // batches
    int64_t tuples_sent = 0;
    int64_t bytes_received = 0;
    int64_t bytes_sent = 0;
    int64_t inputs_ignored = 0;
    double service_time_us_ewma = 0;  // per-batch service time
    int64_t num_kernels = 0;          // GPU kernel launches
    int64_t bytes_h2d = 0;
    int64_t bytes_d2h = 0;
    int64_t start_us = 0;
    int64_t end_us = 0;
};

// ----- runtime context (reference: wf/context.hpp:53) -----
struct RuntimeCtx {
    int replica = 0;
    int parallelism = 1;
    int64_t current_ts = 0;
    int64_t current_wm = 0;   // collector-folded watermark of the current batch
    int current_tag = -1;     // join stream tag of the current batch
    Engine* engine = nullptr;
    int op_id = 0;
    StatsRecord* stats = nullptr;
};

// ===== emitters =====
// An emitter owns the producer->consumer queues of one output edge of one
// replica.  It takes ownership of one reference of every batch passed in.
struct Emitter {
    virtual ~Emitter() = default;
    virtual void emit(Batch* b) = 0;
    virtual void punct(int64_t wm) = 0;  // watermark-only keep-alive
    virtual void flush() = 0;
    virtual void eos() = 0;
    std::atomic<bool>* abort = nullptr;
    StatsRecord* stats = nullptr;
    void account(Batch* b) {
        if (!stats) return;
        stats->outputs_sent++;
        if (b->count > 0) stats->tuples_sent += b->count;
    }
};

// FORWARD / REBALANCING: whole batches, round-robin across destinations.
struct ForwardEmitter : Emitter {
    std::vector<SpscQueue*> dests;
    size_t rr = 0;
    Pool* punct_pool;  // tiny batches for punctuation
    explicit ForwardEmitter(std::vector<SpscQueue*> d, Pool* pp) : dests(std::move(d)), punct_pool(pp) {}
    void emit(Batch* b) override {
        account(b);
        dests[rr]->push(b, abort);
        rr = (rr + 1) % dests.size();
    }
    void punct(int64_t wm) override {
        for (auto* q : dests) {
            Batch* p = punct_pool->get();
            p->punct = true;
            p->watermark = wm;
            q->push(p, abort);
        }
    }
    void flush() override {}
    void eos() override {
        for (auto* q : dests) q->push(EOS_TAG, abort);
    }
};

// BROADCAST: same batch pointer to every destination, refcounted
// (reference: wf/broadcast_emitter.hpp delete_counter scheme).
struct BroadcastEmitter : Emitter {
    std::vector<SpscQueue*> dests;
    Pool* punct_pool;
    explicit BroadcastEmitter(std::vector<SpscQueue*> d, Pool* pp) : dests(std::move(d)), punct_pool(pp) {}
    void emit(Batch* b) override {
        account(b);
        if (b->loc == Loc::DEVICE && dests.size() > 1) {
            // per-batch events are single-consumer: broadcast device
            // batches as D2D clones (reference splitting_emitter_gpu
            // replicates per branch; round 1 rejected this case)
            for (size_t i = 1; i < dests.size(); ++i)
                dests[i]->push(gpu_clone_batch(b), abort);
            dests[0]->push(b, abort);
            return;
        }
        b->refcnt.fetch_add((int)dests.size() - 1, std::memory_order_relaxed);
        for (auto* q : dests) q->push(b, abort);
    }
    void punct(int64_t wm) override {
        for (auto* q : dests) {
            Batch* p = punct_pool->get();
            p->punct = true;
            p->watermark = wm;
            q->push(p, abort);
        }
    }
    void flush() override {}
    void eos() override {
        for (auto* q : dests) q->push(EOS_TAG, abort);
    }
};

// KEYBY: split every input batch by hash(key) % n into per-destination open
// batches; flush them when full or when the watermark cadence expires; idle
// destinations get keep-alive punctuations so downstream watermarks keep
// flowing (reference: wf/keyby_emitter.hpp:305-377 idle-destination logic).
struct KeyByEmitter : Emitter {
    std::vector<SpscQueue*> dests;
    Pool* out_pool;
    Pool* punct_pool;
    int64_t out_batch;
    std::vector<Batch*> open;
    std::vector<int64_t> open_wm;       // min-fold of contributing watermarks
    std::vector<bool> sent_recently;
    int64_t last_cadence_us;
    int64_t cadence_us = 100000;        // WF_DEFAULT_WM_INTERVAL_USEC
    int64_t cur_wm = 0;
    std::vector<uint32_t> cnt;          // per-destination count scratch

    KeyByEmitter(std::vector<SpscQueue*> d, Pool* op, Pool* pp, int64_t ob)
        : dests(std::move(d)), out_pool(op), punct_pool(pp), out_batch(ob) {
        open.assign(dests.size(), nullptr);
        open_wm.assign(dests.size(), WM_MAX);
        sent_recently.assign(dests.size(), false);
        last_cadence_us = now_us();
        cnt.resize(dests.size());
    }

    static inline uint64_t mix(uint64_t k) {
        // splitmix64 finalizer — stable hash used on CPU and GPU alike
        k += 0x9e3779b97f4a7c15ULL;
        k = (k ^ (k >> 30)) * 0xbf58476d1ce4e5b9ULL;
        k = (k ^ (k >> 27)) * 0x94d049bb133111ebULL;
        return k ^ (k >> 31);
    }

    void emit(Batch* b) override;
    void flush_dest(size_t d);
    void punct(int64_t wm) override;
    void flush() override {
        for (size_t d = 0; d < dests.size(); ++d) flush_dest(d);
    }
    void eos() override {
        flush();
        for (auto* q : dests) q->push(EOS_TAG, abort);
    }
};

// ===== collectors =====
enum class CollectorKind : int { WATERMARK = 0, ORDERING = 1, KSLACK = 2, JOIN = 3 };

// Fan-in node in front of every replica after a shuffle.
// WATERMARK: rewrites each batch watermark to min over open channels
// (reference: wf/watermark_collector.hpp:65-137).
struct Collector {
    std::vector<SpscQueue*> chans;
    std::vector<int64_t> chan_wm;
    std::vector<bool> open;
    std::vector<int> chan_tag;   // join stream tag per channel (-1 none)
    size_t n_open = 0;
    size_t rr = 0;
    int64_t last_fwd_wm = -1;
    // the folded watermark/tag of the batch most recently returned by
    // next(): shared (broadcast) batches are NOT rewritten in place — a
    // concurrent consumer would race on the same fields (found by the
    // TSAN harness, csrc/tests/engine_stress.cpp)
    int64_t delivered_wm = 0;
    int delivered_tag = -1;
    std::atomic<bool>* abort = nullptr;

    explicit Collector(std::vector<SpscQueue*> ch, std::vector<int> tags = {})
        : chans(std::move(ch)), chan_tag(std::move(tags)) {
        chan_wm.assign(chans.size(), 0);
        open.assign(chans.size(), true);
        n_open = chans.size();
        if (chan_tag.empty()) chan_tag.assign(chans.size(), -1);
    }
    virtual ~Collector() = default;

    int64_t min_wm() const {
        int64_t m = WM_MAX;
        for (size_t c = 0; c < chans.size(); ++c)
            if (open[c] && chan_wm[c] < m) m = chan_wm[c];
        return m == WM_MAX ? chan_max_wm() : m;
    }
    int64_t chan_max_wm() const {
        int64_t m = 0;
        for (auto w : chan_wm) m = std::max(m, w);
        return m;
    }

    // next data/punct batch with rewritten watermark; nullptr == all-EOS.
    virtual Batch* next();
};

// ORDERING (DETERMINISTIC mode): releases rows in global timestamp order
// across channels — a TUPLE-granular k-way merge (reference
// wf/ordering_collector.hpp:51 orders per-tuple messages).  Batches whose
// rows all precede every other channel's head pass through untouched (the
// common ts-contiguous case, zero copy); otherwise the head batch is SPLIT
// at the merge boundary and the safe prefix is carved into a fresh batch.
// A channel must be non-empty (or closed) before anything is released.
struct OrderingCollector : Collector {
    std::vector<std::deque<Batch*>> pend;
    std::vector<int64_t> off;   // consumed-row offset into each head batch
    // wm of the last *released* rows per channel: a released batch must not
    // carry watermark knowledge from rows still buffered in pend (that
    // would fire downstream windows before their content arrives)
    std::vector<int64_t> rel_wm;
    Pool* out_pool = nullptr;   // sub-batch carving; null => batch granularity
    using Collector::Collector;
    int64_t released_wm() const;
    Batch* carve(Batch* h, int64_t o, int64_t m);
    Batch* next() override;
};

// KSLACK (PROBABILISTIC mode): buffer TUPLES, release them in ts order up
// to the slack horizon t_curr - K, adapting K to observed disorder; late
// tuples (behind the last released ts) are dropped and counted per tuple
// (reference: wf/kslack_collector.hpp:52 buffers/drops per tuple).
struct KSlackCollector : Collector {
    int64_t K = 0;
    int64_t t_curr = 0;
    int64_t last_rel_ts = INT64_MIN;
    // min-heap of buffered rows (ts, seq for stable order, batch, row idx)
    struct Row {
        int64_t ts;
        uint64_t seq;
        Batch* b;
        int64_t i;
        bool operator>(const Row& o) const {
            return ts != o.ts ? ts > o.ts : seq > o.seq;
        }
    };
    std::vector<Row> heap;                        // std::push/pop_heap, min
    std::unordered_map<Batch*, int64_t> remaining;  // unreleased rows per batch
    uint64_t seq = 0;
    Pool* out_pool = nullptr;   // rebuilt output batches; null => batch mode
    Batch* open_out = nullptr;
    std::vector<std::pair<int64_t, Batch*>> buf;  // batch-granularity fallback
    std::atomic<int64_t>* dropped = nullptr;
    using Collector::Collector;
    Batch* next() override;
    Batch* next_batchwise();    // fallback for device batches / no pool
};

// ===== operator logic =====
struct EmitCtx;  // defined below

struct OpLogic {
    virtual ~OpLogic() = default;
    // called once in the replica thread before any processing — GPU logics
    // create streams/pools here so a gated start measures steady state only
    virtual void warm(RuntimeCtx& ctx) {}
    // Take ownership of `in` (release or re-emit it).
    virtual void process(Batch* in, EmitCtx& out, RuntimeCtx& ctx);
    // Watermark-only punctuation: stateful logics (windows, joins) override
    // to advance firing; return true to suppress default re-propagation.
    virtual bool on_punct(int64_t wm, EmitCtx& out, RuntimeCtx& ctx) { return false; }
    virtual void on_eos(EmitCtx& out, RuntimeCtx& ctx) {}
    virtual bool is_source() const { return false; }
    // CPU logics dereference batch columns on the host; device batches
    // must cross gpu_to_host first.  GPU logics override to true and the
    // replica loop turns violations into a clear error, not a segfault.
    virtual bool accepts_device() const { return false; }
    // Source: fill-and-emit loop; return false when exhausted.
    virtual bool source_step(EmitCtx& out, RuntimeCtx& ctx) { return false; }
    // Called by the replica thread AFTER on_eos/flush, while the engine's
    // pools are still alive — the place to return held batches (logic
    // destructors run after Engine::pools is destroyed).
    virtual void post_eos() {}
};

// Emission context handed to operator logic: wraps the replica's emitters.
struct EmitCtx {
    std::vector<Emitter*> emitters;   // usually 1; >1 after a split
    Pool* out_pool = nullptr;         // pool for new output batches
    void emit(Batch* b) {
        if (emitters.empty()) { release(b); return; }
        for (size_t i = 1; i < emitters.size(); ++i) {
            b->refcnt.fetch_add(1, std::memory_order_relaxed);
            emitters[i]->emit(b);
        }
        emitters[0]->emit(b);
    }
    void emit_to(size_t branch, Batch* b) { emitters[branch]->emit(b); }
    size_t n_branches() const { return emitters.size(); }
    Batch* new_batch() { return out_pool->get(); }
};

// ===== graph description (built by the Python layer) =====
// An op may hold several chained stages fused into one replica thread —
// this implements MultiPipe::chain (reference multipipe.hpp:536-590) and
// key-extractor fusion without extra queue hops.
struct StageSpec {
    std::function<std::shared_ptr<OpLogic>()> factory;
    Schema out_schema;
    int64_t out_batch = 1024;
};

struct OpSpec {
    int id = 0;
    std::string name;
    int parallelism = 1;
    std::vector<StageSpec> stages;
    int device = -1;                              // -1 = CPU
    bool pinned_out = false;
    const StageSpec& last() const { return stages.back(); }
};

// Runs stage i's output directly through stage i+1 in the same thread.
struct ChainLogic : OpLogic {
    struct StageEmitter;
    std::vector<std::shared_ptr<OpLogic>> stages;
    std::vector<EmitCtx> ctxs;                  // ctxs[i] = output ctx of stage i
    std::vector<std::unique_ptr<Emitter>> glue; // stage->next adapters
    void wire(const std::vector<Pool*>& pools, EmitCtx& final_ctx, RuntimeCtx& rctx);
    bool is_source() const override { return stages.front()->is_source(); }
    bool accepts_device() const override { return stages.front()->accepts_device(); }
    bool source_step(EmitCtx&, RuntimeCtx& ctx) override {
        return stages.front()->source_step(ctxs[0], ctx);
    }
    void process(Batch* in, EmitCtx&, RuntimeCtx& ctx) override {
        stages.front()->process(in, ctxs[0], ctx);
    }
    bool on_punct(int64_t wm, EmitCtx&, RuntimeCtx& ctx) override;
    void on_eos(EmitCtx&, RuntimeCtx& ctx) override;
    void warm(RuntimeCtx& ctx) override;
    void post_eos() override {
        for (auto& st : stages) st->post_eos();
    }
};

struct EdgeSpec {
    int from = 0, to = 0;
    Routing routing = Routing::FORWARD;
    CollectorKind collector = CollectorKind::WATERMARK;
    int stream_tag = -1;   // tag applied to batches on this edge (joins)
};

// ===== replica =====
struct Replica {
    Engine* engine = nullptr;
    int op_id = 0;
    int idx = 0;
    std::shared_ptr<OpLogic> logic;
    std::unique_ptr<Collector> collector;          // null for sources
    std::vector<std::unique_ptr<Emitter>> emitters;
    EmitCtx ectx;
    RuntimeCtx rctx;
    StatsRecord stats;
    std::thread th;
    void run();
};

// ===== engine =====
struct Engine {
    ExecMode mode = ExecMode::DEFAULT;
    TimePolicy time_policy = TimePolicy::EVENT_TIME;
    std::vector<OpSpec> ops;
    std::vector<EdgeSpec> edges;

    std::vector<std::unique_ptr<Replica>> replicas;
    std::vector<std::unique_ptr<Pool>> pools;
    std::vector<std::unique_ptr<SpscQueue>> queues;
    std::atomic<bool> abort{false};
    std::atomic<int64_t> dropped_tuples{0};
    std::mutex error_mu;
    std::string first_error;  // first replica failure (rethrown by wait())
    // gated start: threads spawn + logics warm, then block until open_gate()
    std::atomic<int> gate{0};
    std::atomic<int> warmed{0};
    bool use_gate = false;
    // per-op sink accumulators (differential-test invariant support)
    std::unordered_map<int, std::atomic<int64_t>> sink_acc_i64;
    std::unordered_map<int, double> sink_acc_f64;
    std::mutex sink_f64_mu;
    std::unordered_map<int, std::atomic<int64_t>> sink_tuples;
    std::unordered_map<int, std::vector<int64_t>> sink_latencies;  // us, GPU sinks
    int64_t queue_capacity = 128;   // batches per SPSC queue
    bool pin_threads = false;
    // distributed (one process per GPU, RCCL over xGMI): set before build()
    int dist_rank = 0;
    int dist_world = 1;
    std::string rccl_id;            // ncclUniqueId bytes (broadcast by rank 0)

    Pool* make_pool(const Schema& s, int64_t cap, bool pinned) {
        pools.push_back(std::make_unique<Pool>(s, cap, pinned));
        return pools.back().get();
    }

    void build();       // instantiate replicas/queues/emitters/collectors
    void start();       // spawn threads
    void wait();        // join all
    void run() { build(); start(); wait(); }
    void start_gated();  // spawn + warm, block replicas at the gate
    void open_gate() { gate.store(1, std::memory_order_release); }
};

// native logic factory (native_logic.cpp)
std::shared_ptr<OpLogic> make_native_logic(const std::string& kind,
                                           const std::string& spec,
                                           const std::vector<double>& fparams,
                                           const std::vector<int64_t>& iparams,
                                           Engine* eng, int op_id);

// ----- CPU window / join suite (windows.cpp) -----
// Contiguous view over one window's rows, handed to user (Python) window
// functions for the non-incremental path (reference wf/iterable.hpp).
struct WinRows {
    int64_t n = 0;
    const int64_t* ts = nullptr;
    uint64_t key = 0;
    int64_t gwid = 0;
    const Schema* schema = nullptr;
    std::vector<const char*> cols;  // base pointer per payload column
};
using WindowFn = std::function<double(const WinRows&)>;

// Vectorized user join function (reference interval_join.hpp:279-307
// arbitrary predicate/result): called once per batch of candidate pairs
// with SoA views; fills keep[] (0/1) and out[] (one result per kept pair).
struct JoinPairs {
    int64_t n = 0;
    const uint64_t* key = nullptr;
    const int64_t* ts_a = nullptr;
    const int64_t* ts_b = nullptr;
    const double* va = nullptr;
    const double* vb = nullptr;
};
using JoinFn =
    std::function<void(const JoinPairs&, uint8_t* keep, double* out)>;

// kinds: win_keyed, win_parallel, win_plq, win_wlq, win_mr_map,
// win_mr_reduce, win_ffat, interval_join  (iparam layouts in windows.cpp)
std::shared_ptr<OpLogic> make_window_logic(const std::string& kind,
                                           const std::vector<double>& fp,
                                           const std::vector<int64_t>& ip,
                                           Engine* eng, int op_id,
                                           WindowFn userfn = nullptr,
                                           JoinFn joinfn = nullptr);

// GPU logic factory (gpu_ops.cpp) — kinds: gpu_source/gpu_map/gpu_filter/
// gpu_reduce/gpu_ffat/gpu_to_host/gpu_count_sink
std::shared_ptr<OpLogic> make_gpu_logic(const std::string& kind, const std::string& spec,
                                        const std::vector<double>& fp,
                                        const std::vector<int64_t>& ip, Engine* eng,
                                        int op_id, int device, const Schema& os,
                                        int64_t out_batch);

// persistent state tier (persist.cpp) — kinds: p_reduce
std::shared_ptr<OpLogic> make_persist_logic(const std::string& kind,
                                            const std::string& spec,
                                            const std::vector<double>& fp,
                                            const std::vector<int64_t>& ip,
                                            Engine* eng, int op_id);
std::shared_ptr<void> open_state_store(const std::string& path, int64_t cache_cap,
                                       void** kv_out, void** cache_out,
                                       bool fresh = true);
std::string* state_cache_get(void* cache, uint64_t key);
void state_cache_put(void* cache, uint64_t key, const std::string& v);
void state_cache_flush(void* cache);
int64_t state_kv_size(void* kv);
bool state_kv_erase(void* kv, void* cache, uint64_t key);

// debug hooks (gpu_ops.cpp): device sort / hash-slot round trips
std::pair<std::vector<uint32_t>, std::vector<uint32_t>> debug_sort_pairs_host(
    const uint32_t* keys, int64_t n, int bits);
std::vector<uint32_t> debug_key_slots_host(const uint64_t* keys, int64_t n,
                                           int64_t max_keys);
std::vector<std::pair<std::string, double>> debug_ffat_stage_times(
    int64_t n, int64_t n_keys, int64_t win, int64_t slide, int iters, int vik);
std::vector<std::pair<std::string, double>> debug_gram_stage_times(
    int64_t n, int64_t n_keys, int64_t win, int iters);
std::vector<std::pair<std::string, double>> debug_tb_stage_times(
    int64_t n, int64_t n_keys, int iters, int mono);
std::vector<std::pair<std::string, double>> debug_a2a_stage_times(
    int64_t n, int64_t n_keys, int iters);

// RCCL bootstrap: rank 0 generates the id, broadcasts it out-of-band
// (torch.distributed store), every rank passes it to Engine::rccl_id.
std::string wfa_rccl_unique_id();

// JIT fold codegen (gpu_jit.cpp): returns the hiprtc source that would be
// compiled for the given jit logic — lets the CPU test tier cross-compile
// generated kernels with hipcc and catch codegen errors without a GPU.
std::string debug_jit_fold_source(const std::string& kind,
                                  const std::string& spec,
                                  const std::vector<double>& fp,
                                  const std::vector<int64_t>& ip);

}  // namespace wfa
