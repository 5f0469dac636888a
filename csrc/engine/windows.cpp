// CPU window / join operator suite.
//
// MI355X-native re-design of WindFlow's CPU windowing layer (reference:
// wf/window_replica.hpp, wf/window_structure.hpp, wf/keyed_windows.hpp,
// wf/parallel_windows.hpp, wf/paned_windows.hpp, wf/mapreduce_windows.hpp,
// wf/ffat_windows.hpp + wf/ffat_replica.hpp + wf/flatfat.hpp,
// wf/interval_join.hpp).  Differences from the reference by design:
//  - batch-SoA processing (no per-tuple Single_t messages);
//  - paned/mapreduce stages exchange explicit (gwid, partial) rows rather
//    than relying on channel ordering, so DEFAULT mode stays correct;
//  - TB windows for a key open at the first window CONTAINING the key's
//    first tuple (the reference opens from lwid 0, which floods empty
//    windows for late-appearing keys with large event timestamps).
//
// Engine op kinds and iparam layouts (see make_window_logic at bottom):
//  win_keyed     [wintype, win, slide, lateness, comb, col, own_mode]
//  win_parallel  same (own_mode=1: replica r owns windows gwid%par==r)
//  win_plq       same, win==slide==pane; emits (ts=start, key, c0=pane_gwid,
//                c1=partial) — the PLQ stage of Paned_Windows
//  win_wlq       [wintype, win_p, slide_p, lateness, comb, pane_len, own_mode]
//                consumes PLQ rows, fires windows of win_p panes
//  win_mr_map    [wintype, win, slide, lateness, comb, col] — MAP stage of
//                MapReduce_Windows: accumulates round-robin subset, emits one
//                (gwid, partial) per window per replica
//  win_mr_reduce [n_partials, comb] — merges n partials per (key, gwid)
//  win_ffat      [wintype, win, slide, lateness, comb, col] — per-key FlatFAT
//  interval_join [mode(0 KP,1 DP), lower, upper, colA, colB]
#include <algorithm>
#include <cmath>
#include <deque>
#include <map>
#include <numeric>
#include <queue>
#include <unordered_map>

#include "engine.hpp"

namespace wfa {
namespace {

enum Comb { C_SUM = 0, C_MIN = 1, C_MAX = 2, C_COUNT = 3, C_AVG = 4 };

// value union: integer path preserves exactness for I64 payloads
struct ValU {
    int64_t i = 0;
    double f = 0.0;
};

struct Acc {
    int64_t i = 0;
    double f = 0.0;
    int64_t cnt = 0;
};

struct AggCfg {
    int comb = C_SUM;
    bool use_int = true;  // int accumulate path (I64/I32 col, comb != AVG)

    void add(Acc& a, const ValU& v) const {
        if (a.cnt == 0 && (comb == C_MIN || comb == C_MAX)) {
            a.i = v.i;
            a.f = v.f;
        } else {
            switch (comb) {
                case C_SUM: a.i += v.i; a.f += v.f; break;
                case C_MIN: a.i = std::min(a.i, v.i); a.f = std::min(a.f, v.f); break;
                case C_MAX: a.i = std::max(a.i, v.i); a.f = std::max(a.f, v.f); break;
                default: break;  // COUNT/AVG use cnt / f-sum
            }
            if (comb == C_AVG) a.f += v.f;
        }
        a.cnt++;
    }
    // merge partial b into a (paned/mapreduce recombination)
    void merge(Acc& a, const Acc& b) const {
        if (b.cnt == 0) return;
        if (a.cnt == 0) {
            a = b;
            return;
        }
        switch (comb) {
            case C_SUM: case C_AVG: a.i += b.i; a.f += b.f; break;
            case C_MIN: a.i = std::min(a.i, b.i); a.f = std::min(a.f, b.f); break;
            case C_MAX: a.i = std::max(a.i, b.i); a.f = std::max(a.f, b.f); break;
            case C_COUNT: break;  // cnt carries it
        }
        a.cnt += b.cnt;
    }
    ValU result(const Acc& a) const {
        ValU r;
        if (comb == C_COUNT) {
            r.i = a.cnt;
            r.f = (double)a.cnt;
        } else if (comb == C_AVG) {
            r.f = a.cnt ? a.f / (double)a.cnt : 0.0;
            r.i = (int64_t)r.f;
        } else {
            r.i = a.i;
            r.f = a.f;
        }
        return r;
    }
};

inline ValU read_val(Batch* b, int col, int64_t i) {
    ValU v;
    switch (b->schema.payload[col]) {
        case DType::I64: v.i = b->col<int64_t>(col)[i]; v.f = (double)v.i; break;
        case DType::I32: v.i = b->col<int32_t>(col)[i]; v.f = (double)v.i; break;
        case DType::F64: v.f = b->col<double>(col)[i]; v.i = (int64_t)v.f; break;
        case DType::F32: v.f = b->col<float>(col)[i]; v.i = (int64_t)v.f; break;
        case DType::U64: v.i = (int64_t)b->col<uint64_t>(col)[i]; v.f = (double)v.i; break;
        default: throw std::runtime_error("unsupported window value dtype");
    }
    return v;
}

inline void write_val(Batch* o, int col, int64_t i, const ValU& v, bool int_path) {
    switch (o->schema.payload[col]) {
        case DType::I64: o->col<int64_t>(col)[i] = int_path ? v.i : (int64_t)v.f; break;
        case DType::I32: o->col<int32_t>(col)[i] = (int32_t)(int_path ? v.i : (int64_t)v.f); break;
        case DType::F64: o->col<double>(col)[i] = int_path ? (double)v.i : v.f; break;
        case DType::F32: o->col<float>(col)[i] = (float)(int_path ? (double)v.i : v.f); break;
        case DType::U64: o->col<uint64_t>(col)[i] = (uint64_t)(int_path ? v.i : (int64_t)v.f); break;
        default: throw std::runtime_error("unsupported window out dtype");
    }
}

// batches result rows; flushed with the current watermark
struct OutBuf {
    Batch* b = nullptr;
    Batch* row(EmitCtx& out) {
        if (!b) b = out.new_batch();
        return b;
    }
    int64_t slot(EmitCtx& out, int64_t wm) {
        Batch* o = row(out);
        if (o->count == o->capacity) {
            flush(out, wm);
            o = row(out);
        }
        return o->count++;
    }
    void flush(EmitCtx& out, int64_t wm) {
        if (!b) return;
        if (b->count) {
            b->watermark = wm;
            out.emit(b);
        } else {
            release(b);
        }
        b = nullptr;
    }
};

// ===================== incremental window core =====================
// One open window = one accumulator (reference window_replica.hpp incremental
// path).  Covers keyed/parallel windows, the PLQ pane stage and the
// MapReduce MAP stage via flags.
struct WindowCore : OpLogic {
    // config
    WinType wt;
    int64_t win, slide, lateness;
    AggCfg agg;
    int col;
    int own_mode;       // 0: own all; 1: own gwid % parallelism == replica
    bool emit_meta;     // plq/mr_map: rows are (c0=gwid, c1=partial)
    bool emit_start_ts; // plq: result ts = window start (pane start)
    bool subset_rr;     // mr_map: accumulate only arrival idx % par == replica
    bool sparse = false;  // plq TB: tumbling panes, only touched panes open
                          // (gap panes would flood sparse keys; the WLQ
                          // treats missing panes as the combine identity)
    Engine* eng;

    struct Open {
        int64_t gwid;
        int64_t start;  // CB: per-key tuple index; TB: event ts
        Acc acc;
        bool owned;
    };
    struct KeyState {
        int64_t idx = 0;        // CB arrival count
        int64_t next_gwid = -1; // next window to open (-1: uninitialized)
        int64_t first_unfired = 0;  // sparse mode: panes below this fired
        int64_t last_ts = 0;
        int64_t cum_i = 0;      // CB sum prefix (prefix mode)
        double cum_f = 0;
        int64_t cum_cnt = 0;    // wedge mode: prefix count of (mine) tuples
        std::deque<std::pair<int64_t, ValU>> wd;  // monotonic wedge (idx, v)
        std::deque<Open> open;
    };
    std::unordered_map<uint64_t, KeyState> keys;
    // TB fire schedule: (fire_at = end + lateness, key)
    using HeapEl = std::pair<int64_t, uint64_t>;
    std::priority_queue<HeapEl, std::vector<HeapEl>, std::greater<HeapEl>> fire_heap;
    OutBuf ob;
    int64_t cur_wm = 0;
    // wm stamped on INTERMEDIATE output batches of one processing round:
    // the pre-round watermark.  Stamping the updated cur_wm on a batch
    // whose successors (same round, OutBuf overflow) still carry ts <= wm
    // breaks the watermark contract downstream (found by the 300-config
    // fuzz: paned TB windows fired before their late panes arrived).
    int64_t emit_wm = 0;
    int64_t rr_ctr = 0;  // mr_map round-robin arrival counter
    int replica = 0, par = 1;
    int64_t ignored = 0;
    // CB + sum/count: window result = prefix@fire - prefix@open, O(1) per
    // tuple instead of one add per open window (win/slide adds)
    bool prefix = false;
    // CB + min/max: monotonic wedge (sliding-window minimum), O(1)
    // amortized per tuple; window result = wedge front at fire
    bool wedge = false;

    WindowCore(WinType wt_, int64_t w, int64_t s, int64_t lat, AggCfg a, int c,
               int own, bool meta, bool start_ts, bool rr, Engine* e)
        : wt(wt_), win(w), slide(s), lateness(lat), agg(a), col(c), own_mode(own),
          emit_meta(meta), emit_start_ts(start_ts), subset_rr(rr), eng(e) {
        prefix = wt == WinType::CB &&
                 (agg.comb == C_SUM || agg.comb == C_COUNT);
        wedge = wt == WinType::CB && (agg.comb == C_MIN || agg.comb == C_MAX);
    }

    void warm(RuntimeCtx& ctx) override {
        replica = ctx.replica;
        par = ctx.parallelism;
    }
    bool owned(int64_t gwid) const {
        return own_mode == 0 || (gwid % par) == (int64_t)replica;
    }

    void fire(Open& w, uint64_t key, int64_t res_ts, EmitCtx& out) {
        if (!w.owned) return;
        int64_t i = ob.slot(out, emit_wm);
        Batch* o = ob.b;
        o->ts[i] = res_ts;
        o->key[i] = key;
        ValU r = agg.result(w.acc);
        if (emit_meta) {
            o->col<int64_t>(0)[i] = w.gwid;
            // AVG partials carry the raw SUM (the slice count rides in c2);
            // emitting the slice MEAN would make the recombiner divide twice
            if (agg.comb == C_AVG) {
                r.i = w.acc.i;
                r.f = w.acc.f;
            }
            write_val(o, 1, i, r, agg.use_int);
            // tuple count of the slice: empty partials (cnt 0) must not
            // poison MIN/MAX downstream and AVG needs real weights
            o->col<int64_t>(2)[i] = w.acc.cnt;
        } else {
            write_val(o, 0, i, r, agg.use_int);
        }
    }

    void add_cb(uint64_t key, KeyState& ks, int64_t ts, const ValU& v, bool mine,
                EmitCtx& out) {
        if (ks.next_gwid < 0) ks.next_gwid = 0;
        while (ks.next_gwid * slide <= ks.idx) {
            Open w{ks.next_gwid, ks.next_gwid * slide, {}, owned(ks.next_gwid)};
            if (prefix) {  // remember the prefix at window open
                w.acc.i = ks.cum_i;
                w.acc.f = ks.cum_f;
            }
            if (wedge) w.acc.cnt = ks.cum_cnt;  // count@open (empty detection)
            ks.open.push_back(w);
            ks.next_gwid++;
        }
        if (prefix) {
            if (mine) {  // COUNT runs the prefix over 1s (subset-correct)
                ks.cum_i += agg.comb == C_COUNT ? 1 : v.i;
                ks.cum_f += agg.comb == C_COUNT ? 1.0 : v.f;
            }
        } else if (wedge) {
            if (mine) {
                auto& wd = ks.wd;
                if (agg.use_int) {
                    while (!wd.empty() &&
                           (agg.comb == C_MIN ? wd.back().second.i >= v.i
                                              : wd.back().second.i <= v.i))
                        wd.pop_back();
                } else {
                    while (!wd.empty() &&
                           (agg.comb == C_MIN ? wd.back().second.f >= v.f
                                              : wd.back().second.f <= v.f))
                        wd.pop_back();
                }
                wd.emplace_back(ks.idx, v);
                ks.cum_cnt++;
            }
        } else {
            for (auto& w : ks.open)
                if (w.owned && mine && ks.idx >= w.start) agg.add(w.acc, v);
        }
        ks.last_ts = ts;
        while (!ks.open.empty() && ks.idx == ks.open.front().start + win - 1) {
            Open& w = ks.open.front();
            if (prefix) materialize_prefix(ks, w);
            if (wedge) materialize_wedge(ks, w);
            fire(w, key, ts, out);
            ks.open.pop_front();
        }
        ks.idx++;
    }

    void materialize_wedge(KeyState& ks, Open& w) {
        while (!ks.wd.empty() && ks.wd.front().first < w.start) ks.wd.pop_front();
        Acc a;
        a.cnt = ks.cum_cnt - w.acc.cnt;
        if (a.cnt > 0 && !ks.wd.empty()) {
            a.i = ks.wd.front().second.i;
            a.f = ks.wd.front().second.f;
        }
        w.acc = a;
    }

    // turn (prefix@open stored in acc) into the actual window accumulator
    void materialize_prefix(KeyState& ks, Open& w) {
        Acc a;
        a.i = ks.cum_i - w.acc.i;
        a.f = ks.cum_f - w.acc.f;
        a.cnt = agg.comb == C_COUNT
                    ? a.i  // prefix of 1s (exact under mr_map subsets)
                    : std::min(ks.idx + 1, w.start + win) - w.start;
        w.acc = a;
    }

    void add_tb_sparse(uint64_t key, KeyState& ks, int64_t ts, const ValU& v,
                       bool mine, EmitCtx& out) {
        // tumbling panes (win == slide): exactly one pane contains ts
        const int64_t g = ts / slide;
        if (g < ks.first_unfired) {
            ignored++;
            return;
        }
        // opens are few and mostly increasing: scan from the back
        Open* w = nullptr;
        for (auto it = ks.open.rbegin(); it != ks.open.rend(); ++it) {
            if (it->gwid == g) {
                w = &*it;
                break;
            }
            if (it->gwid < g) break;
        }
        if (!w) {
            auto pos = ks.open.end();
            while (pos != ks.open.begin() && std::prev(pos)->gwid > g) --pos;
            pos = ks.open.insert(pos, {g, g * slide, {}, owned(g)});
            fire_heap.emplace(g * slide + win + lateness, key);
            w = &*pos;
        }
        if (w->owned && mine) agg.add(w->acc, v);
        ks.last_ts = std::max(ks.last_ts, ts);
    }

    void add_tb(uint64_t key, KeyState& ks, int64_t ts, const ValU& v, bool mine,
                EmitCtx& out) {
        if (sparse) {
            add_tb_sparse(key, ks, ts, v, mine, out);
            return;
        }
        if (ks.next_gwid < 0) {
            // first window containing the key's first tuple
            int64_t w0 = (ts - win + 1);
            w0 = w0 <= 0 ? 0 : (w0 + slide - 1) / slide;
            ks.next_gwid = w0;
        }
        while (ks.next_gwid * slide <= ts) {
            int64_t g = ks.next_gwid++;
            ks.open.push_back({g, g * slide, {}, owned(g)});
            fire_heap.emplace(g * slide + win + lateness, key);
        }
        bool hit = false;
        for (auto& w : ks.open) {
            if (ts >= w.start && ts < w.start + win) {
                hit = true;
                if (w.owned && mine) agg.add(w.acc, v);
            }
        }
        if (!hit) ignored++;  // late beyond every open window
        ks.last_ts = std::max(ks.last_ts, ts);
    }

    void fire_tb(EmitCtx& out) {
        while (!fire_heap.empty() && fire_heap.top().first <= cur_wm) {
            uint64_t key = fire_heap.top().second;
            fire_heap.pop();
            auto it = keys.find(key);
            if (it == keys.end()) continue;
            auto& ks = it->second;
            while (!ks.open.empty() &&
                   ks.open.front().start + win + lateness <= cur_wm) {
                Open& w = ks.open.front();
                fire(w, key, emit_start_ts ? w.start : w.start + win - 1, out);
                ks.first_unfired = std::max(ks.first_unfired, w.gwid + 1);
                ks.open.pop_front();
            }
        }
    }

    void process(Batch* b, EmitCtx& out, RuntimeCtx& ctx) override {
        emit_wm = cur_wm;
        const int64_t n = b->count;
        for (int64_t i = 0; i < n; ++i) {
            bool mine = !subset_rr || (rr_ctr % par) == (int64_t)replica;
            rr_ctr++;
            auto& ks = keys[b->key[i]];
            ValU v = read_val(b, col, i);
            if (wt == WinType::CB)
                add_cb(b->key[i], ks, b->ts[i], v, mine, out);
            else
                add_tb(b->key[i], ks, b->ts[i], v, mine, out);
        }
        cur_wm = std::max(cur_wm, ctx.current_wm);
        release(b);
        if (wt == WinType::TB) fire_tb(out);
        ob.flush(out, cur_wm);
    }

    bool on_punct(int64_t wm, EmitCtx& out, RuntimeCtx&) override {
        emit_wm = cur_wm;
        cur_wm = std::max(cur_wm, wm);
        if (wt == WinType::TB) fire_tb(out);
        ob.flush(out, cur_wm);
        return false;  // propagate the punctuation after any results
    }

    void on_eos(EmitCtx& out, RuntimeCtx&) override {
        emit_wm = cur_wm;
        // flush every remaining open window (reference window_replica
        // eosnotify, :356-408)
        for (auto& [key, ks] : keys) {
            for (auto& w : ks.open) {
                // empty TB windows fire with the default result (0), same as
                // the steady-state heap path and the reference's FIRED
                // triggerer — skipping them here made EOS semantics depend
                // on how far the watermark happened to advance (found by
                // the 300-config fuzz campaign)
                if (prefix) {
                    Acc a;
                    a.i = ks.cum_i - w.acc.i;
                    a.f = ks.cum_f - w.acc.f;
                    a.cnt = agg.comb == C_COUNT ? a.i : ks.idx - w.start;
                    w.acc = a;
                }
                if (wedge && wt == WinType::CB) materialize_wedge(ks, w);
                fire(w, key,
                     wt == WinType::CB ? ks.last_ts
                                       : (emit_start_ts ? w.start
                                                        : std::min(ks.last_ts, w.start + win - 1)),
                     out);
            }
            ks.open.clear();
        }
        ob.flush(out, cur_wm);
    }
};

// ===================== WLQ: window over pane partials =====================
// Consumes (ts = pane start ts or per-key pane index, c0 = pane gwid,
// c1 = partial) rows from win_plq.  A window w covers panes
// [w*slide_p, w*slide_p + win_p).  CB fires on completeness (every pane of
// the window received — PLQ emits gap panes); TB fires on the watermark.
struct KeyGwidHash {
    size_t operator()(const std::pair<uint64_t, int64_t>& p) const {
        uint64_t h = p.first * 0x9e3779b97f4a7c15ULL ^ (uint64_t)p.second;
        h ^= h >> 29;
        h *= 0xbf58476d1ce4e5b9ULL;
        return (size_t)(h ^ (h >> 32));
    }
};

struct WlqLogic : OpLogic {
    WinType wt;
    int64_t win_p, slide_p, lateness, pane_len;
    AggCfg agg;
    int own_mode;
    int replica = 0, par = 1;

    struct WinAcc {
        Acc acc;
        int64_t panes = 0;
        int64_t last_ts = 0;
    };
    // (key, gwid) -> partial window
    std::unordered_map<std::pair<uint64_t, int64_t>, WinAcc, KeyGwidHash> wins;
    using HeapEl = std::pair<int64_t, std::pair<uint64_t, int64_t>>;
    std::priority_queue<HeapEl, std::vector<HeapEl>, std::greater<HeapEl>> fire_heap;
    OutBuf ob;
    int64_t cur_wm = 0;
    // wm stamped on INTERMEDIATE output batches of one processing round:
    // the pre-round watermark.  Stamping the updated cur_wm on a batch
    // whose successors (same round, OutBuf overflow) still carry ts <= wm
    // breaks the watermark contract downstream (found by the 300-config
    // fuzz: paned TB windows fired before their late panes arrived).
    int64_t emit_wm = 0;

    WlqLogic(WinType wt_, int64_t wp, int64_t sp, int64_t lat, AggCfg a,
             int64_t pl, int own)
        : wt(wt_), win_p(wp), slide_p(sp), lateness(lat), pane_len(pl), agg(a),
          own_mode(own) {}

    void warm(RuntimeCtx& ctx) override {
        replica = ctx.replica;
        par = ctx.parallelism;
    }
    bool owned(int64_t g) const {
        return own_mode == 0 || (g % par) == (int64_t)replica;
    }

    void fire(const std::pair<uint64_t, int64_t>& wk, WinAcc& w, EmitCtx& out) {
        int64_t i = ob.slot(out, emit_wm);
        Batch* o = ob.b;
        o->key[i] = wk.first;
        o->ts[i] = wt == WinType::CB ? w.last_ts
                                     : (wk.second * slide_p + win_p) * pane_len - 1;
        write_val(o, 0, i, agg.result(w.acc), agg.use_int);
    }

    void process(Batch* b, EmitCtx& out, RuntimeCtx& ctx) override {
        emit_wm = cur_wm;
        const int64_t n = b->count;
        for (int64_t i = 0; i < n; ++i) {
            uint64_t key = b->key[i];
            int64_t p = b->col<int64_t>(0)[i];
            Acc part;
            ValU v = read_val(b, 1, i);
            part.i = v.i;
            part.f = v.f;
            part.cnt = b->col<int64_t>(2)[i];  // real slice count (0 = empty)
            int64_t w_lo = p - win_p + 1;
            w_lo = w_lo <= 0 ? 0 : (w_lo + slide_p - 1) / slide_p;
            int64_t w_hi = p / slide_p;
            for (int64_t w = w_lo; w <= w_hi; ++w) {
                if (!owned(w)) continue;
                auto kk = std::make_pair(key, w);
                auto [it, inserted] = wins.try_emplace(kk);
                if (inserted && wt == WinType::TB)
                    fire_heap.emplace((w * slide_p + win_p) * pane_len + lateness, kk);
                agg.merge(it->second.acc, part);
                it->second.panes++;
                it->second.last_ts = std::max(it->second.last_ts, b->ts[i]);
                if (wt == WinType::CB && it->second.panes == win_p) {
                    fire(kk, it->second, out);
                    wins.erase(it);
                }
            }
        }
        cur_wm = std::max(cur_wm, ctx.current_wm);
        release(b);
        if (wt == WinType::TB) fire_tb(out);
        ob.flush(out, cur_wm);
    }

    void fire_tb(EmitCtx& out) {
        while (!fire_heap.empty() && fire_heap.top().first <= cur_wm) {
            auto kk = fire_heap.top().second;
            fire_heap.pop();
            auto it = wins.find(kk);
            if (it == wins.end()) continue;
            fire(kk, it->second, out);
            wins.erase(it);
        }
    }

    bool on_punct(int64_t wm, EmitCtx& out, RuntimeCtx&) override {
        emit_wm = cur_wm;
        cur_wm = std::max(cur_wm, wm);
        if (wt == WinType::TB) fire_tb(out);
        ob.flush(out, cur_wm);
        return false;
    }

    void on_eos(EmitCtx& out, RuntimeCtx&) override {
        emit_wm = cur_wm;
        for (auto& [kk, w] : wins) fire(kk, const_cast<WinAcc&>(w), out);
        wins.clear();
        ob.flush(out, cur_wm);
    }
};

// WLQ, keyby-partitioned fast path: a key's panes arrive IN ORDER from its
// single owning PLQ replica, so the window recombination runs over an
// ordered per-key deque (front-scan per fire) instead of per-(key,gwid)
// hash accumulation — no hash ops, cache-friendly, ~5x on dense configs.
struct WlqKeyedLogic : OpLogic {
    WinType wt;
    int64_t win_p, slide_p, lateness, pane_len;
    AggCfg agg;

    struct KeyState {
        std::deque<std::pair<int64_t, Acc>> buf;  // (pane id, partial), ordered
        int64_t next_w = -1;      // next window to fire
        int64_t last_id = -1;     // newest pane id seen
        int64_t last_ts = 0;
        bool armed = false;       // TB: heap entry outstanding for next_w
    };
    std::unordered_map<uint64_t, KeyState> keys;
    using HeapEl = std::pair<int64_t, uint64_t>;
    std::priority_queue<HeapEl, std::vector<HeapEl>, std::greater<HeapEl>> fire_heap;
    OutBuf ob;
    int64_t cur_wm = 0;
    // wm stamped on INTERMEDIATE output batches of one processing round:
    // the pre-round watermark.  Stamping the updated cur_wm on a batch
    // whose successors (same round, OutBuf overflow) still carry ts <= wm
    // breaks the watermark contract downstream (found by the 300-config
    // fuzz: paned TB windows fired before their late panes arrived).
    int64_t emit_wm = 0;
    int64_t ignored = 0;

    WlqKeyedLogic(WinType wt_, int64_t wp, int64_t sp, int64_t lat, AggCfg a,
                  int64_t pl)
        : wt(wt_), win_p(wp), slide_p(sp), lateness(lat), pane_len(pl), agg(a) {}

    int64_t fire_at(int64_t w) const {
        return (w * slide_p + win_p) * pane_len + lateness;
    }

    void fire(uint64_t key, KeyState& ks, EmitCtx& out) {
        const int64_t wS = ks.next_w * slide_p, wE = wS + win_p;
        Acc acc;
        for (auto& [id, part] : ks.buf) {
            if (id >= wE) break;
            agg.merge(acc, part);  // ids < wS were dropped after earlier fires
        }
        int64_t i = ob.slot(out, emit_wm);
        Batch* o = ob.b;
        o->key[i] = key;
        o->ts[i] = wt == WinType::CB ? ks.last_ts : wE * pane_len - 1;
        write_val(o, 0, i, agg.result(acc), agg.use_int);
        ks.next_w++;
        const int64_t keep_from = ks.next_w * slide_p;
        while (!ks.buf.empty() && ks.buf.front().first < keep_from)
            ks.buf.pop_front();
    }

    void process(Batch* b, EmitCtx& out, RuntimeCtx& ctx) override {
        emit_wm = cur_wm;
        const int64_t n = b->count;
        for (int64_t i = 0; i < n; ++i) {
            uint64_t key = b->key[i];
            int64_t p = b->col<int64_t>(0)[i];
            auto& ks = keys[key];
            if (ks.next_w < 0) {
                int64_t w0 = p - win_p + 1;
                ks.next_w = w0 <= 0 ? 0 : (w0 + slide_p - 1) / slide_p;
            }
            if (p < ks.next_w * slide_p) {  // pane for an already-fired window
                ignored++;
                continue;
            }
            Acc part;
            ValU v = read_val(b, 1, i);
            part.i = v.i;
            part.f = v.f;
            part.cnt = b->col<int64_t>(2)[i];  // real slice count (0 = empty)
            ks.buf.emplace_back(p, part);
            ks.last_id = std::max(ks.last_id, p);
            ks.last_ts = std::max(ks.last_ts, b->ts[i]);
            if (wt == WinType::CB) {
                // CB panes are dense per key: window complete when its last
                // pane (wS + win_p - 1) arrives
                while (ks.next_w * slide_p + win_p - 1 <= ks.last_id)
                    fire(key, ks, out);
            } else if (!ks.armed) {
                fire_heap.emplace(fire_at(ks.next_w), key);
                ks.armed = true;
            }
        }
        cur_wm = std::max(cur_wm, ctx.current_wm);
        release(b);
        if (wt == WinType::TB) fire_tb(out);
        ob.flush(out, cur_wm);
    }

    void fire_tb(EmitCtx& out) {
        while (!fire_heap.empty() && fire_heap.top().first <= cur_wm) {
            uint64_t key = fire_heap.top().second;
            fire_heap.pop();
            auto& ks = keys[key];
            ks.armed = false;
            while (fire_at(ks.next_w) <= cur_wm &&
                   ks.next_w * slide_p <= ks.last_id)
                fire(key, ks, out);
            if (ks.next_w * slide_p <= ks.last_id || !ks.buf.empty()) {
                fire_heap.emplace(fire_at(ks.next_w), key);
                ks.armed = true;
            }
        }
    }

    bool on_punct(int64_t wm, EmitCtx& out, RuntimeCtx&) override {
        emit_wm = cur_wm;
        cur_wm = std::max(cur_wm, wm);
        if (wt == WinType::TB) fire_tb(out);
        ob.flush(out, cur_wm);
        return false;
    }

    void on_eos(EmitCtx& out, RuntimeCtx&) override {
        emit_wm = cur_wm;
        for (auto& [key, ks] : keys)
            while (ks.next_w >= 0 && ks.next_w * slide_p <= ks.last_id &&
                   !ks.buf.empty())
                fire(key, ks, out);
        ob.flush(out, cur_wm);
    }
};

// ============== MapReduce REDUCE: merge n partials per window ==============
struct MrReduceLogic : OpLogic {
    int64_t n_partials;
    AggCfg agg;
    struct WinAcc {
        Acc acc;
        int64_t got = 0;
        int64_t last_ts = 0;
    };
    std::unordered_map<std::pair<uint64_t, int64_t>, WinAcc, KeyGwidHash> wins;
    OutBuf ob;
    int64_t cur_wm = 0;
    // wm stamped on INTERMEDIATE output batches of one processing round:
    // the pre-round watermark.  Stamping the updated cur_wm on a batch
    // whose successors (same round, OutBuf overflow) still carry ts <= wm
    // breaks the watermark contract downstream (found by the 300-config
    // fuzz: paned TB windows fired before their late panes arrived).
    int64_t emit_wm = 0;

    MrReduceLogic(int64_t n, AggCfg a) : n_partials(n), agg(a) {}

    void fire(const std::pair<uint64_t, int64_t>& kk, WinAcc& w, EmitCtx& out) {
        int64_t i = ob.slot(out, emit_wm);
        Batch* o = ob.b;
        o->key[i] = kk.first;
        o->ts[i] = w.last_ts;
        write_val(o, 0, i, agg.result(w.acc), agg.use_int);
    }

    void process(Batch* b, EmitCtx& out, RuntimeCtx& ctx) override {
        emit_wm = cur_wm;
        for (int64_t i = 0; i < b->count; ++i) {
            auto kk = std::make_pair(b->key[i], b->col<int64_t>(0)[i]);
            auto& w = wins[kk];
            Acc part;
            ValU v = read_val(b, 1, i);
            part.i = v.i;
            part.f = v.f;
            part.cnt = b->col<int64_t>(2)[i];  // real slice count (0 = empty)
            agg.merge(w.acc, part);
            w.got++;
            w.last_ts = std::max(w.last_ts, b->ts[i]);
            if (w.got == n_partials) {
                fire(kk, w, out);
                wins.erase(kk);
            }
        }
        cur_wm = std::max(cur_wm, ctx.current_wm);
        release(b);
        ob.flush(out, cur_wm);
    }

    void on_eos(EmitCtx& out, RuntimeCtx&) override {
        emit_wm = cur_wm;
        for (auto& [kk, w] : wins) fire(kk, const_cast<WinAcc&>(w), out);
        wins.clear();
        ob.flush(out, cur_wm);
    }
};

// ===================== FlatFAT (CPU) =====================
// Flat binary aggregation tree over a circular leaf window (reference
// wf/flatfat.hpp, Tangwongsan VLDB'15): insert = leaf write + root path
// update (O(log n)); window result = O(log n) segment-tree range combine.
struct FlatFAT {
    int64_t cap = 0;  // leaves (power of two)
    std::vector<Acc> tree;  // [2*cap]; node 1 = root, leaves cap..2cap-1
    int64_t head = 0;       // absolute index of next insert
    const AggCfg* agg = nullptr;

    void init(int64_t min_leaves, const AggCfg* a) {
        cap = 1;
        while (cap < min_leaves) cap <<= 1;
        tree.assign(2 * cap, Acc{});
        agg = a;
        head = 0;
    }
    void insert(const Acc& leaf) {
        int64_t n = cap + (head & (cap - 1));
        tree[n] = leaf;
        for (n >>= 1; n >= 1; n >>= 1) {
            Acc a = tree[2 * n];
            agg->merge(a, tree[2 * n + 1]);
            tree[n] = a;
        }
        head++;
    }
    // combine leaves [head-win, head) (absolute), circular
    Acc query_last(int64_t win) {
        Acc r;
        int64_t lo = head - win;
        if (lo < 0) lo = 0;
        int64_t a = lo & (cap - 1), b = ((head - 1) & (cap - 1)) + 1;
        if (head - lo == cap) { a = 0; b = cap; }
        if (a < b) {
            range(a, b, r);
        } else {
            range(a, cap, r);
            range(0, b, r);
        }
        return r;
    }
    void range(int64_t l, int64_t r, Acc& out) {  // leaf indices [l, r)
        int64_t lo = l + cap, hi = r + cap;
        std::vector<Acc> right;  // keep combine order (non-commutative safe)
        while (lo < hi) {
            if (lo & 1) agg->merge(out, tree[lo++]);
            if (hi & 1) right.push_back(tree[--hi]);
            lo >>= 1;
            hi >>= 1;
        }
        for (auto it = right.rbegin(); it != right.rend(); ++it)
            agg->merge(out, *it);
    }
};

// Keyed FFAT windows: CB inserts tuples as leaves; TB lifts tuples into
// pane partials (pane = gcd(win, slide)), completes panes on the watermark
// and inserts panes as leaves (reference ffat_replica.hpp CB/TB paths).
struct FfatCpuLogic : OpLogic {
    WinType wt;
    int64_t win, slide, lateness;
    AggCfg agg;
    int col;
    int64_t pane_len = 1, win_p = 0, slide_p = 0;

    struct KeyState {
        FlatFAT fat;
        int64_t idx = 0;       // CB tuples inserted / TB panes inserted
        int64_t last_ts = 0;
        // TB pending panes (absolute pane id -> partial)
        std::map<int64_t, Acc> pending;
        int64_t next_pane = -1;  // next absolute pane to insert
        int64_t first_pane = 0;
        bool init = false;
    };
    std::unordered_map<uint64_t, KeyState> keys;
    OutBuf ob;
    int64_t cur_wm = 0;
    // wm stamped on INTERMEDIATE output batches of one processing round:
    // the pre-round watermark.  Stamping the updated cur_wm on a batch
    // whose successors (same round, OutBuf overflow) still carry ts <= wm
    // breaks the watermark contract downstream (found by the 300-config
    // fuzz: paned TB windows fired before their late panes arrived).
    int64_t emit_wm = 0;

    FfatCpuLogic(WinType wt_, int64_t w, int64_t s, int64_t lat, AggCfg a, int c)
        : wt(wt_), win(w), slide(s), lateness(lat), agg(a), col(c) {
        if (wt == WinType::TB) {
            pane_len = std::gcd(win, slide);
            win_p = win / pane_len;
            slide_p = slide / pane_len;
        }
    }

    void fire(uint64_t key, KeyState& ks, int64_t res_ts, int64_t span,
              EmitCtx& out) {
        Acc r = ks.fat.query_last(span);
        int64_t i = ob.slot(out, emit_wm);
        Batch* o = ob.b;
        o->ts[i] = res_ts;
        o->key[i] = key;
        write_val(o, 0, i, agg.result(r), agg.use_int);
    }

    void process(Batch* b, EmitCtx& out, RuntimeCtx& ctx) override {
        emit_wm = cur_wm;
        for (int64_t i = 0; i < b->count; ++i) {
            auto& ks = keys[b->key[i]];
            if (!ks.init) {
                ks.fat.init(wt == WinType::CB ? win : win_p, &agg);
                ks.init = true;
            }
            ValU v = read_val(b, col, i);
            if (wt == WinType::CB) {
                Acc leaf;
                agg.add(leaf, v);
                ks.fat.insert(leaf);
                ks.idx++;
                ks.last_ts = b->ts[i];
                if (ks.idx >= win && (ks.idx - win) % slide == 0)
                    fire(b->key[i], ks, b->ts[i], win, out);
            } else {
                int64_t p = b->ts[i] / pane_len;
                if (ks.next_pane < 0) {
                    // align to the absolute window grid: start inserting at
                    // the first pane of the first window containing this
                    // tuple, so results match Keyed_Windows exactly
                    int64_t w0 = b->ts[i] - win + 1;
                    w0 = w0 <= 0 ? 0 : (w0 + slide - 1) / slide;
                    ks.next_pane = w0 * slide_p;
                    ks.first_pane = ks.next_pane;
                }
                if (p < ks.next_pane) continue;  // late: pane already closed
                agg.add(ks.pending[p], v);
                ks.last_ts = std::max(ks.last_ts, b->ts[i]);
            }
        }
        cur_wm = std::max(cur_wm, ctx.current_wm);
        release(b);
        if (wt == WinType::TB) complete_panes(out, false);
        ob.flush(out, cur_wm);
    }

    // insert panes completed by the watermark: pane p is complete when
    // (p+1)*pane_len + lateness <= wm (reference ffat_replica_gpu:875-881)
    void complete_panes(EmitCtx& out, bool eos) {
        for (auto& [key, ks] : keys) {
            if (ks.next_pane < 0) continue;
            int64_t limit = eos ? INT64_MAX : (cur_wm - lateness) / pane_len - 1;
            int64_t last_have = ks.pending.empty() ? -1 : ks.pending.rbegin()->first;
            while (ks.next_pane <= limit &&
                   (ks.next_pane <= last_have)) {
                Acc pane;  // default (gap pane) unless pending
                auto it = ks.pending.find(ks.next_pane);
                if (it != ks.pending.end()) {
                    pane = it->second;
                    ks.pending.erase(it);
                }
                ks.fat.insert(pane);
                int64_t inserted = ks.next_pane - ks.first_pane + 1;
                if (inserted >= win_p && (inserted - win_p) % slide_p == 0)
                    fire(key, ks, (ks.next_pane + 1) * pane_len - 1, win_p, out);
                ks.next_pane++;
            }
        }
    }

    bool on_punct(int64_t wm, EmitCtx& out, RuntimeCtx&) override {
        emit_wm = cur_wm;
        cur_wm = std::max(cur_wm, wm);
        if (wt == WinType::TB) complete_panes(out, false);
        ob.flush(out, cur_wm);
        return false;
    }

    void on_eos(EmitCtx& out, RuntimeCtx&) override {
        emit_wm = cur_wm;
        if (wt == WinType::TB) {
            complete_panes(out, true);
            // flush partial pane-windows (every slide_p, shrinking span) —
            // mirrors Keyed_Windows' EOS flush of open windows
            for (auto& [key, ks] : keys) {
                if (ks.next_pane < 0) continue;
                int64_t ins = ks.next_pane - ks.first_pane;  // panes inserted
                if (ins == 0) continue;
                int64_t first = ins >= win_p ? (ins - win_p) / slide_p + 1 : 0;
                for (int64_t w = first; w * slide_p < ins; ++w)
                    fire(key, ks, ks.last_ts, ins - w * slide_p, out);
            }
        } else {
            // flush every open (partial) window — windows start each slide
            for (auto& [key, ks] : keys) {
                if (ks.idx == 0) continue;
                int64_t first = ks.idx >= win ? (ks.idx - win) / slide + 1 : 0;
                for (int64_t w = first; w * slide < ks.idx; ++w)
                    fire(key, ks, ks.last_ts, ks.idx - w * slide, out);
            }
        }
        ob.flush(out, cur_wm);
    }
};

// ===================== interval join =====================
// Pair (a, b) joins iff b.ts - a.ts in [lower, upper] and key matches.
// KP: keys partitioned (KEYBY).  DP: broadcast, replica stores its
// round-robin slice but probes everything; requires a consistent arrival
// order (ordering collector) so each pair fires exactly once
// (reference interval_join.hpp KP/DP, join_collector.hpp).
struct IntervalJoinLogic : OpLogic {
    int mode;  // 0 KP, 1 DP
    int64_t lower, upper;
    int colA, colB;
    Engine* eng;
    int replica = 0, par = 1;

    struct Entry {
        int64_t ts;
        ValU v;
    };
    struct KeyState {
        std::vector<Entry> a, b;  // sorted by ts
    };
    std::unordered_map<uint64_t, KeyState> keys;
    OutBuf ob;
    int64_t cur_wm = 0;
    // wm stamped on INTERMEDIATE output batches of one processing round:
    // the pre-round watermark.  Stamping the updated cur_wm on a batch
    // whose successors (same round, OutBuf overflow) still carry ts <= wm
    // breaks the watermark contract downstream (found by the 300-config
    // fuzz: paned TB windows fired before their late panes arrived).
    int64_t emit_wm = 0;
    int64_t store_ctr = 0;
    bool a_int = true, b_int = true;
    // user predicate/result (reference interval_join.hpp:279-307): when
    // set, matched pairs are STAGED and the vectorized callback decides
    // keep + result value per pair (one call per processed batch)
    JoinFn pairfn = nullptr;
    std::vector<uint64_t> p_key;
    std::vector<int64_t> p_tsa, p_tsb;
    std::vector<double> p_va, p_vb;

    IntervalJoinLogic(int m, int64_t lo, int64_t up, int ca, int cb, Engine* e)
        : mode(m), lower(lo), upper(up), colA(ca), colB(cb), eng(e) {}

    void warm(RuntimeCtx& ctx) override {
        replica = ctx.replica;
        par = ctx.parallelism;
    }

    static void insert_sorted(std::vector<Entry>& v, Entry e) {
        auto it = std::upper_bound(v.begin(), v.end(), e,
                                   [](const Entry& x, const Entry& y) { return x.ts < y.ts; });
        v.insert(it, e);
    }

    void emit_pair(uint64_t key, const Entry& ea, const Entry& eb, EmitCtx& out) {
        if (pairfn) {
            p_key.push_back(key);
            p_tsa.push_back(ea.ts);
            p_tsb.push_back(eb.ts);
            p_va.push_back(a_int ? (double)ea.v.i : ea.v.f);
            p_vb.push_back(b_int ? (double)eb.v.i : eb.v.f);
            return;
        }
        int64_t i = ob.slot(out, emit_wm);
        Batch* o = ob.b;
        o->ts[i] = std::max(ea.ts, eb.ts);
        o->key[i] = key;
        write_val(o, 0, i, ea.v, a_int);
        write_val(o, 1, i, eb.v, b_int);
    }

    void flush_pairs(EmitCtx& out) {
        if (!pairfn || p_key.empty()) return;
        const int64_t n = (int64_t)p_key.size();
        std::vector<uint8_t> keep(n, 0);
        std::vector<double> res(n, 0.0);
        JoinPairs jp{n, p_key.data(), p_tsa.data(), p_tsb.data(), p_va.data(),
                     p_vb.data()};
        pairfn(jp, keep.data(), res.data());
        for (int64_t i = 0; i < n; ++i) {
            if (!keep[i]) continue;
            int64_t w = ob.slot(out, emit_wm);
            Batch* o = ob.b;
            o->ts[w] = std::max(p_tsa[i], p_tsb[i]);
            o->key[w] = p_key[i];
            ValU v;
            v.f = res[i];
            v.i = (int64_t)res[i];
            bool int_out = o->schema.payload.size() > 0 &&
                           (o->schema.payload[0] == DType::I64 ||
                            o->schema.payload[0] == DType::I32);
            write_val(o, 0, w, v, int_out);
        }
        p_key.clear();
        p_tsa.clear();
        p_tsb.clear();
        p_va.clear();
        p_vb.clear();
    }

    void process(Batch* b, EmitCtx& out, RuntimeCtx& ctx) override {
        emit_wm = cur_wm;
        int tag = ctx.current_tag;
        int vcol = tag == 1 ? colB : colA;
        bool is_int = b->schema.payload[vcol] == DType::I64 ||
                      b->schema.payload[vcol] == DType::I32;
        if (tag == 1) b_int = is_int; else a_int = is_int;
        for (int64_t i = 0; i < b->count; ++i) {
            int64_t ts = b->ts[i];
            if (eng->mode == ExecMode::DEFAULT && ts < cur_wm) {
                eng->dropped_tuples.fetch_add(1, std::memory_order_relaxed);
                continue;
            }
            uint64_t key = b->key[i];
            auto& ks = keys[key];
            Entry e{ts, read_val(b, vcol, i)};
            // probe the other archive
            if (tag == 1) {
                // b arrived: a.ts in [ts-upper, ts-lower]
                auto& arc = ks.a;
                auto lo = std::lower_bound(arc.begin(), arc.end(), ts - upper,
                                           [](const Entry& x, int64_t t) { return x.ts < t; });
                for (auto it = lo; it != arc.end() && it->ts <= ts - lower; ++it)
                    emit_pair(key, *it, e, out);
            } else {
                // a arrived: b.ts in [ts+lower, ts+upper]
                auto& arc = ks.b;
                auto lo = std::lower_bound(arc.begin(), arc.end(), ts + lower,
                                           [](const Entry& x, int64_t t) { return x.ts < t; });
                for (auto it = lo; it != arc.end() && it->ts <= ts + upper; ++it)
                    emit_pair(key, e, *it, out);
            }
            // store (DP: only my round-robin slice)
            bool store = mode == 0 || (store_ctr % par) == (int64_t)replica;
            store_ctr++;
            if (store) insert_sorted(tag == 1 ? ks.b : ks.a, e);
            purge_key(ks);
        }
        cur_wm = std::max(cur_wm, ctx.current_wm);
        release(b);
        flush_pairs(out);
        ob.flush(out, cur_wm);
    }

    // purge one key's archives up to the current watermark bounds (called
    // inline per touched key — a full-keyspace sweep per batch is O(keys))
    void purge_key(KeyState& ks) {
        int64_t a_min = cur_wm - upper, b_min = cur_wm + lower;
        if (!ks.a.empty() && ks.a.front().ts < a_min) {
            auto pa = std::lower_bound(ks.a.begin(), ks.a.end(), a_min,
                                       [](const Entry& x, int64_t t) { return x.ts < t; });
            ks.a.erase(ks.a.begin(), pa);
        }
        if (!ks.b.empty() && ks.b.front().ts < b_min) {
            auto pb = std::lower_bound(ks.b.begin(), ks.b.end(), b_min,
                                       [](const Entry& x, int64_t t) { return x.ts < t; });
            ks.b.erase(ks.b.begin(), pb);
        }
    }

    void purge() {
        // a needed while future b may reach back: a.ts >= wm - upper
        // b needed while future a may reach forward: b.ts >= wm + lower
        int64_t a_min = cur_wm - upper, b_min = cur_wm + lower;
        for (auto& [k, ks] : keys) {
            auto pa = std::lower_bound(ks.a.begin(), ks.a.end(), a_min,
                                       [](const Entry& x, int64_t t) { return x.ts < t; });
            ks.a.erase(ks.a.begin(), pa);
            auto pb = std::lower_bound(ks.b.begin(), ks.b.end(), b_min,
                                       [](const Entry& x, int64_t t) { return x.ts < t; });
            ks.b.erase(ks.b.begin(), pb);
        }
    }

    bool on_punct(int64_t wm, EmitCtx& out, RuntimeCtx&) override {
        emit_wm = cur_wm;
        cur_wm = std::max(cur_wm, wm);
        purge();
        flush_pairs(out);
        ob.flush(out, cur_wm);
        return false;
    }

    void on_eos(EmitCtx& out, RuntimeCtx&) override {
        emit_wm = cur_wm;
        flush_pairs(out);
        ob.flush(out, cur_wm);
    }
};

// ============== non-incremental (user-function) keyed windows ==============
// Archive-based path: per-key SoA archive with sorted insert; the user
// function sees a contiguous WinRows view per fired window (reference
// window_replica non-incremental + wf/iterable.hpp).
struct PyWindowLogic : OpLogic {
    WinType wt;
    int64_t win, slide, lateness;
    WindowFn fn;

    struct KeyState {
        std::vector<int64_t> ts;
        std::vector<std::vector<char>> cols;  // SoA archive per payload col
        int64_t purged = 0;   // rows removed from the front (CB index base)
        int64_t idx = 0;      // CB arrival count
        int64_t next_gwid = 0;
        int64_t max_ts = INT64_MIN;
        bool tb_init = false;
    };
    std::unordered_map<uint64_t, KeyState> keys;
    Schema in_schema;
    bool have_schema = false;
    OutBuf ob;
    int64_t cur_wm = 0;
    // wm stamped on INTERMEDIATE output batches of one processing round:
    // the pre-round watermark.  Stamping the updated cur_wm on a batch
    // whose successors (same round, OutBuf overflow) still carry ts <= wm
    // breaks the watermark contract downstream (found by the 300-config
    // fuzz: paned TB windows fired before their late panes arrived).
    int64_t emit_wm = 0;

    PyWindowLogic(WinType wt_, int64_t w, int64_t s, int64_t lat, WindowFn f)
        : wt(wt_), win(w), slide(s), lateness(lat), fn(std::move(f)) {}

    void append(KeyState& ks, Batch* b, int64_t i) {
        // sorted insert by ts (rows usually arrive in order: append fast path)
        size_t np = b->schema.payload.size();
        if (ks.cols.size() != np) ks.cols.resize(np);
        int64_t pos = ks.ts.size();
        while (pos > 0 && ks.ts[pos - 1] > b->ts[i]) pos--;
        ks.ts.insert(ks.ts.begin() + pos, b->ts[i]);
        ks.max_ts = std::max(ks.max_ts, b->ts[i]);
        for (size_t c = 0; c < np; ++c) {
            size_t es = dsize(b->schema.payload[c]);
            const char* src = (const char*)b->cols[c] + i * es;
            ks.cols[c].insert(ks.cols[c].begin() + pos * es, src, src + es);
        }
    }

    void fire(uint64_t key, KeyState& ks, int64_t lo, int64_t hi, int64_t gwid,
              int64_t res_ts, EmitCtx& out) {
        // rows [lo, hi) relative to the archive front
        WinRows wr;
        wr.n = hi - lo;
        wr.ts = ks.ts.data() + lo;
        wr.key = key;
        wr.gwid = gwid;
        wr.schema = &in_schema;
        for (size_t c = 0; c < ks.cols.size(); ++c)
            wr.cols.push_back(ks.cols[c].data() + lo * dsize(in_schema.payload[c]));
        double r = fn(wr);
        int64_t i = ob.slot(out, emit_wm);
        Batch* o = ob.b;
        o->ts[i] = res_ts;
        o->key[i] = key;
        ValU v;
        v.f = r;
        v.i = (int64_t)llround(r);
        write_val(o, 0, i, v, o->schema.payload[0] == DType::I64);
    }

    void purge_front(KeyState& ks, int64_t n) {
        if (n <= 0) return;
        n = std::min<int64_t>(n, ks.ts.size());
        ks.ts.erase(ks.ts.begin(), ks.ts.begin() + n);
        for (size_t c = 0; c < ks.cols.size(); ++c) {
            size_t es = dsize(in_schema.payload[c]);
            ks.cols[c].erase(ks.cols[c].begin(), ks.cols[c].begin() + n * es);
        }
        ks.purged += n;
    }

    void process(Batch* b, EmitCtx& out, RuntimeCtx& ctx) override {
        emit_wm = cur_wm;
        if (!have_schema) {
            in_schema = b->schema;
            have_schema = true;
        }
        for (int64_t i = 0; i < b->count; ++i) {
            auto& ks = keys[b->key[i]];
            append(ks, b, i);
            ks.idx++;
            if (wt == WinType::CB) {
                // fire window gwid when its last tuple (index start+win-1) is in
                int64_t g = ks.next_gwid;
                while (g * slide + win <= ks.idx) {
                    int64_t lo = g * slide - ks.purged;
                    fire(b->key[i], ks, lo, lo + win, g, b->ts[i], out);
                    ks.next_gwid = ++g;
                    purge_front(ks, g * slide - ks.purged);
                }
            }
        }
        cur_wm = std::max(cur_wm, ctx.current_wm);
        release(b);
        if (wt == WinType::TB) fire_tb(out, false);
        ob.flush(out, cur_wm);
    }

    void fire_tb(EmitCtx& out, bool eos) {
        for (auto& [key, ks] : keys) {
            if (ks.ts.empty() && !eos) continue;
            if (!ks.tb_init) {
                if (ks.ts.empty()) continue;
                int64_t t0 = ks.ts.front();
                int64_t w0 = t0 - win + 1;
                ks.next_gwid = w0 <= 0 ? 0 : (w0 + slide - 1) / slide;
                ks.tb_init = true;
            }
            for (;;) {
                int64_t start = ks.next_gwid * slide, end = start + win;
                if (start > ks.max_ts) break;  // never open past the key's data
                // at EOS every remaining in-range window fires, INCLUDING
                // empty ones and windows past the front tuple (firing only
                // windows containing the front lost trailing windows —
                // found by the python-fn TB fuzz)
                bool ready = eos || end + lateness <= cur_wm;
                if (!ready) break;
                auto lo = std::lower_bound(ks.ts.begin(), ks.ts.end(), start) - ks.ts.begin();
                auto hi = std::lower_bound(ks.ts.begin(), ks.ts.end(), end) - ks.ts.begin();
                fire(key, ks, lo, hi, ks.next_gwid, end - 1, out);
                ks.next_gwid++;
                // purge rows older than the next window's start
                auto nxt = std::lower_bound(ks.ts.begin(), ks.ts.end(),
                                            ks.next_gwid * slide) - ks.ts.begin();
                purge_front(ks, nxt);
            }
        }
    }

    bool on_punct(int64_t wm, EmitCtx& out, RuntimeCtx&) override {
        emit_wm = cur_wm;
        cur_wm = std::max(cur_wm, wm);
        if (wt == WinType::TB) fire_tb(out, false);
        ob.flush(out, cur_wm);
        return false;
    }

    void on_eos(EmitCtx& out, RuntimeCtx&) override {
        emit_wm = cur_wm;
        if (wt == WinType::TB) {
            fire_tb(out, true);
        } else {
            for (auto& [key, ks] : keys) {
                // flush every open (partial) window, one per slide
                for (int64_t g = ks.next_gwid; g * slide < ks.idx; ++g) {
                    int64_t lo = std::max<int64_t>(g * slide - ks.purged, 0);
                    if ((int64_t)ks.ts.size() > lo)
                        fire(key, ks, lo, ks.ts.size(), g, ks.ts.back(), out);
                }
            }
        }
        ob.flush(out, cur_wm);
    }
};

}  // namespace

std::shared_ptr<OpLogic> make_window_logic(const std::string& kind,
                                           const std::vector<double>& fp,
                                           const std::vector<int64_t>& ip,
                                           Engine* eng, int op_id,
                                           WindowFn userfn, JoinFn joinfn) {
    auto geti = [&](size_t i, int64_t dflt = 0) { return i < ip.size() ? ip[i] : dflt; };
    if (kind == "interval_join") {
        auto j = std::make_shared<IntervalJoinLogic>((int)geti(0), geti(1), geti(2),
                                                     (int)geti(3), (int)geti(4, geti(3)), eng);
        j->pairfn = joinfn;
        return j;
    }
    WinType wt = (WinType)geti(0);
    // defense in depth: win=0/slide=0 would loop forever opening windows
    if (kind != "win_mr_reduce" && (geti(1) < 1 || geti(2) < 1))
        throw std::runtime_error(kind + ": window length and slide must be >= 1");
    if (kind == "win_mr_reduce") {
        AggCfg a{(int)geti(1), geti(2, 1) != 0};
        return std::make_shared<MrReduceLogic>(geti(0, 1), a);
    }
    if (kind == "win_wlq") {
        AggCfg a{(int)geti(4), geti(7, 1) != 0};
        if (geti(6) == 0)  // keyby-partitioned: ordered per-key fast path
            return std::make_shared<WlqKeyedLogic>(wt, geti(1), geti(2), geti(3),
                                                   a, geti(5, 1));
        return std::make_shared<WlqLogic>(wt, geti(1), geti(2), geti(3), a,
                                          geti(5, 1), (int)geti(6));
    }
    if (userfn && kind == "win_keyed")
        return std::make_shared<PyWindowLogic>(wt, geti(1), geti(2), geti(3),
                                               std::move(userfn));
    if (userfn)  // never silently drop a user window function
        throw std::runtime_error("python window function not supported for " +
                                 kind + " (lowering should map it to win_keyed)");
    AggCfg a{(int)geti(4), geti(7, 1) != 0};
    int col = (int)geti(5);
    if (kind == "win_ffat")
        return std::make_shared<FfatCpuLogic>(wt, geti(1), geti(2), geti(3), a, col);
    if (kind == "win_keyed")
        return std::make_shared<WindowCore>(wt, geti(1), geti(2), geti(3), a, col,
                                            0, false, false, false, eng);
    if (kind == "win_parallel")
        return std::make_shared<WindowCore>(wt, geti(1), geti(2), geti(3), a, col,
                                            (int)geti(6, 1), false, false, false, eng);
    if (kind == "win_plq") {
        auto w = std::make_shared<WindowCore>(wt, geti(1), geti(2), geti(3), a, col,
                                              (int)geti(6, 1), true, true, false, eng);
        // TB tumbling panes: materialize only touched panes (the WLQ treats
        // missing panes as the combine identity); CB panes are dense per key
        if (wt == WinType::TB) w->sparse = true;
        return w;
    }
    if (kind == "win_mr_map")
        return std::make_shared<WindowCore>(wt, geti(1), geti(2), geti(3), a, col,
                                            0, true, false, true, eng);
    throw std::runtime_error("unknown window logic kind: " + kind);
}

}  // namespace wfa
