// Native engine stress harness — runs representative graphs directly
// against the C++ engine (no Python), designed to run under
// ThreadSanitizer: `python build_ext.py --tsan && ./build/engine_stress`.
//
// The reference has no race detection at all (SURVEY.md §5.2 — ownership
// discipline only); this harness puts the SPSC queues, emitters,
// collectors, window logics and recycling pools under TSAN across every
// routing mode and execution mode.
#include <cassert>
#include <cstdio>
#include <cstdlib>

#include "../engine/engine.hpp"

using namespace wfa;

static int failures = 0;
#define CHECK(cond, ...)                                                       \
    do {                                                                       \
        if (!(cond)) {                                                         \
            fprintf(stderr, "CHECK failed (%s:%d): " #cond "\n", __FILE__,     \
                    __LINE__);                                                 \
            failures++;                                                        \
        }                                                                      \
    } while (0)

static void add_native(Engine& e, const char* name, int par, const char* kind,
                       const char* spec, std::vector<int64_t> ip,
                       std::vector<DType> schema, int64_t ob) {
    OpSpec op;
    op.id = (int)e.ops.size();
    op.name = name;
    op.parallelism = par;
    StageSpec st;
    st.out_schema.payload = schema;
    st.out_batch = ob;
    Engine* ep = &e;
    int id = op.id;
    std::string k = kind, s = spec;
    std::vector<double> fp;
    if (k == "sink") {
        e.sink_acc_i64[id].store(0);
        e.sink_tuples[id].store(0);
    }
    if (k.rfind("win_", 0) == 0 || k == "interval_join")
        st.factory = [k, fp, ip, ep, id] {
            return make_window_logic(k, fp, ip, ep, id, nullptr);
        };
    else
        st.factory = [k, s, fp, ip, ep, id] {
            return make_native_logic(k, s, fp, ip, ep, id);
        };
    op.stages.push_back(std::move(st));
    e.ops.push_back(std::move(op));
}

// Source -> Map -> Filter(keyby) -> Sink across modes and random degrees
static void stress_linear(ExecMode mode, int seed) {
    srand(seed);
    int64_t n = 40000;
    int sdeg = 1 + rand() % 3, mdeg = 1 + rand() % 4, fdeg = 1 + rand() % 4;
    Engine e;
    e.mode = mode;
    add_native(e, "src", sdeg, "source", "seq", {n, 13, 256, 0}, {DType::I64}, 256);
    add_native(e, "map", mdeg, "map", "affine_i64", {0, 3, 1}, {DType::I64}, 256);
    add_native(e, "fil", fdeg, "filter", "mod_i64", {0, 5, 0, 0}, {DType::I64}, 256);
    add_native(e, "snk", 1, "sink", "sum_i64", {0}, {}, 256);
    CollectorKind ck = mode == ExecMode::DETERMINISTIC ? CollectorKind::ORDERING
                     : mode == ExecMode::PROBABILISTIC ? CollectorKind::KSLACK
                                                       : CollectorKind::WATERMARK;
    e.edges.push_back({0, 1, Routing::FORWARD, ck, -1});
    e.edges.push_back({1, 2, Routing::KEYBY, ck, -1});
    e.edges.push_back({2, 3, Routing::FORWARD, ck, -1});
    e.run();
    int64_t exp1 = 0;
    for (int64_t v = 1; v <= n; ++v) {
        int64_t x = 3 * v + 1;
        if (x % 5 != 0) exp1 += x;
    }
    if (mode != ExecMode::PROBABILISTIC)
        CHECK(e.sink_acc_i64[3].load() == exp1 * sdeg);
}

// Source -> keyed TB windows -> Sink (fires through puncts + shuffles)
static void stress_windows(int seed) {
    srand(seed);
    int64_t n = 30000;
    int wdeg = 1 + rand() % 3;
    Engine e;
    add_native(e, "src", 1, "source", "seq", {n, 7, 128, 0}, {DType::I64}, 128);
    add_native(e, "win", wdeg, "win_keyed", "", {1, 100, 25, 0, 0, 0, 0, 1},
               {DType::I64}, 128);
    add_native(e, "snk", 1, "sink", "count", {}, {}, 128);
    e.edges.push_back({0, 1, Routing::KEYBY, CollectorKind::WATERMARK, -1});
    e.edges.push_back({1, 2, Routing::FORWARD, CollectorKind::WATERMARK, -1});
    e.run();
    CHECK(e.sink_tuples[2].load() > 0);
}

// Broadcast fan-out + multi-sink
static void stress_broadcast(int seed) {
    srand(seed);
    int64_t n = 20000;
    int deg = 2 + rand() % 3;
    Engine e;
    add_native(e, "src", 1, "source", "seq", {n, 3, 128, 0}, {DType::I64}, 128);
    add_native(e, "snk", deg, "sink", "sum_i64", {0}, {}, 128);
    e.edges.push_back({0, 1, Routing::BROADCAST, CollectorKind::WATERMARK, -1});
    e.run();
    int64_t exp = 0;
    for (int64_t v = 1; v <= n; ++v) exp += v;
    CHECK(e.sink_acc_i64[1].load() == exp * deg);
}

int main(int argc, char** argv) {
    int rounds = argc > 1 ? atoi(argv[1]) : 3;
    for (int r = 0; r < rounds; ++r) {
        stress_linear(ExecMode::DEFAULT, 100 + r);
        stress_linear(ExecMode::DETERMINISTIC, 200 + r);
        stress_linear(ExecMode::PROBABILISTIC, 300 + r);
        stress_windows(400 + r);
        stress_broadcast(500 + r);
        printf("round %d ok\n", r);
    }
    if (failures) {
        printf("FAILURES: %d\n", failures);
        return 1;
    }
    printf("engine_stress OK\n");
    return 0;
}
