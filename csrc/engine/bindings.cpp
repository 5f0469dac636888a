// pybind11 bindings: windflow_amd._core
//
// Exposes the native engine (Engine/OpSpec/EdgeSpec), the native logic
// catalog, and per-batch Python-callback logic.  Python callbacks receive
// zero-copy numpy views of the batch columns (valid only during the call).
#include <pybind11/functional.h>
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "engine.hpp"

namespace py = pybind11;
using namespace wfa;

// ---- numpy view helpers ----
static py::array col_view(DType dt, void* data, int64_t n) {
    switch (dt) {
        case DType::I64: return py::array_t<int64_t>({n}, {8}, (int64_t*)data, py::none());
        case DType::F64: return py::array_t<double>({n}, {8}, (double*)data, py::none());
        case DType::F32: return py::array_t<float>({n}, {4}, (float*)data, py::none());
        case DType::U64: return py::array_t<uint64_t>({n}, {8}, (uint64_t*)data, py::none());
        case DType::I32: return py::array_t<int32_t>({n}, {4}, (int32_t*)data, py::none());
        case DType::U16: return py::array_t<uint16_t>({n}, {2}, (uint16_t*)data, py::none());
        case DType::U8: return py::array_t<uint8_t>({n}, {1}, (uint8_t*)data, py::none());
    }
    throw std::runtime_error("bad dtype");
}

static py::dict batch_views(Batch* b, RuntimeCtx* ctx = nullptr) {
    py::dict d;
    d["ts"] = col_view(DType::I64, b->ts, b->count);
    d["key"] = col_view(DType::U64, b->key, b->count);
    for (size_t c = 0; c < b->schema.payload.size(); ++c)
        d[py::str("c" + std::to_string(c))] = col_view(b->schema.payload[c], b->cols[c], b->count);
    d["watermark"] = ctx ? ctx->current_wm : b->watermark;
    if (ctx) {  // RuntimeContext parity (reference wf/context.hpp:53)
        d["replica"] = ctx->replica;
        d["parallelism"] = ctx->parallelism;
        d["stream_tag"] = ctx->current_tag;
    }
    return d;
}

// copy a python dict-of-arrays into fresh batches (splitting on capacity)
// Rows grabbed from a python dict UNDER the GIL, emitted WITHOUT it.
// Holding the GIL across a blocking queue push (or pool get) deadlocks a
// pipeline of python operators once queues fill: the downstream python
// stage needs the GIL to drain the very queue the upstream is pushing to
// (found by the WFA_QUEUE_CAP=2 backpressure campaign).
struct PyRows {
    std::vector<int64_t> ts;
    std::vector<uint64_t> key;
    std::vector<std::vector<char>> cols;
    int64_t n = 0;
    bool has_ts = false, has_key = false;
};

static PyRows grab_pydict(py::dict out_d) {  // caller holds the GIL
    PyRows r;
    if (out_d.size() == 0) return r;
    py::array ts = out_d.contains("ts") ? out_d["ts"].cast<py::array>() : py::array();
    py::array key = out_d.contains("key") ? out_d["key"].cast<py::array>() : py::array();
    std::vector<py::array> cols;
    for (size_t c = 0;; ++c) {
        std::string nm = "c" + std::to_string(c);
        if (!out_d.contains(nm.c_str())) break;
        cols.push_back(out_d[nm.c_str()].cast<py::array>());
    }
    if (cols.size()) r.n = cols[0].shape(0);
    else if (ts.ndim()) r.n = ts.shape(0);
    if (ts.ndim() == 1) {
        auto a = ts.cast<py::array_t<int64_t>>();
        r.ts.assign(a.data(), a.data() + r.n);
        r.has_ts = true;
    }
    if (key.ndim() == 1) {
        auto a = key.cast<py::array_t<uint64_t>>();
        r.key.assign(a.data(), a.data() + r.n);
        r.has_key = true;
    }
    for (auto& c : cols) {
        py::buffer_info bi = c.request();
        r.cols.emplace_back((char*)bi.ptr,
                            (char*)bi.ptr + (size_t)r.n * bi.itemsize);
    }
    return r;
}

static void emit_rows(const PyRows& r, EmitCtx& out, int64_t wm) {  // no GIL
    int64_t done = 0;
    while (done < r.n) {
        Batch* o = out.new_batch();
        int64_t take = std::min<int64_t>(o->capacity, r.n - done);
        if (r.has_ts)
            memcpy(o->ts, r.ts.data() + done, take * 8);
        else
            for (int64_t i = 0; i < take; ++i) o->ts[i] = done + i;
        if (r.has_key)
            memcpy(o->key, r.key.data() + done, take * 8);
        else
            memset(o->key, 0, take * 8);
        for (size_t c = 0; c < r.cols.size() && c < o->schema.payload.size(); ++c) {
            size_t es = dsize(o->schema.payload[c]);
            size_t have = r.cols[c].size() / (r.n ? r.n : 1);
            if (have != es)
                throw std::runtime_error("dtype mismatch on c" + std::to_string(c));
            memcpy(o->cols[c], r.cols[c].data() + done * es, take * es);
        }
        o->count = take;
        o->watermark = wm;
        out.emit(o);
        done += take;
    }
}

// ---- Python-callback logic ----
struct PyMapLogic : OpLogic {  // in-place mutate
    py::function fn;
    explicit PyMapLogic(py::function f) : fn(std::move(f)) {}
    void process(Batch* b, EmitCtx& out, RuntimeCtx& rctx) override {
        if (b->refcnt.load(std::memory_order_acquire) > 1) {
            Batch* c = clone(b, *b->pool);
            c->watermark = rctx.current_wm;  // folded (shared original is read-only)
            release(b);
            b = c;
        }
        {
            py::gil_scoped_acquire gil;
            fn(batch_views(b, &rctx));
        }
        out.emit(b);
    }
};

struct PyTransformLogic : OpLogic {  // returns new column dict (map/flatmap)
    py::function fn;
    int64_t last_wm = 0;
    explicit PyTransformLogic(py::function f) : fn(std::move(f)) {}
    void process(Batch* b, EmitCtx& out, RuntimeCtx& rctx) override {
        int64_t wm = rctx.current_wm;
        last_wm = wm;
        PyRows rows;
        {
            py::gil_scoped_acquire gil;
            py::object r = fn(batch_views(b, &rctx));
            release(b);
            if (!r.is_none()) rows = grab_pydict(r.cast<py::dict>());
        }
        emit_rows(rows, out, wm);
    }
    void on_eos(EmitCtx& out, RuntimeCtx& ctx) override {
        // stateful python transforms (e.g. store-backed windows) flush their
        // pending results at stream end via fn.on_eos(replica) -> dict; the
        // replica id is passed because the python callable is SHARED across
        // replicas and must only flush the keys this replica owns
        PyRows rows;
        {
            py::gil_scoped_acquire gil;
            if (!py::hasattr(fn, "on_eos")) return;
            py::object r = fn.attr("on_eos")(ctx.replica);
            if (!r.is_none()) rows = grab_pydict(r.cast<py::dict>());
        }
        emit_rows(rows, out, last_wm);
    }
};

struct PyFilterLogic : OpLogic {  // returns bool mask
    py::function fn;
    explicit PyFilterLogic(py::function f) : fn(std::move(f)) {}
    void process(Batch* b, EmitCtx& out, RuntimeCtx& rctx) override {
        if (b->refcnt.load(std::memory_order_acquire) > 1) {
            Batch* c = clone(b, *b->pool);
            c->watermark = rctx.current_wm;
            release(b);
            b = c;
        }
        std::vector<char> keep;
        {
            py::gil_scoped_acquire gil;
            py::object r = fn(batch_views(b, &rctx));
            auto mask = r.cast<py::array_t<bool>>();
            keep.assign(mask.data(), mask.data() + mask.shape(0));
        }
        const size_t np_ = b->schema.payload.size();
        int64_t w = 0;
        for (int64_t i = 0; i < b->count; ++i) {
            if (!keep[i]) continue;
            if (w != i) {
                b->ts[w] = b->ts[i];
                b->key[w] = b->key[i];
                for (size_t cc = 0; cc < np_; ++cc) {
                    size_t es = dsize(b->schema.payload[cc]);
                    memcpy((char*)b->cols[cc] + w * es, (char*)b->cols[cc] + i * es, es);
                }
            }
            ++w;
        }
        b->count = w;
        if (w)
            out.emit(b);
        else {
            int64_t wm = rctx.current_wm;
            release(b);
            for (auto* e : out.emitters) e->punct(wm);
        }
    }
};

struct PySinkLogic : OpLogic {
    py::function fn;
    explicit PySinkLogic(py::function f) : fn(std::move(f)) {}
    void process(Batch* b, EmitCtx&, RuntimeCtx& rctx) override {
        {
            py::gil_scoped_acquire gil;
            fn(batch_views(b, &rctx));
        }
        release(b);
    }
    void on_eos(EmitCtx&, RuntimeCtx&) override {
        py::gil_scoped_acquire gil;
        if (py::hasattr(fn, "on_eos")) fn.attr("on_eos")();
    }
};

struct PySplitLogic : OpLogic {  // fn(cols) -> branch ids OR per-branch masks
    py::function fn;
    explicit PySplitLogic(py::function f) : fn(std::move(f)) {}
    void process(Batch* b, EmitCtx& out, RuntimeCtx& rctx) override {
        const size_t nb = out.n_branches();
        const size_t np_ = b->schema.payload.size();
        // per-branch keep masks; a 1-D int array routes each row to ONE
        // branch, a sequence of bool masks routes rows to ANY subset
        // (reference splitting_emitter.hpp: integral_t OR vector<integral_t>)
        std::vector<std::vector<char>> keep(nb, std::vector<char>(b->count, 0));
        {
            py::gil_scoped_acquire gil;
            py::object r = fn(batch_views(b, &rctx));
            if (py::isinstance<py::array>(r)) {
                auto ids = r.cast<py::array_t<int32_t>>();
                for (int64_t i = 0; i < b->count; ++i) {
                    int32_t d = ids.at(i);
                    if (d >= 0 && (size_t)d < nb) keep[d][i] = 1;
                }
            } else {
                size_t br = 0;
                for (auto m : r.cast<py::sequence>()) {
                    if (br >= nb) break;
                    auto mask = m.cast<py::array_t<bool>>();
                    for (int64_t i = 0; i < b->count; ++i)
                        if (mask.at(i)) keep[br][i] = 1;
                    ++br;
                }
            }
        }
        int64_t wm = rctx.current_wm;
        for (size_t br = 0; br < nb; ++br) {
            int64_t cnt = 0;
            for (int64_t i = 0; i < b->count; ++i) cnt += keep[br][i];
            if (!cnt) {
                out.emitters[br]->punct(wm);  // keep idle branches alive
                continue;
            }
            Batch* o = out.new_batch();
            int64_t w = 0;
            for (int64_t i = 0; i < b->count; ++i) {
                if (!keep[br][i]) continue;
                if (w == o->capacity) {
                    o->count = w;
                    o->watermark = wm;
                    out.emit_to(br, o);
                    o = out.new_batch();
                    w = 0;
                }
                o->ts[w] = b->ts[i];
                o->key[w] = b->key[i];
                for (size_t cc = 0; cc < np_; ++cc) {
                    size_t es = dsize(b->schema.payload[cc]);
                    memcpy((char*)o->cols[cc] + w * es, (char*)b->cols[cc] + i * es,
                           es);
                }
                ++w;
            }
            o->count = w;
            o->watermark = wm;
            out.emit_to(br, o);
        }
        release(b);
    }
};

struct PySourceLogic : OpLogic {
    py::function fn;  // fn(replica, parallelism) -> dict|None
    explicit PySourceLogic(py::function f) : fn(std::move(f)) {}
    bool is_source() const override { return true; }
    bool source_step(EmitCtx& out, RuntimeCtx& ctx) override {
        PyRows rows;
        int64_t wm = 0;
        {
            py::gil_scoped_acquire gil;
            py::object r = fn(ctx.replica, ctx.parallelism);
            if (r.is_none()) return false;
            py::dict d = r.cast<py::dict>();
            wm = d.contains("watermark") ? d["watermark"].cast<int64_t>() : 0;
            rows = grab_pydict(d);
            if (ctx.engine && ctx.engine->time_policy == TimePolicy::INGRESS_TIME) {
                // ingress time: stamp arrival clock, ignore user ts/wm
                int64_t nowus = now_us();
                rows.ts.assign((size_t)rows.n, nowus);
                rows.has_ts = true;
                wm = nowus;
            }
        }
        emit_rows(rows, out, wm);
        return true;
    }
};

// key-extractor map: computes the key column from a payload column
struct KeyByColLogic : OpLogic {
    int col;
    explicit KeyByColLogic(int c) : col(c) {}
    void process(Batch* b, EmitCtx& out, RuntimeCtx&) override {
        if (b->refcnt.load(std::memory_order_acquire) > 1) {
            Batch* c2 = clone(b, *b->pool);
            release(b);
            b = c2;
        }
        switch (b->schema.payload[col]) {
            case DType::I64:
                for (int64_t i = 0; i < b->count; ++i) b->key[i] = (uint64_t)b->col<int64_t>(col)[i];
                break;
            case DType::U64:
                for (int64_t i = 0; i < b->count; ++i) b->key[i] = b->col<uint64_t>(col)[i];
                break;
            case DType::I32:
                for (int64_t i = 0; i < b->count; ++i) b->key[i] = (uint64_t)b->col<int32_t>(col)[i];
                break;
            default:
                throw std::runtime_error("keyby col must be integer");
        }
        out.emit(b);
    }
};

static StageSpec make_stage(Engine& e, int id, const std::string& kind, const std::string& spec,
                            std::vector<double> fp, std::vector<int64_t> ip,
                            std::vector<int> out_schema, int64_t out_batch, py::object pyfn,
                            int device = -1) {
    StageSpec st;
    for (int d : out_schema) st.out_schema.payload.push_back((DType)d);
    st.out_batch = out_batch;
    if (kind == "sink") {
        e.sink_acc_i64[id].store(0);
        e.sink_tuples[id].store(0);
    }
    if (kind.rfind("p_", 0) == 0) {
        Engine* ep = &e;
        st.factory = [kind, spec, fp, ip, ep, id] {
            return make_persist_logic(kind, spec, fp, ip, ep, id);
        };
        return st;
    }
    if (kind.rfind("win_", 0) == 0 || kind == "interval_join") {
        Engine* ep = &e;
        WindowFn wf = nullptr;
        JoinFn jf = nullptr;
        if (!pyfn.is_none() && kind == "interval_join") {
            // vectorized user predicate/result (reference
            // interval_join.hpp:279-307): fn(pairs_dict) -> None (drop all)
            // | values array (keep all) | (mask, values) tuple
            auto f = pyfn.cast<py::function>();
            jf = [f](const JoinPairs& p, uint8_t* keep, double* out) {
                py::gil_scoped_acquire gil;
                py::dict d;
                d["key"] = py::array_t<uint64_t>({p.n}, {(int64_t)8}, p.key,
                                                 py::none());
                d["ts_a"] = py::array_t<int64_t>({p.n}, {(int64_t)8}, p.ts_a,
                                                 py::none());
                d["ts_b"] = py::array_t<int64_t>({p.n}, {(int64_t)8}, p.ts_b,
                                                 py::none());
                d["a"] = py::array_t<double>({p.n}, {(int64_t)8}, p.va,
                                             py::none());
                d["b"] = py::array_t<double>({p.n}, {(int64_t)8}, p.vb,
                                             py::none());
                py::object r = f(d);
                if (r.is_none()) return;  // keep[] stays all-zero
                py::object mask = py::none(), vals;
                if (py::isinstance<py::tuple>(r)) {
                    auto t = r.cast<py::tuple>();
                    mask = t[0];
                    vals = t[1];
                } else {
                    vals = r;
                }
                auto va = py::array_t<double, py::array::c_style |
                                                  py::array::forcecast>(vals);
                if (va.shape(0) != p.n)
                    throw std::runtime_error(
                        "join function must return one value per pair");
                memcpy(out, va.data(), 8 * p.n);
                if (mask.is_none()) {
                    memset(keep, 1, p.n);
                } else {
                    auto ma = py::array_t<bool, py::array::c_style |
                                                    py::array::forcecast>(mask);
                    if (ma.shape(0) != p.n)
                        throw std::runtime_error(
                            "join mask must have one entry per pair");
                    for (int64_t i = 0; i < p.n; ++i)
                        keep[i] = ma.data()[i] ? 1 : 0;
                }
            };
        } else if (!pyfn.is_none()) {
            auto f = pyfn.cast<py::function>();
            wf = [f](const WinRows& wr) -> double {
                py::gil_scoped_acquire gil;
                py::dict d;
                d["ts"] = py::array_t<int64_t>({wr.n}, {(int64_t)8}, wr.ts, py::none());
                d["key"] = wr.key;
                d["gwid"] = wr.gwid;
                for (size_t c = 0; c < wr.schema->payload.size(); ++c)
                    d[py::str("c" + std::to_string(c))] =
                        col_view(wr.schema->payload[c], (void*)wr.cols[c], wr.n);
                return f(d).cast<double>();
            };
        }
        st.factory = [kind, fp, ip, ep, id, wf, jf] {
            return make_window_logic(kind, fp, ip, ep, id, wf, jf);
        };
        return st;
    }
    if (kind.rfind("gpu_", 0) == 0) {
        Engine* ep = &e;
        Schema os = st.out_schema;
        st.factory = [kind, spec, fp, ip, ep, id, device, os, out_batch] {
            return make_gpu_logic(kind, spec, fp, ip, ep, id, device, os, out_batch);
        };
        return st;
    }
    if (!pyfn.is_none()) {
        auto fn = pyfn.cast<py::function>();
        if (kind == "map")
            st.factory = [fn] { return std::make_shared<PyMapLogic>(fn); };
        else if (kind == "split")
            st.factory = [fn] { return std::make_shared<PySplitLogic>(fn); };
        else if (kind == "transform" || kind == "flatmap")
            st.factory = [fn] { return std::make_shared<PyTransformLogic>(fn); };
        else if (kind == "filter")
            st.factory = [fn] { return std::make_shared<PyFilterLogic>(fn); };
        else if (kind == "sink")
            st.factory = [fn] { return std::make_shared<PySinkLogic>(fn); };
        else if (kind == "source")
            st.factory = [fn] { return std::make_shared<PySourceLogic>(fn); };
        else
            throw std::runtime_error("no python form for op kind " + kind);
    } else if (kind == "keyby_col") {
        int col = (int)ip.at(0);
        st.factory = [col] { return std::make_shared<KeyByColLogic>(col); };
    } else {
        Engine* ep = &e;
        st.factory = [kind, spec, fp, ip, ep, id] {
            return make_native_logic(kind, spec, fp, ip, ep, id);
        };
    }
    return st;
}

PYBIND11_MODULE(_core, m) {
    m.doc() = "windflow_amd native engine (MI355X-native WindFlow re-design)";

    py::enum_<ExecMode>(m, "ExecMode")
        .value("DEFAULT", ExecMode::DEFAULT)
        .value("DETERMINISTIC", ExecMode::DETERMINISTIC)
        .value("PROBABILISTIC", ExecMode::PROBABILISTIC);
    py::enum_<TimePolicy>(m, "TimePolicy")
        .value("INGRESS_TIME", TimePolicy::INGRESS_TIME)
        .value("EVENT_TIME", TimePolicy::EVENT_TIME);
    py::enum_<Routing>(m, "Routing")
        .value("FORWARD", Routing::FORWARD)
        .value("KEYBY", Routing::KEYBY)
        .value("BROADCAST", Routing::BROADCAST)
        .value("REBALANCING", Routing::REBALANCING);
    py::enum_<CollectorKind>(m, "CollectorKind")
        .value("WATERMARK", CollectorKind::WATERMARK)
        .value("ORDERING", CollectorKind::ORDERING)
        .value("KSLACK", CollectorKind::KSLACK)
        .value("JOIN", CollectorKind::JOIN);
    py::enum_<DType>(m, "DType")
        .value("I64", DType::I64)
        .value("F64", DType::F64)
        .value("F32", DType::F32)
        .value("U64", DType::U64)
        .value("I32", DType::I32)
        .value("U16", DType::U16)
        .value("U8", DType::U8);

    m.def("rccl_unique_id",
          [] { return py::bytes(wfa_rccl_unique_id()); },
          "ncclUniqueId bytes for Engine.set_dist (call on rank 0, broadcast)");

    py::class_<Engine>(m, "Engine")
        .def(py::init<>())
        .def_readwrite("mode", &Engine::mode)
        .def_readwrite("time_policy", &Engine::time_policy)
        .def_readwrite("queue_capacity", &Engine::queue_capacity)
        .def_readwrite("pin_threads", &Engine::pin_threads)
        .def("set_dist",
             [](Engine& e, int rank, int world, py::bytes id) {
                 e.dist_rank = rank;
                 e.dist_world = world;
                 e.rccl_id = std::string(id);
             },
             py::arg("rank"), py::arg("world"), py::arg("rccl_id"))
        .def("add_op",
             [](Engine& e, const std::string& name, int par, const std::string& kind,
                const std::string& spec, std::vector<double> fp, std::vector<int64_t> ip,
                std::vector<int> out_schema, int64_t out_batch, py::object pyfn,
                int device) {
                 OpSpec op;
                 op.id = (int)e.ops.size();
                 op.name = name;
                 op.parallelism = par;
                 op.device = device;
                 int id = op.id;
                 op.stages.push_back(make_stage(e, id, kind, spec, fp, ip, out_schema,
                                                out_batch, pyfn, device));
                 e.ops.push_back(std::move(op));
                 return id;
             },
             py::arg("name"), py::arg("parallelism"), py::arg("kind"), py::arg("spec") = "",
             py::arg("fparams") = std::vector<double>{}, py::arg("iparams") = std::vector<int64_t>{},
             py::arg("out_schema") = std::vector<int>{}, py::arg("out_batch") = 1024,
             py::arg("pyfn") = py::none(), py::arg("device") = -1)
        .def("chain_stage",
             [](Engine& e, int op_id, const std::string& kind, const std::string& spec,
                std::vector<double> fp, std::vector<int64_t> ip, std::vector<int> out_schema,
                int64_t out_batch, py::object pyfn) {
                 e.ops.at(op_id).stages.push_back(
                     make_stage(e, op_id, kind, spec, fp, ip, out_schema, out_batch, pyfn,
                                e.ops.at(op_id).device));
             },
             py::arg("op_id"), py::arg("kind"), py::arg("spec") = "",
             py::arg("fparams") = std::vector<double>{}, py::arg("iparams") = std::vector<int64_t>{},
             py::arg("out_schema") = std::vector<int>{}, py::arg("out_batch") = 1024,
             py::arg("pyfn") = py::none())
        .def("add_edge",
             [](Engine& e, int from, int to, Routing r, CollectorKind ck, int tag) {
                 e.edges.push_back(EdgeSpec{from, to, r, ck, tag});
             },
             py::arg("from_op"), py::arg("to_op"), py::arg("routing") = Routing::FORWARD,
             py::arg("collector") = CollectorKind::WATERMARK, py::arg("stream_tag") = -1)
        .def("run",
             [](Engine& e) {
                 e.build();  // copies py::function logic — needs the GIL
                 {
                     py::gil_scoped_release rel;
                     e.start();
                     e.wait();
                 }
             })
        .def("start",
             [](Engine& e) {
                 e.build();
                 e.start();
             })
        .def("start_gated",
             [](Engine& e) {
                 e.build();
                 {
                     py::gil_scoped_release rel;
                     e.start_gated();
                 }
             })
        .def("open_gate_and_wait",
             [](Engine& e) {
                 py::gil_scoped_release rel;
                 e.open_gate();
                 e.wait();
             })
        .def("wait",
             [](Engine& e) {
                 py::gil_scoped_release rel;
                 e.wait();
             })
        .def("abort_now", [](Engine& e) { e.abort.store(true); })
        .def("sink_sum", [](Engine& e, int op) { return e.sink_acc_i64.at(op).load(); })
        .def("sink_sum_f",
             [](Engine& e, int op) {
                 std::lock_guard<std::mutex> g(e.sink_f64_mu);
                 auto it = e.sink_acc_f64.find(op);
                 return it == e.sink_acc_f64.end() ? 0.0 : it->second;
             })
        .def("sink_count", [](Engine& e, int op) { return e.sink_tuples.at(op).load(); })
        .def("dropped", [](Engine& e) { return e.dropped_tuples.load(); })
        .def("sink_latencies",
             [](Engine& e, int op) {
                 auto it = e.sink_latencies.find(op);
                 return it == e.sink_latencies.end() ? std::vector<int64_t>{}
                                                     : it->second;
             })
        .def("stats", [](Engine& e) {
            py::list out;
            for (auto& r : e.replicas) {
                py::dict d;
                d["op"] = r->op_id;
                d["name"] = e.ops[r->op_id].name;
                d["replica"] = r->idx;
                d["inputs"] = r->stats.inputs_received;
                d["tuples_in"] = r->stats.tuples_received;
                d["outputs"] = r->stats.outputs_sent;
                d["tuples_out"] = r->stats.tuples_sent;
                d["svc_us_ewma"] = r->stats.service_time_us_ewma;
                d["start_us"] = r->stats.start_us;
                d["end_us"] = r->stats.end_us;
                d["kernels"] = r->stats.num_kernels;
                d["bytes_h2d"] = r->stats.bytes_h2d;
                d["bytes_d2h"] = r->stats.bytes_d2h;
                out.append(d);
            }
            return out;
        });

    m.def("hash_key", [](uint64_t k) { return KeyByEmitter::mix(k); });

#ifdef WFA_WITH_HIP
    // JIT codegen probe: the generated hiprtc source for a jit fold spec —
    // the CPU test tier cross-compiles it with hipcc (no GPU needed)
    m.def("debug_jit_fold_source",
          [](const std::string& kind, const std::string& spec,
             const std::vector<double>& fp, const std::vector<int64_t>& ip) {
              return debug_jit_fold_source(kind, spec, fp, ip);
          });
    // debug hooks: round-trip device primitives for isolation tests
    m.def("debug_sort_pairs",
          [](py::array_t<uint32_t> keys, int bits) {
              auto r = debug_sort_pairs_host(keys.data(), keys.shape(0), bits);
              py::array_t<uint32_t> ks((py::ssize_t)r.first.size());
              py::array_t<uint32_t> vs((py::ssize_t)r.second.size());
              memcpy(ks.mutable_data(), r.first.data(), 4 * r.first.size());
              memcpy(vs.mutable_data(), r.second.data(), 4 * r.second.size());
              return py::make_tuple(ks, vs);
          });
    m.def("debug_ffat_stage_times",
          [](int64_t n, int64_t n_keys, int64_t win, int64_t slide, int iters,
             int vik) {
              py::dict d;
              for (auto& [k, v] :
                   debug_ffat_stage_times(n, n_keys, win, slide, iters, vik))
                  d[py::str(k)] = v;
              return d;
          },
          py::arg("n"), py::arg("n_keys"), py::arg("win") = 1000,
          py::arg("slide") = 100, py::arg("iters") = 20, py::arg("vik") = 0);
    m.def("debug_tb_stage_times",
          [](int64_t n, int64_t n_keys, int iters, int mono) {
              std::map<std::string, double> m2;
              for (auto& kv : debug_tb_stage_times(n, n_keys, iters, mono))
                  m2[kv.first] = kv.second;
              return m2;
          },
          py::arg("n"), py::arg("n_keys"), py::arg("iters") = 20,
          py::arg("mono") = 1);
    m.def("debug_gram_stage_times",
          [](int64_t n, int64_t n_keys, int64_t win, int iters) {
              std::map<std::string, double> m2;
              for (auto& kv : debug_gram_stage_times(n, n_keys, win, iters))
                  m2[kv.first] = kv.second;
              return m2;
          },
          py::arg("n"), py::arg("n_keys"), py::arg("win") = 32,
          py::arg("iters") = 20);
    m.def("debug_a2a_stage_times",
          [](int64_t n, int64_t n_keys, int iters) {
              py::list out;
              for (auto& [k, v] : debug_a2a_stage_times(n, n_keys, iters))
                  out.append(py::make_tuple(k, v));
              return out;
          },
          py::arg("n"), py::arg("n_keys") = 8192, py::arg("iters") = 20);
    m.def("debug_key_slots",
          [](py::array_t<uint64_t> keys, int64_t max_keys) {
              auto r = debug_key_slots_host(keys.data(), keys.shape(0), max_keys);
              py::array_t<uint32_t> s((py::ssize_t)r.size());
              memcpy(s.mutable_data(), r.data(), 4 * r.size());
              return s;
          });
#endif

    // persistent keyed state handle for Python P_* logic (reference
    // DBHandle<T>: get/modify/put with user serialize/deserialize)
    struct StateStore {
        std::shared_ptr<void> holder;
        void* kv = nullptr;
        void* cache = nullptr;
    };
    py::class_<StateStore>(m, "StateStore")
        .def(py::init([](const std::string& path, int64_t cache_cap, bool fresh) {
                 auto s = new StateStore();
                 s->holder =
                     open_state_store(path, cache_cap, &s->kv, &s->cache, fresh);
                 return s;
             }),
             py::arg("path"), py::arg("cache_capacity") = 1 << 16,
             py::arg("fresh") = true)
        .def("erase",
             [](StateStore& s, uint64_t key) {
                 return state_kv_erase(s.kv, s.cache, key);
             })
        .def("get",
             [](StateStore& s, uint64_t key) -> py::object {
                 std::string* v = state_cache_get(s.cache, key);
                 if (!v) return py::none();
                 return py::bytes(*v);
             })
        .def("put",
             [](StateStore& s, uint64_t key, py::bytes val) {
                 state_cache_put(s.cache, key, std::string(val));
             })
        .def("flush", [](StateStore& s) { state_cache_flush(s.cache); })
        .def("__len__", [](StateStore& s) { return state_kv_size(s.kv); });
}
