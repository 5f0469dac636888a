// Persistent keyed-state tier.
//
// The reference's wf/persistent/ operators (P_Filter/P_Map/P_Reduce/
// P_Keyed_Windows) keep per-key state in RocksDB via
// DBHandle<T> (db_handle.hpp:54) with a get-modify-put per tuple and an
// optional LRU cache (persistent/cache/cache.hpp:46).  RocksDB is not in
// this image; the MI355X-native build ships its own embedded store:
//  - VarKV: append-only value log + in-memory key index (u64 key ->
//    offset/len), crash-simple, compacting when garbage exceeds half the
//    log; values are opaque byte strings (user serialize/deserialize, as
//    in the reference).
//  - LruCache: bounded write-back cache in front of the log, flushed on
//    eviction and at operator EOS (reference cache_lru semantics).
// P_Reduce (keyed running i64 sum) runs entirely native; generic P_Map /
// P_Filter logic reaches the store through the Python binding (StateStore).
#include <fcntl.h>
#include <sys/stat.h>
#include <unistd.h>

#include <cstring>
#include <list>

#include "engine.hpp"

namespace wfa {

struct VarKV {
    std::string path;
    int fd = -1;
    struct Loc {
        uint64_t off;   // offset of the VALUE bytes (header skipped)
        uint32_t len;
    };
    std::unordered_map<uint64_t, Loc> index;
    uint64_t tail = 0;      // append offset
    uint64_t live = 0;      // live bytes
    std::mutex mu;

    // Records are self-describing ([u64 key][u32 len][bytes]) so a log can
    // be REOPENED and its index rebuilt — checkpoint/resume the reference
    // entirely lacks (DBs are destroyed on teardown).
    explicit VarKV(const std::string& p, bool fresh = true) : path(p) {
        open_log(fresh);
    }
    ~VarKV() {
        if (fd >= 0) close(fd);
    }
    void open_log(bool fresh) {
        fd = ::open(path.c_str(), O_RDWR | O_CREAT | (fresh ? O_TRUNC : 0), 0644);
        if (fd < 0) throw std::runtime_error("VarKV: cannot open " + path);
        tail = live = 0;
        index.clear();
        if (!fresh) replay();
    }
    void replay() {
        struct stat st {};
        if (fstat(fd, &st) != 0) throw std::runtime_error("VarKV: fstat failed");
        const uint64_t fsize = (uint64_t)st.st_size;
        char hdr[12];
        uint64_t off = 0;
        for (;;) {
            if (off + 12 > fsize || pread(fd, hdr, 12, (off_t)off) != 12) break;
            uint64_t key;
            uint32_t len;
            memcpy(&key, hdr, 8);
            memcpy(&len, hdr + 8, 4);
            if (len == 0xFFFFFFFFu) {  // tombstone: key erased
                auto it = index.find(key);
                if (it != index.end()) {
                    live -= it->second.len;
                    index.erase(it);
                }
                off += 12;
                continue;
            }
            // torn tail record (crash mid-write): ignore it and let the
            // next put() overwrite from here
            if (off + 12 + len > fsize) break;
            auto it = index.find(key);
            if (it != index.end()) live -= it->second.len;
            index[key] = {off + 12, len};
            live += len;
            off += 12 + len;
        }
        tail = off;
        if (ftruncate(fd, (off_t)tail) != 0)
            throw std::runtime_error("VarKV: truncate failed");
    }

    void put(uint64_t key, const void* data, uint32_t len) {
        std::lock_guard<std::mutex> g(mu);
        auto it = index.find(key);
        if (it != index.end()) live -= it->second.len;
        char hdr[12];
        memcpy(hdr, &key, 8);
        memcpy(hdr + 8, &len, 4);
        if (pwrite(fd, hdr, 12, (off_t)tail) != 12 ||
            pwrite(fd, data, len, (off_t)(tail + 12)) != (ssize_t)len)
            throw std::runtime_error("VarKV: write failed");
        index[key] = {tail + 12, len};
        tail += 12 + len;
        live += len;
        if (tail > compact_bytes() && live * 2 < tail) compact();
    }
    static uint64_t compact_bytes() {
        // tunable so tests can exercise compaction without writing 64 MB
        static uint64_t v = [] {
            const char* e = getenv("WFA_KV_COMPACT_BYTES");
            return e ? (uint64_t)atoll(e) : (uint64_t)(64u << 20);
        }();
        return v;
    }
    bool get(uint64_t key, std::string& out) {
        std::lock_guard<std::mutex> g(mu);
        auto it = index.find(key);
        if (it == index.end()) return false;
        out.resize(it->second.len);
        if (pread(fd, out.data(), it->second.len, (off_t)it->second.off) !=
            (ssize_t)it->second.len)
            throw std::runtime_error("VarKV: read failed");
        return true;
    }
    bool erase(uint64_t key) {
        std::lock_guard<std::mutex> g(mu);
        auto it = index.find(key);
        if (it == index.end()) return false;
        live -= it->second.len;
        index.erase(it);
        // tombstone record (len = UINT32_MAX): without it a reopened log
        // would replay the old put and resurrect the key
        char hdr[12];
        uint32_t ts = 0xFFFFFFFFu;
        memcpy(hdr, &key, 8);
        memcpy(hdr + 8, &ts, 4);
        if (pwrite(fd, hdr, 12, (off_t)tail) != 12)
            throw std::runtime_error("VarKV: tombstone write failed");
        tail += 12;
        return true;
    }
    size_t size() const { return index.size(); }

    void compact() {  // called under mu; keeps the self-describing format
        std::string tmp = path + ".compact";
        int nfd = ::open(tmp.c_str(), O_RDWR | O_CREAT | O_TRUNC, 0644);
        if (nfd < 0) throw std::runtime_error("VarKV: compact open failed");
        uint64_t ntail = 0;
        std::string buf;
        uint64_t nlive = 0;
        for (auto& [k, loc] : index) {
            buf.resize(loc.len);
            if (pread(fd, buf.data(), loc.len, (off_t)loc.off) != (ssize_t)loc.len)
                throw std::runtime_error("VarKV: compact read failed");
            char hdr[12];
            uint64_t kk = k;
            uint32_t ln = loc.len;
            memcpy(hdr, &kk, 8);
            memcpy(hdr + 8, &ln, 4);
            if (pwrite(nfd, hdr, 12, (off_t)ntail) != 12 ||
                pwrite(nfd, buf.data(), ln, (off_t)(ntail + 12)) != (ssize_t)ln)
                throw std::runtime_error("VarKV: compact write failed");
            index[k] = {ntail + 12, ln};
            ntail += 12 + ln;
            nlive += ln;
        }
        close(fd);
        if (rename(tmp.c_str(), path.c_str()) != 0)
            throw std::runtime_error("VarKV: compact rename failed");
        fd = nfd;
        tail = ntail;
        live = nlive;
    }
};

// bounded write-back LRU in front of the log (reference cache_lru)
struct LruCache {
    VarKV* kv;
    size_t cap;
    std::list<uint64_t> order;  // front = most recent
    struct Ent {
        std::string val;
        std::list<uint64_t>::iterator pos;
        bool dirty = false;
    };
    std::unordered_map<uint64_t, Ent> map;
    int64_t hits = 0, misses = 0;

    LruCache(VarKV* k, size_t c) : kv(k), cap(c) {}

    std::string* get(uint64_t key) {
        auto it = map.find(key);
        if (it != map.end()) {
            order.splice(order.begin(), order, it->second.pos);
            hits++;
            return &it->second.val;
        }
        misses++;
        std::string v;
        if (!kv->get(key, v)) return nullptr;
        return &insert(key, std::move(v), false);
    }
    std::string& put(uint64_t key, std::string v) {
        auto it = map.find(key);
        if (it != map.end()) {
            it->second.val = std::move(v);
            it->second.dirty = true;
            order.splice(order.begin(), order, it->second.pos);
            return it->second.val;
        }
        return insert(key, std::move(v), true);
    }
    std::string& insert(uint64_t key, std::string v, bool dirty) {
        while (map.size() >= cap) {
            uint64_t old = order.back();
            auto oe = map.find(old);
            if (oe->second.dirty)
                kv->put(old, oe->second.val.data(), (uint32_t)oe->second.val.size());
            map.erase(oe);
            order.pop_back();
        }
        order.push_front(key);
        auto& e = map[key];
        e.val = std::move(v);
        e.pos = order.begin();
        e.dirty = dirty;
        return e.val;
    }
    bool evict(uint64_t key) {  // true if the key was cached
        auto it = map.find(key);
        if (it == map.end()) return false;
        order.erase(it->second.pos);
        map.erase(it);
        return true;
    }
    void flush() {
        for (auto& [k, e] : map)
            if (e.dirty) {
                kv->put(k, e.val.data(), (uint32_t)e.val.size());
                e.dirty = false;
            }
    }
};

// ----- P_Reduce: persistent keyed running sum (reference p_reduce.hpp) -----
struct PReduceLogic : OpLogic {
    int col;
    std::string dir;
    int op_id;
    bool keep;
    int64_t cache_cap;
    std::unique_ptr<VarKV> kv;
    std::unique_ptr<LruCache> cache;
    PReduceLogic(int c, std::string dir_, int id, bool keep_, int64_t cc)
        : col(c), dir(std::move(dir_)), op_id(id), keep(keep_), cache_cap(cc) {}

    void warm(RuntimeCtx& ctx) override {
        // deterministic per-replica log path -> withKeepState() resumes the
        // accumulators of a previous run
        std::string path = dir + ".op" + std::to_string(op_id) + ".r" +
                           std::to_string(ctx.replica) + ".log";
        kv = std::make_unique<VarKV>(path, /*fresh=*/!keep);
        cache = std::make_unique<LruCache>(kv.get(), (size_t)cache_cap);
    }

    void process(Batch* b, EmitCtx& out, RuntimeCtx&) override {
        Batch* o = out.new_batch();
        int64_t* x = b->col<int64_t>(col);
        for (int64_t i = 0; i < b->count; ++i) {
            uint64_t key = b->key[i];
            int64_t acc = 0;
            std::string* v = cache->get(key);
            if (v && v->size() == 8) memcpy(&acc, v->data(), 8);
            acc += x[i];
            cache->put(key, std::string((char*)&acc, 8));
            if (o->count == o->capacity) {
                o->watermark = b->watermark;
                out.emit(o);
                o = out.new_batch();
            }
            int64_t w = o->count++;
            o->ts[w] = b->ts[i];
            o->key[w] = key;
            o->col<int64_t>(0)[w] = acc;
        }
        o->watermark = b->watermark;
        if (o->count)
            out.emit(o);
        else
            release(o);
        release(b);
    }
    void on_eos(EmitCtx&, RuntimeCtx&) override { cache->flush(); }
};

std::shared_ptr<OpLogic> make_persist_logic(const std::string& kind,
                                            const std::string& spec,
                                            const std::vector<double>&,
                                            const std::vector<int64_t>& ip,
                                            Engine*, int op_id) {
    if (kind == "p_reduce") {
        // spec = state dir; ip: [col, cache_capacity, keep]
        std::string dir = spec.empty() ? "/tmp/wfa_preduce" : spec;
        return std::make_shared<PReduceLogic>(
            (int)(ip.empty() ? 0 : ip[0]), dir, op_id,
            ip.size() > 2 && ip[2] != 0, ip.size() > 1 ? ip[1] : 1 << 16);
    }
    throw std::runtime_error("unknown persistent logic: " + kind);
}

// Python-facing store handle (P_Map / P_Filter user logic)
std::shared_ptr<void> open_state_store(const std::string& path, int64_t cache_cap,
                                       void** kv_out, void** cache_out,
                                       bool fresh) {
    struct Holder {
        VarKV kv;
        LruCache cache;
        Holder(const std::string& p, size_t c, bool f) : kv(p, f), cache(&kv, c) {}
    };
    auto h = std::make_shared<Holder>(path, (size_t)cache_cap, fresh);
    *kv_out = &h->kv;
    *cache_out = &h->cache;
    return h;
}

std::string* state_cache_get(void* cache, uint64_t key) {
    return ((LruCache*)cache)->get(key);
}
void state_cache_put(void* cache, uint64_t key, const std::string& v) {
    ((LruCache*)cache)->put(key, v);
}
void state_cache_flush(void* cache) { ((LruCache*)cache)->flush(); }
int64_t state_kv_size(void* kv) { return (int64_t)((VarKV*)kv)->size(); }
bool state_kv_erase(void* kv, void* cache, uint64_t key) {
    // drop any cached copy first (a dirty cache entry would re-put it);
    // the key existed if EITHER tier had it (a fresh put may still live
    // only in the write-back cache)
    bool cached = ((LruCache*)cache)->evict(key);
    bool logged = ((VarKV*)kv)->erase(key);
    return cached || logged;
}

}  // namespace wfa
