// Native gateway data plane: batch decision + response assembly.
//
// Reference analog: the Rust edge runtime's direct tools/call resolve +
// fast path (crates/mcp_runtime/src/lib.rs:998-1108) — the per-request
// decision work the reference moved out of Python is here one C++ pass over
// the whole micro-batch. Python prepares row-aligned numpy arrays (kernel
// mask outputs, per-unique-tool flag words) and this module:
//
//   forge_decide: applies the plugin-chain decision order (deny → rewrite →
//     moderation → harm → schema-shape → semantic-cache hit → exact-cache →
//     breaker), writes finished responses (errors + cache hits) into an
//     output arena, and classifies remaining rows (native dispatch /
//     python dispatch / rewrite / host-schema).
//
//   forge_finalize: splices JSON-RPC result responses for dispatched rows,
//     inserts into the exact cache, and assigns semantic-cache slots.
//
// Result payload bytes for cache hits live in a slot-indexed C++ store
// (forge_store_*) mirroring the HBM key matrix's slots.

#ifndef _GNU_SOURCE
#define _GNU_SOURCE
#endif
#include <stdint.h>
#include <string.h>
#include <stdio.h>
#include <functional>
#include <mutex>
#include <string>
#include <thread>
#include <unordered_map>
#include <vector>

extern "C" void forge_parallel_for(int n, void (*fn)(int, void*), void* ctx);
namespace {
inline void run_parallel(int n, const std::function<void(int)>& f) {
    forge_parallel_for(
        n, [](int i, void* c) { (*static_cast<const std::function<void(int)>*>(c))(i); },
        (void*)&f);
}
}  // namespace

namespace {

// ---------------------------------------------------------------- stores

struct SlotStore {
    std::vector<std::string> slots;
    std::mutex mu;   // concurrent batches: put_batch vs decide-side get
};

struct ExactEntry {
    std::string key;     // name \x00 args
    std::string value;   // response result bytes
    double expires;
};

struct ExactCache {
    // sharded by hash: finalize's insert phase runs one thread per shard,
    // and with two batches in flight decide(A) can overlap finalize(B) —
    // each shard carries its own mutex
    static constexpr int SHARDS = 8;
    std::unordered_map<uint64_t, ExactEntry> shards[SHARDS];
    std::mutex shard_mu[SHARDS];
    double ttl = 300.0;
    int shard_of(uint64_t h) const { return (int)(h & (SHARDS - 1)); }
    std::unordered_map<uint64_t, ExactEntry>& map_for(uint64_t h) {
        return shards[h & (SHARDS - 1)];
    }
};

uint64_t fnv64(const uint8_t* p, size_t n, uint64_t h = 1469598103934665603ull) {
    for (size_t i = 0; i < n; ++i) h = (h ^ p[i]) * 1099511628211ull;
    return h;
}

void append_json_escaped(std::string& out, const char* s, size_t n) {
    for (size_t i = 0; i < n; ++i) {
        unsigned char c = s[i];
        switch (c) {
            case '"': out += "\\\""; break;
            case '\\': out += "\\\\"; break;
            case '\n': out += "\\n"; break;
            case '\r': out += "\\r"; break;
            case '\t': out += "\\t"; break;
            default:
                if (c < 0x20) { char b[8]; snprintf(b, sizeof(b), "\\u%04x", c); out += b; }
                else out += (char)c;
        }
    }
}

struct Arena {
    std::string buf;
    int64_t* beg;
    int64_t* end;
    void open(int row) { beg[row] = (int64_t)buf.size(); }
    void close(int row) { end[row] = (int64_t)buf.size(); }
};

void emit_error(Arena& a, int row, const uint8_t* blob, int32_t idb, int32_t ide,
                int code, const std::string& msg) {
    if (idb < 0) { a.beg[row] = a.end[row] = -1; return; }  // notification: no response
    a.open(row);
    a.buf += "{\"jsonrpc\":\"2.0\",\"id\":";
    a.buf.append((const char*)blob + idb, (size_t)(ide - idb));
    a.buf += ",\"error\":{\"code\":";
    a.buf += std::to_string(code);
    a.buf += ",\"message\":\"";
    append_json_escaped(a.buf, msg.data(), msg.size());
    a.buf += "\"}}";
    a.close(row);
}

void emit_result(Arena& a, int row, const uint8_t* blob, int32_t idb, int32_t ide,
                 const char* res, size_t res_n) {
    if (idb < 0) { a.beg[row] = a.end[row] = -1; return; }
    a.open(row);
    a.buf += "{\"jsonrpc\":\"2.0\",\"id\":";
    a.buf.append((const char*)blob + idb, (size_t)(ide - idb));
    a.buf += ",\"result\":";
    a.buf.append(res, res_n);
    a.buf += "}";
    a.close(row);
}

std::string span_str(const uint8_t* blob, int32_t b, int32_t e) {
    return std::string((const char*)blob + b, (size_t)(e - b));
}

}  // namespace

extern "C" {

void* forge_store_new(int capacity) {
    auto* s = new SlotStore();
    s->slots.resize(capacity);
    return s;
}
void forge_store_put(void* store, int slot, const uint8_t* data, int64_t n) {
    auto* s = (SlotStore*)store;
    std::lock_guard<std::mutex> g(s->mu);
    s->slots[slot].assign((const char*)data, (size_t)n);
}
void forge_store_free(void* store) { delete (SlotStore*)store; }

void* forge_cache_new(double ttl) {
    auto* c = new ExactCache();
    c->ttl = ttl;
    return c;
}
void forge_cache_free(void* cache) { delete (ExactCache*)cache; }

// per-tool flag bits
enum : uint32_t {
    TF_REACHABLE = 1u << 0,
    TF_DENY = 1u << 1,
    TF_PII = 1u << 2,
    TF_REGEX = 1u << 3,
    TF_NORM = 1u << 4,
    TF_MOD = 1u << 5,
    TF_HARM = 1u << 6,
    TF_SCHEMA_FAST = 1u << 7,
    TF_SCHEMA_HOST = 1u << 8,
    TF_CACHE = 1u << 9,
    TF_EXACT = 1u << 10,
    TF_BREAKER_OPEN = 1u << 11,
    TF_KNOWN = 1u << 12,
};

// output states
enum : int32_t {
    ST_ANSWERED = 1,
    ST_DISPATCH_NATIVE = 0,
    ST_DISPATCH_PY = 4,
    ST_REWRITE = 2,
    ST_HOST_SCHEMA = 3,
};

// Decide + answer for m rows. Returns total arena bytes (call pattern:
// caller passes a generously sized arena; on overflow returns -needed).
int64_t forge_decide(
    const uint8_t* blob,
    const int32_t* id_beg, const int32_t* id_end,       // [m] (indices into blob; -1 = notification)
    const int32_t* args_beg, const int32_t* args_end,   // [m]
    const int32_t* tool_idx,                            // [m] into tool table; -1 unknown
    const int32_t* name_beg, const int32_t* name_end,   // [m] raw tool-name spans (unknown-tool message)
    const uint32_t* deny_m, const uint32_t* harm_m, const uint32_t* pii_m,
    const uint32_t* regex_m, const uint32_t* norm_m, const uint32_t* schema_m,
    const uint8_t* mod_block, const int32_t* mod_cat, const float* mod_score,
    const uint8_t* hit, const int32_t* hit_slot,
    const uint64_t* user_hash,                          // [m] tenant scope for cache keys
    int m,
    // unique-tool table (nt tools)
    const uint32_t* tool_flags, const uint32_t* tool_required_bits,
    const uint64_t* tool_typed_pairs,   // [nt] up to 4 packed (present<<8|typed) u16 pairs; 0xFFFF = none
    const int32_t* tname_beg, const int32_t* tname_end, const uint8_t* tname_blob,
    const int8_t* tool_native_kind,
    uint32_t nest_bits,
    // string tables
    const uint8_t* deny_words, const int32_t* deny_off, int n_deny,
    const uint8_t* harm_cats, const int32_t* harm_off, int n_harm,
    const uint8_t* mod_cats, const int32_t* mod_off, int n_mod,
    // stores
    void* slot_store, void* exact_cache, double now,
    // outputs
    int32_t* state, int32_t* native_kind_out, int8_t* reason_out,
    uint8_t* arena, int64_t arena_cap, int64_t* resp_beg, int64_t* resp_end)
{
    auto* store = (SlotStore*)slot_store;
    auto* ec = (ExactCache*)exact_cache;
    // rows are independent (stores are mutex/shard-guarded): chunk across
    // threads with per-chunk arenas, stitch + rebase at the end — same
    // pattern as forge_finalize
    int nthreads = m >= 2048 ? 8 : (m >= 256 ? 4 : 1);
    int chunk_sz = (m + nthreads - 1) / nthreads;
    std::vector<std::string> bufs((size_t)nthreads);

    auto run_rows = [&](int t) {
    Arena a;
    a.buf.reserve((size_t)chunk_sz * 48);
    a.beg = resp_beg;
    a.end = resp_end;
    int i0 = t * chunk_sz, i1 = i0 + chunk_sz < m ? i0 + chunk_sz : m;
    for (int i = i0; i < i1; ++i) {
        resp_beg[i] = -1;
        resp_end[i] = -1;
        native_kind_out[i] = -1;
        reason_out[i] = 0;
        int32_t ti = tool_idx[i];
        int32_t idb = id_beg[i], ide = id_end[i];
        if (ti < 0) {
            std::string name = span_str(blob, name_beg[i], name_end[i]);
            emit_error(a, i, blob, idb, ide, -32602, "Tool not found: " + name);
            state[i] = ST_ANSWERED;
            reason_out[i] = 1;
            continue;
        }
        uint32_t fl = tool_flags[ti];
        std::string name = span_str(tname_blob, tname_beg[ti], tname_end[ti]);
        if (!(fl & TF_REACHABLE)) {
            emit_error(a, i, blob, idb, ide, -32002, "Tool " + name + " currently unreachable");
            state[i] = ST_ANSWERED;
            reason_out[i] = 2;
            continue;
        }
        if ((fl & TF_DENY) && deny_m[i]) {
            int pid = __builtin_ctz(deny_m[i]);
            std::string word = pid < n_deny ? span_str(deny_words, deny_off[pid], deny_off[pid + 1]) : "?";
            emit_error(a, i, blob, idb, ide, -32003, "deny_filter: deny word '" + word + "' present");
            state[i] = ST_ANSWERED;
            reason_out[i] = 3;
            continue;
        }
        if (((fl & TF_PII) && pii_m[i]) || ((fl & TF_REGEX) && regex_m[i]) || ((fl & TF_NORM) && norm_m[i])) {
            state[i] = ST_REWRITE;
            continue;
        }
        if ((fl & TF_MOD) && mod_block[i]) {
            int c = mod_cat[i];
            std::string cat = c < n_mod ? span_str(mod_cats, mod_off[c], mod_off[c + 1]) : "?";
            char sc[16];
            snprintf(sc, sizeof(sc), "%.3f", mod_score[i]);
            emit_error(a, i, blob, idb, ide, -32003,
                       "content_moderation: moderation: category " + cat + " score " + sc);
            state[i] = ST_ANSWERED;
            reason_out[i] = 4;
            continue;
        }
        if ((fl & TF_HARM) && harm_m[i]) {
            int pid = __builtin_ctz(harm_m[i]);
            std::string cat = pid < n_harm ? span_str(harm_cats, harm_off[pid], harm_off[pid + 1]) : "?";
            emit_error(a, i, blob, idb, ide, -32003,
                       "harmful_content_detector: harmful content (" + cat + ")");
            state[i] = ST_ANSWERED;
            reason_out[i] = 5;
            continue;
        }
        if (fl & TF_SCHEMA_HOST) {
            state[i] = ST_HOST_SCHEMA;
            continue;
        }
        if (fl & TF_SCHEMA_FAST) {
            uint32_t sm = schema_m[i];
            bool ok = ((sm & tool_required_bits[ti]) == tool_required_bits[ti]) && !(sm & nest_bits);
            if (ok) {
                uint64_t pairs = tool_typed_pairs[ti];
                for (int k = 0; k < 4 && ok; ++k) {
                    uint16_t pk = (uint16_t)(pairs >> (k * 16));
                    if (pk == 0xFFFF) break;
                    uint32_t present = pk >> 8, typed = pk & 0xFF;
                    if ((sm >> present) & 1u) {
                        if (!((sm >> typed) & 1u)) ok = false;
                    }
                }
            }
            if (!ok) {
                state[i] = ST_HOST_SCHEMA;
                continue;
            }
        }
        if ((fl & TF_CACHE) && hit[i] && store != nullptr) {
            bool answered_from_store = false;
            {
                std::lock_guard<std::mutex> g(store->mu);
                const std::string& res = store->slots[hit_slot[i]];
                if (!res.empty()) {
                    emit_result(a, i, blob, idb, ide, res.data(), res.size());
                    answered_from_store = true;
                }
            }
            if (answered_from_store) {
                state[i] = ST_ANSWERED;
                reason_out[i] = 6;
                continue;
            }
        }
        if ((fl & TF_EXACT) && ec != nullptr) {
            uint64_t h = fnv64(tname_blob + tname_beg[ti], (size_t)(tname_end[ti] - tname_beg[ti]));
            h = fnv64((const uint8_t*)"\x00", 1, h);
            uint64_t uh = user_hash ? user_hash[i] : 0;
            h = fnv64((const uint8_t*)&uh, 8, h);  // tenant scope
            h = fnv64(blob + args_beg[i], (size_t)(args_end[i] - args_beg[i]), h);
            int sh = ec->shard_of(h);
            bool answered_from_exact = false;
            {
                std::lock_guard<std::mutex> g(ec->shard_mu[sh]);
                auto& shard = ec->shards[sh];
                auto it = shard.find(h);
                if (it != shard.end() && it->second.expires > now) {
                    emit_result(a, i, blob, idb, ide, it->second.value.data(), it->second.value.size());
                    answered_from_exact = true;
                }
            }
            if (answered_from_exact) {
                state[i] = ST_ANSWERED;
                reason_out[i] = 7;
                continue;
            }
        }
        if (fl & TF_BREAKER_OPEN) {
            emit_error(a, i, blob, idb, ide, -32003, "circuit_breaker: circuit open for tool " + name);
            state[i] = ST_ANSWERED;
            reason_out[i] = 8;
            continue;
        }
        int8_t nk = tool_native_kind[ti];
        if (nk >= 0) {
            native_kind_out[i] = nk;
            state[i] = ST_DISPATCH_NATIVE;
        } else {
            state[i] = ST_DISPATCH_PY;
        }
    }
    bufs[(size_t)t] = std::move(a.buf);
    };
    if (nthreads == 1) {
        run_rows(0);
    } else {
        run_parallel(nthreads, run_rows);
    }
    int64_t total = 0;
    std::vector<int64_t> base((size_t)nthreads, 0);
    for (int t = 0; t < nthreads; ++t) { base[(size_t)t] = total; total += (int64_t)bufs[(size_t)t].size(); }
    if (total > arena_cap) return -total;
    for (int t = 0; t < nthreads; ++t) {
        if (!bufs[(size_t)t].empty())
            memcpy(arena + base[(size_t)t], bufs[(size_t)t].data(), bufs[(size_t)t].size());
        int i0 = t * chunk_sz, i1 = i0 + chunk_sz < m ? i0 + chunk_sz : m;
        for (int i = i0; i < i1; ++i)
            if (resp_beg[i] >= 0) { resp_beg[i] += base[(size_t)t]; resp_end[i] += base[(size_t)t]; }
    }
    return total;
}

// Finalize dispatched rows: splice responses, detect isError, exact-cache
// insert, semcache slot bookkeeping is python-side. Rows with needs_host=1
// are skipped (python handles them).
int64_t forge_finalize(
    const uint8_t* blob,
    const int32_t* id_beg, const int32_t* id_end,
    const int32_t* args_beg, const int32_t* args_end,
    const int32_t* tool_idx,
    const uint64_t* user_hash,                     // [m] tenant scope for cache keys
    int m,
    const int32_t* rows, int n_rows,               // row indices being finalized
    const uint8_t* res_blob, const int64_t* res_beg, const int64_t* res_end,  // [n_rows]
    const uint8_t* needs_host,                      // [n_rows]
    const int32_t* tname_beg, const int32_t* tname_end, const uint8_t* tname_blob,
    const uint32_t* tool_flags,
    void* exact_cache, double now, double exact_ttl,
    uint8_t* arena, int64_t arena_cap, int64_t* resp_beg, int64_t* resp_end,  // [n_rows]
    uint8_t* is_error_out, uint8_t* cacheable_out)                            // [n_rows]
{
    auto* ec = (ExactCache*)exact_cache;
    static const char ERRMARK[] = "\"isError\":true";

    // phase A (parallel): splice responses into per-chunk arenas, compute
    // is_error/cacheable, collect (hash, span) pairs for the exact cache
    int nthreads = n_rows >= 2048 ? 8 : (n_rows >= 256 ? 4 : 1);
    int chunk = (n_rows + nthreads - 1) / nthreads;
    std::vector<std::string> bufs((size_t)nthreads);
    std::vector<std::vector<std::pair<uint64_t, std::pair<int64_t, int64_t>>>> inserts((size_t)nthreads);

    auto run_chunk = [&](int t) {
        int j0 = t * chunk, j1 = j0 + chunk < n_rows ? j0 + chunk : n_rows;
        Arena a;
        a.buf.reserve((size_t)(j1 - j0) * 64);
        a.beg = resp_beg;
        a.end = resp_end;
        auto& ins = inserts[(size_t)t];
        for (int j = j0; j < j1; ++j) {
            resp_beg[j] = -1;
            resp_end[j] = -1;
            is_error_out[j] = 0;
            cacheable_out[j] = 0;
            if (needs_host[j]) continue;
            int i = rows[j];
            const char* res = (const char*)res_blob + res_beg[j];
            size_t rn = (size_t)(res_end[j] - res_beg[j]);
            bool is_err = rn >= sizeof(ERRMARK) - 1 &&
                          memmem(res, rn, ERRMARK, sizeof(ERRMARK) - 1) != nullptr;
            is_error_out[j] = is_err ? 1 : 0;
            emit_result(a, j, blob, id_beg[i], id_end[i], res, rn);
            if (!is_err) {
                cacheable_out[j] = 1;
                int32_t ti = tool_idx[i];
                if (ec != nullptr && ti >= 0 && (tool_flags[ti] & TF_EXACT)) {
                    uint64_t h = fnv64(tname_blob + tname_beg[ti], (size_t)(tname_end[ti] - tname_beg[ti]));
                    h = fnv64((const uint8_t*)"\x00", 1, h);
                    uint64_t uh = user_hash ? user_hash[i] : 0;
                    h = fnv64((const uint8_t*)&uh, 8, h);  // tenant scope (must match decide)
                    h = fnv64(blob + args_beg[i], (size_t)(args_end[i] - args_beg[i]), h);
                    ins.emplace_back(h, std::make_pair(res_beg[j], res_end[j]));
                }
            }
        }
        bufs[(size_t)t] = std::move(a.buf);
    };
    if (nthreads == 1) {
        run_chunk(0);
    } else {
        run_parallel(nthreads, run_chunk);
    }
    int64_t total = 0;
    std::vector<int64_t> base((size_t)nthreads, 0);
    for (int t = 0; t < nthreads; ++t) { base[(size_t)t] = total; total += (int64_t)bufs[(size_t)t].size(); }
    if (total > arena_cap) return -total;
    // phase B (parallel by shard): exact-cache inserts, one thread per shard
    if (ec != nullptr) {
        auto insert_shard = [&](int s) {
            double exp = now + ec->ttl;
            std::lock_guard<std::mutex> g(ec->shard_mu[s]);
            for (auto& vec : inserts)
                for (auto& kv : vec) {
                    if ((int)(kv.first & (ExactCache::SHARDS - 1)) != s) continue;
                    ExactEntry& e = ec->shards[s][kv.first];
                    e.value.assign((const char*)res_blob + kv.second.first,
                                   (size_t)(kv.second.second - kv.second.first));
                    e.expires = exp;
                }
        };
        bool any = false;
        for (auto& vec : inserts) if (!vec.empty()) { any = true; break; }
        if (any && nthreads > 1) {
            run_parallel(ExactCache::SHARDS, insert_shard);
        } else if (any) {
            for (int s = 0; s < ExactCache::SHARDS; ++s) insert_shard(s);
        }
    }
    // stitch chunk arenas + rebase spans
    for (int t = 0; t < nthreads; ++t) {
        if (!bufs[(size_t)t].empty())
            memcpy(arena + base[(size_t)t], bufs[(size_t)t].data(), bufs[(size_t)t].size());
        int j0 = t * chunk, j1 = j0 + chunk < n_rows ? j0 + chunk : n_rows;
        for (int j = j0; j < j1; ++j)
            if (resp_beg[j] >= 0) { resp_beg[j] += base[(size_t)t]; resp_end[j] += base[(size_t)t]; }
    }
    return total;
}

}  // extern "C"

// -------------------------------------------------------------- tool map

extern "C" {

void* forge_toolmap_new(const uint8_t* blob, const int32_t* beg, const int32_t* end, int nt) {
    auto* m = new std::unordered_map<std::string, int32_t>();
    m->reserve((size_t)nt * 2);
    for (int i = 0; i < nt; ++i)
        (*m)[std::string((const char*)blob + beg[i], (size_t)(end[i] - beg[i]))] = i;
    return m;
}

void forge_toolmap_free(void* map) { delete (std::unordered_map<std::string, int32_t>*)map; }

// Resolve raw tool-name spans to tool indices (-1 unknown).
void forge_toolmap_resolve(void* map, const uint8_t* blob,
                           const int32_t* name_beg, const int32_t* name_end, int m,
                           int32_t* tool_idx_out) {
    auto* mp = (std::unordered_map<std::string, int32_t>*)map;
    // tiny per-batch memo: repeated names hit the last-seen fast check
    std::string key;
    for (int i = 0; i < m; ++i) {
        key.assign((const char*)blob + name_beg[i], (size_t)(name_end[i] - name_beg[i]));
        auto it = mp->find(key);
        tool_idx_out[i] = it == mp->end() ? -1 : it->second;
    }
}

// Batch insert of result payloads into the slot store (semcache mirror).
void forge_store_put_batch(void* store, const int32_t* slots, int n,
                           const uint8_t* blob, const int64_t* beg, const int64_t* end) {
    auto* s = (SlotStore*)store;
    std::lock_guard<std::mutex> g(s->mu);
    for (int i = 0; i < n; ++i) {
        if (slots[i] >= 0 && slots[i] < (int32_t)s->slots.size())
            s->slots[slots[i]].assign((const char*)blob + beg[i], (size_t)(end[i] - beg[i]));
    }
}

}  // extern "C"


// copy-out under the lock: a concurrent put may reassign the slot string,
// so returning an interior pointer would dangle with two batches in flight
extern "C" int64_t forge_store_get(void* store, int slot, uint8_t* out, int64_t cap) {
    auto* s = (SlotStore*)store;
    std::lock_guard<std::mutex> g(s->mu);
    if (slot < 0 || slot >= (int)s->slots.size()) return -1;
    const std::string& v = s->slots[slot];
    if ((int64_t)v.size() <= cap && !v.empty())
        memcpy(out, v.data(), v.size());
    return (int64_t)v.size();
}
