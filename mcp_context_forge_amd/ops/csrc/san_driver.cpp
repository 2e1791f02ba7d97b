// Sanitizer driver for the C++ host data plane (ASan / TSan CI tier).
//
// Reference analog: SURVEY.md §5.2 — the reference has no TSan/ASan (pure
// Python hot path); this build's decision plane is native C++, so it gets
// real sanitizer coverage. This driver exercises, with generated data:
//
//   forge_parse_envelopes  (threaded row parsing over a request blob)
//   forge_toolmap_*        (name resolution)
//   forge_decide           (threaded decision pass incl. arena growth)
//   forge_upstream_call_batch (threaded native upstream, canonical gate)
//   forge_finalize         (threaded splice + sharded exact-cache insert)
//   forge_store_* / forge_cache_* (mutex'd stores)
//
// and — the part TSan is for — runs decide/finalize/store_put/store_get
// CONCURRENTLY from several threads against the same stores, the exact
// overlap the pipeline produces with two batches in flight.
//
// Built by tests/test_sanitizers.py with g++ -fsanitize={address,thread}
// (no HIP: these translation units are pure host C++). Exit 0 = clean.

#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

#include <string>
#include <thread>
#include <vector>

extern "C" {
int forge_parse_envelopes(const uint8_t*, const int64_t*, int, int32_t*, int32_t*, int32_t*,
                          int32_t*, int32_t*, int32_t*, int32_t*);
void* forge_toolmap_new(const uint8_t*, const int32_t*, const int32_t*, int);
void forge_toolmap_free(void*);
void forge_toolmap_resolve(void*, const uint8_t*, const int32_t*, const int32_t*, int, int32_t*);
void* forge_store_new(int);
void forge_store_put(void*, int, const uint8_t*, int64_t);
int64_t forge_store_get(void*, int, uint8_t*, int64_t);
void forge_store_free(void*);
void* forge_cache_new(double);
void forge_cache_free(void*);
int64_t forge_decide(
    const uint8_t*, const int32_t*, const int32_t*, const int32_t*, const int32_t*,
    const int32_t*, const int32_t*, const int32_t*,
    const uint32_t*, const uint32_t*, const uint32_t*, const uint32_t*, const uint32_t*,
    const uint32_t*, const uint8_t*, const int32_t*, const float*, const uint8_t*,
    const int32_t*, const uint64_t*, int,
    const uint32_t*, const uint32_t*, const uint64_t*,
    const int32_t*, const int32_t*, const uint8_t*, const int8_t*, uint32_t,
    const uint8_t*, const int32_t*, int, const uint8_t*, const int32_t*, int,
    const uint8_t*, const int32_t*, int,
    void*, void*, double,
    int32_t*, int32_t*, int8_t*, uint8_t*, int64_t, int64_t*, int64_t*);
int64_t forge_upstream_call_batch(const uint8_t*, const int32_t*, const int32_t*,
                                  const int32_t*, int, const char*,
                                  uint8_t*, int64_t, int64_t*, int64_t*);
int64_t forge_rewrite_rows(const uint8_t*, const int32_t*, const int32_t*, int,
                           const uint8_t*, const uint32_t*, uint32_t, int, int, int,
                           const uint8_t*, const int32_t*, int, int,
                           int32_t*, uint32_t*, int32_t*, uint8_t*, int64_t,
                           int64_t*, int64_t*, int64_t*, int64_t*,
                           const uint8_t*, const int32_t*, int, int32_t*,
                           const uint8_t*, const int32_t*, const int32_t*,
                           const int8_t*, const uint8_t*,
                           const int32_t*, const int32_t*, uint8_t*);
int64_t forge_post_rows(const uint8_t*, const int64_t*, const int64_t*, int,
                        const uint8_t*, uint32_t, int,
                        const uint8_t*, const int32_t*, int, int64_t, double,
                        int32_t*, uint32_t*, int32_t*, uint8_t*,
                        uint8_t*, int64_t, int64_t*, int64_t*);
int64_t forge_finalize(
    const uint8_t*, const int32_t*, const int32_t*, const int32_t*, const int32_t*,
    const int32_t*, const uint64_t*, int, const int32_t*, int,
    const uint8_t*, const int64_t*, const int64_t*, const uint8_t*,
    const int32_t*, const int32_t*, const uint8_t*, const uint32_t*,
    void*, double, double,
    uint8_t*, int64_t, int64_t*, int64_t*, uint8_t*, uint8_t*);
}

namespace {

struct Batch {
    std::string blob;
    std::vector<int64_t> offs;
    int n;
};

Batch make_batch(int n, unsigned seed) {
    Batch b;
    b.n = n;
    b.offs.push_back(0);
    unsigned s = seed;
    auto rnd = [&]() { s = s * 1103515245u + 12345u; return (s >> 16) & 0x7fff; };
    for (int i = 0; i < n; ++i) {
        char buf[512];
        int tool = (int)(rnd() % 4);
        int kind = (int)(rnd() % 8);
        if (kind == 0) {
            snprintf(buf, sizeof(buf),
                     "{\"jsonrpc\":\"2.0\",\"id\":%d,\"method\":\"tools/call\","
                     "\"params\":{\"name\":\"tool%d\",\"arguments\":{\"msg\":\"forbidden %u\"}}}",
                     i, tool, rnd());
        } else if (kind == 1) {
            snprintf(buf, sizeof(buf),
                     "{\"jsonrpc\":\"2.0\",\"id\":%d,\"method\":\"ping\"}", i);
        } else if (kind == 2) {
            snprintf(buf, sizeof(buf), "{broken json %u", rnd());
        } else {
            snprintf(buf, sizeof(buf),
                     "{\"jsonrpc\":\"2.0\",\"id\":%d,\"method\":\"tools/call\","
                     "\"params\":{\"name\":\"tool%d\",\"arguments\":"
                     "{\"time\":\"2026-01-01T0%u:00:00Z\",\"source_timezone\":\"UTC\","
                     "\"target_timezone\":\"UTC\",\"n\":%u}}}",
                     i, tool, rnd() % 10, rnd());
        }
        b.blob += buf;
        b.offs.push_back((int64_t)b.blob.size());
    }
    return b;
}

int run_pipeline_once(void* toolmap, void* store, void* cache, int n, unsigned seed) {
    Batch b = make_batch(n, seed);
    std::vector<int32_t> kind(n), idb(n), ide(n), nb(n), ne(n), ab(n), ae(n);
    forge_parse_envelopes((const uint8_t*)b.blob.data(), b.offs.data(), n, kind.data(),
                          idb.data(), ide.data(), nb.data(), ne.data(), ab.data(), ae.data());
    std::vector<int32_t> ti(n);
    forge_toolmap_resolve(toolmap, (const uint8_t*)b.blob.data(), nb.data(), ne.data(), n, ti.data());

    std::vector<uint32_t> zero(n, 0), deny(n, 0);
    std::vector<uint8_t> mod_block(n, 0), hit(n, 0);
    std::vector<int32_t> mod_cat(n, 0), hit_slot(n, -1);
    std::vector<float> mod_score(n, 0.f);
    std::vector<uint64_t> uh(n, 0x1234567ull);
    for (int i = 0; i < n; ++i)
        if (ti[i] >= 0 && (i % 7) == 0) deny[i] = 1;  // some deny blocks

    // tool table: 4 tools, tool3 native echo (kind 2)
    const char* names = "tool0tool1tool2tool3";
    int32_t tnb[4] = {0, 5, 10, 15}, tne[4] = {5, 10, 15, 20};
    uint32_t flags[4] = {0x1 | 0x2 | 0x400, 0x1 | 0x2 | 0x400, 0x1, 0x1 | 0x2};
    uint32_t req_bits[4] = {0, 0, 0, 0};
    uint64_t typed[4] = {~0ull, ~0ull, ~0ull, ~0ull};
    int8_t nk[4] = {-1, -1, -1, 2};
    const char* dw = "forbidden";
    int32_t dwo[2] = {0, 9};

    std::vector<int32_t> state(n), nko(n);
    std::vector<int8_t> reason(n);
    std::vector<int64_t> rb(n), re(n);
    int64_t cap = (int64_t)n * 96 + 4096;
    std::vector<uint8_t> arena((size_t)cap);
    int64_t used = forge_decide(
        (const uint8_t*)b.blob.data(), idb.data(), ide.data(), ab.data(), ae.data(),
        ti.data(), nb.data(), ne.data(),
        deny.data(), zero.data(), zero.data(), zero.data(), zero.data(), zero.data(),
        mod_block.data(), mod_cat.data(), mod_score.data(), hit.data(), hit_slot.data(),
        uh.data(), n,
        flags, req_bits, typed, tnb, tne, (const uint8_t*)names, nk, 0,
        (const uint8_t*)dw, dwo, 1, (const uint8_t*)dw, dwo, 1, (const uint8_t*)dw, dwo, 1,
        store, cache, 1000.0,
        state.data(), nko.data(), reason.data(), arena.data(), cap, rb.data(), re.data());
    if (used < 0) { fprintf(stderr, "arena overflow\n"); return 1; }

    // native upstream for rows decide marked DISPATCH_NATIVE (state 0)
    std::vector<int32_t> njs;
    for (int i = 0; i < n; ++i)
        if (state[i] == 0) njs.push_back(i);
    int m = (int)njs.size();
    std::vector<int64_t> ub(m), ue(m);
    std::vector<int32_t> uab(m), uae(m), kinds(m, 2);
    for (int i = 0; i < m; ++i) { uab[i] = ab[njs[i]]; uae[i] = ae[njs[i]]; }
    int64_t need = forge_upstream_call_batch((const uint8_t*)b.blob.data(), uab.data(), uae.data(),
                                             kinds.data(), m, "2026-01-01T00:00:00Z",
                                             nullptr, 0, ub.data(), ue.data());
    std::vector<uint8_t> ublob((size_t)(need > 0 ? need : 1));
    forge_upstream_call_batch((const uint8_t*)b.blob.data(), uab.data(), uae.data(),
                              kinds.data(), m, "2026-01-01T00:00:00Z",
                              ublob.data(), (int64_t)ublob.size(), ub.data(), ue.data());

    std::vector<uint8_t> needs_host((size_t)(m > 0 ? m : 1), 0);
    std::vector<int64_t> rb2((size_t)(m > 0 ? m : 1)), re2((size_t)(m > 0 ? m : 1));
    std::vector<uint8_t> iserr((size_t)(m > 0 ? m : 1)), cacheable((size_t)(m > 0 ? m : 1));
    int64_t cap2 = (int64_t)ublob.size() + (int64_t)m * 64 + 4096;
    std::vector<uint8_t> arena2((size_t)cap2);
    if (m > 0) {
        int64_t u2 = forge_finalize(
            (const uint8_t*)b.blob.data(), idb.data(), ide.data(), ab.data(), ae.data(),
            ti.data(), uh.data(), n, njs.data(), m,
            ublob.data(), ub.data(), ue.data(), needs_host.data(),
            tnb, tne, (const uint8_t*)names, flags,
            cache, 1000.0, 300.0,
            arena2.data(), cap2, rb2.data(), re2.data(), iserr.data(), cacheable.data());
        if (u2 < 0) { fprintf(stderr, "finalize overflow\n"); return 1; }
    }

    // store churn (semcache slot mirror)
    for (int i = 0; i < 16; ++i) {
        char v[64];
        snprintf(v, sizeof(v), "result-%u-%d", seed, i);
        forge_store_put(store, (int)((seed + i) % 256), (const uint8_t*)v, (int64_t)strlen(v));
        uint8_t out[128];
        forge_store_get(store, (int)((seed + i) % 256), out, sizeof(out));
    }
    return 0;
}

}  // namespace

// the threaded rewrite + post lanes (rewrite.cpp): internal 4/8-way row
// split with per-thread arena stitching — run them ALSO from several
// outer threads to model two batches in flight
static int run_rewrite_lanes_once(int n, unsigned seed) {
    std::string blob;
    std::vector<int32_t> beg((size_t)n), end_((size_t)n);
    for (int i = 0; i < n; ++i) {
        beg[(size_t)i] = (int32_t)blob.size();
        char row[256];
        snprintf(row, sizeof(row),
                 "{\"msg\":\"row %u mail a%u@ex%u.co and 123-45-6789\",\"n\":%d}",
                 seed + (unsigned)i, seed % 97, (unsigned)i % 89, i);
        blob += row;
        end_[(size_t)i] = (int32_t)blob.size();
    }
    std::vector<uint8_t> fl((size_t)n, 3 | 4 | 8);
    std::vector<uint32_t> want((size_t)n, 0x3Fu);
    const char* deny = "forbiddenblocked";
    int32_t deny_off[3] = {0, 9, 16};
    std::vector<int32_t> st((size_t)n), dh((size_t)n), hh((size_t)n);
    std::vector<uint32_t> fb((size_t)n);
    std::vector<int64_t> ob((size_t)n), oe((size_t)n), sb((size_t)n), se((size_t)n);
    int64_t cap = (int64_t)blob.size() * 3 + n * 64 + 4096;
    std::vector<uint8_t> arena((size_t)cap);
    int64_t rc = forge_rewrite_rows((const uint8_t*)blob.data(), beg.data(), end_.data(), n,
                                    fl.data(), want.data(), 0x3Fu, 0, 1, 1,
                                    (const uint8_t*)deny, deny_off, 2, 1,
                                    st.data(), fb.data(), dh.data(), arena.data(), cap,
                                    ob.data(), oe.data(), sb.data(), se.data(),
                                    (const uint8_t*)deny, deny_off, 2, hh.data(),
                                    nullptr, nullptr, nullptr, nullptr, nullptr,
                                    nullptr, nullptr, nullptr);
    if (rc < 0) return 1;
    // feed the rewritten rows (as results) through the post lane
    std::string rblob;
    std::vector<int64_t> rb2((size_t)n), re2((size_t)n);
    for (int i = 0; i < n; ++i) {
        rb2[(size_t)i] = (int64_t)rblob.size();
        if (st[(size_t)i] == 0)
            rblob.append((const char*)arena.data() + ob[(size_t)i],
                         (size_t)(oe[(size_t)i] - ob[(size_t)i]));
        re2[(size_t)i] = (int64_t)rblob.size();
    }
    std::vector<uint8_t> pfl((size_t)n, 7);
    std::vector<int32_t> pst((size_t)n), phh((size_t)n);
    std::vector<uint32_t> pfb((size_t)n);
    std::vector<uint8_t> pie((size_t)n);
    std::vector<int64_t> pob((size_t)n), poe((size_t)n);
    int64_t pcap = (int64_t)rblob.size() * 3 + n * 128 + 4096;
    std::vector<uint8_t> parena((size_t)pcap);
    int64_t prc = forge_post_rows((const uint8_t*)rblob.data(), rb2.data(), re2.data(), n,
                                  pfl.data(), 0x3Fu, 0,
                                  (const uint8_t*)deny, deny_off, 2, 64, 0.01,
                                  pst.data(), pfb.data(), phh.data(), pie.data(),
                                  parena.data(), pcap, pob.data(), poe.data());
    return prc < 0 ? 1 : 0;
}

int main(int argc, char** argv) {
    int iters = argc > 1 ? atoi(argv[1]) : 6;
    int nthreads = argc > 2 ? atoi(argv[2]) : 4;
    int n = argc > 3 ? atoi(argv[3]) : 512;

    const char* names = "tool0tool1tool2tool3";
    int32_t tnb[4] = {0, 5, 10, 15}, tne[4] = {5, 10, 15, 20};
    void* toolmap = forge_toolmap_new((const uint8_t*)names, tnb, tne, 4);
    void* store = forge_store_new(256);
    void* cache = forge_cache_new(300.0);

    // the TSan-relevant shape: several "batches in flight" sharing the
    // same toolmap/stores, exactly like depth-2 edge serving
    int fails = 0;
    for (int it = 0; it < iters; ++it) {
        std::vector<std::thread> ts;
        std::vector<int> rc((size_t)nthreads, 0);
        for (int t = 0; t < nthreads; ++t)
            ts.emplace_back([&, t] { rc[(size_t)t] = run_pipeline_once(toolmap, store, cache, n,
                                                                       (unsigned)(it * 131 + t))
                                                    + run_rewrite_lanes_once(n, (unsigned)(it * 17 + t)); });
        for (auto& th : ts) th.join();
        for (int r : rc) fails += r;
    }
    forge_toolmap_free(toolmap);
    forge_store_free(store);
    forge_cache_free(cache);
    if (fails) { fprintf(stderr, "driver failures: %d\n", fails); return 1; }
    printf("san driver ok: %d iters x %d threads x %d rows\n", iters, nthreads, n);
    return 0;
}
