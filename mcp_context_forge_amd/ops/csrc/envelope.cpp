// JSON-RPC envelope scanner — the native host fast path.
//
// Reference analog: the Rust edge runtime's JSON-RPC id fast parse
// (crates/wrapper/src/json_rpc_id_fast.rs) and the orjson body parse on /rpc
// (main.py:11225). One C++ pass extracts, per request: method kind, the id
// span (spliced verbatim into the response), the tool-name string, and the
// raw `arguments` object span (scanned/featurized on-GPU directly — the
// arguments are never JSON-parsed in Python on the fast path).
//
// Conservative contract: anything that doesn't match the canonical
// tools/call shape exactly is marked NEEDS_PY and handled by the Python
// path, so fast-path coverage never changes semantics.

#include <stdint.h>
#include <string.h>
#include <thread>
#include <functional>
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

struct Cur {
    const uint8_t* p;
    const uint8_t* end;
    bool ok = true;

    bool eof() const { return p >= end; }
    uint8_t peek() const { return *p; }
    void ws() { while (p < end && (*p == ' ' || *p == '\t' || *p == '\n' || *p == '\r')) ++p; }
    bool lit(char c) {
        ws();
        if (eof() || *p != (uint8_t)c) return false;
        ++p;
        return true;
    }
};

// skip a JSON string; p at opening quote. returns false on malformed.
bool skip_string(Cur& c) {
    if (!c.lit('"')) return false;
    while (!c.eof()) {
        uint8_t b = *c.p++;
        if (b == '\\') {
            if (c.eof()) return false;
            ++c.p;
        } else if (b == '"') {
            return true;
        }
    }
    return false;
}

// skip any JSON value; p at first char of value.
bool skip_value(Cur& c) {
    c.ws();
    if (c.eof()) return false;
    uint8_t b = c.peek();
    if (b == '"') return skip_string(c);
    if (b == '{' || b == '[') {
        int depth = 0;
        bool in_str = false, esc = false;
        while (!c.eof()) {
            uint8_t x = *c.p++;
            if (in_str) {
                if (esc) esc = false;
                else if (x == '\\') esc = true;
                else if (x == '"') in_str = false;
            } else if (x == '"') {
                in_str = true;
            } else if (x == '{' || x == '[') {
                ++depth;
            } else if (x == '}' || x == ']') {
                if (--depth == 0) return true;
            }
        }
        return false;
    }
    // number / literal
    const uint8_t* start = c.p;
    while (!c.eof()) {
        uint8_t x = c.peek();
        if (x == ',' || x == '}' || x == ']' || x == ' ' || x == '\t' || x == '\n' || x == '\r') break;
        ++c.p;
    }
    return c.p > start;
}

inline bool key_is(const uint8_t* kb, int klen, const char* name) {
    return klen == (int)strlen(name) && memcmp(kb, name, klen) == 0;
}

}  // namespace

// method kinds
enum : int32_t {
    ENV_NEEDS_PY = -2,   // valid-ish but unusual — Python handles
    ENV_PARSE_ERR = -1,  // malformed — Python re-parses to produce the exact error
    ENV_OTHER = 0,       // valid envelope, not tools/call
    ENV_TOOLS_CALL = 1,  // canonical tools/call with name + arguments object
};

static void parse_rows(
    const uint8_t* data, const int64_t* offsets, int r0, int r1,
    int32_t* kind,
    int32_t* id_beg, int32_t* id_end,
    int32_t* name_beg, int32_t* name_end,
    int32_t* args_beg, int32_t* args_end)
{
    for (int r = r0; r < r1; ++r) {
        const uint8_t* base = data;
        Cur c{data + offsets[r], data + offsets[r + 1]};
        kind[r] = ENV_PARSE_ERR;
        id_beg[r] = id_end[r] = -1;
        name_beg[r] = name_end[r] = -1;
        args_beg[r] = args_end[r] = -1;

        bool saw_jsonrpc = false, saw_method = false, is_tools_call = false, odd = false;
        if (!c.lit('{')) continue;
        c.ws();
        if (!c.eof() && c.peek() == '}') { kind[r] = ENV_NEEDS_PY; continue; }
        bool bad = false;
        while (true) {
            c.ws();
            const uint8_t* kstart = c.p + 1;
            if (!skip_string(c)) { bad = true; break; }
            const uint8_t* kend = c.p - 1;
            if (!c.lit(':')) { bad = true; break; }
            c.ws();
            int klen = (int)(kend - kstart);
            if (key_is(kstart, klen, "jsonrpc")) {
                const uint8_t* v0 = c.p;
                if (!skip_string(c)) { bad = true; break; }
                if (c.p - v0 != 5 || memcmp(v0 + 1, "2.0", 3) != 0) odd = true;
                saw_jsonrpc = true;
            } else if (key_is(kstart, klen, "id")) {
                const uint8_t* v0 = c.p;
                if (!skip_value(c)) { bad = true; break; }
                // only string/number/null ids ride the fast path
                uint8_t f = *v0;
                if (f == '{' || f == '[' || f == 't' || f == 'f') odd = true;
                if (!(f == 'n' && c.p - v0 == 4)) {  // null id = notification-ish → absent
                    id_beg[r] = (int32_t)(v0 - base);
                    id_end[r] = (int32_t)(c.p - base);
                }
            } else if (key_is(kstart, klen, "method")) {
                const uint8_t* v0 = c.p;
                if (c.eof() || c.peek() != '"' || !skip_string(c)) { bad = true; break; }
                int mlen = (int)(c.p - v0 - 2);
                saw_method = mlen > 0;
                if (mlen == 10 && memcmp(v0 + 1, "tools/call", 10) == 0) is_tools_call = true;
                else if (mlen >= 4 && memcmp(v0 + 1, "rpc.", 4) == 0) odd = true;
            } else if (key_is(kstart, klen, "params")) {
                c.ws();
                if (c.eof()) { bad = true; break; }
                if (c.peek() != '{') {
                    if (c.peek() == '[') { odd = true; if (!skip_value(c)) { bad = true; break; } }
                    else { odd = true; if (!skip_value(c)) { bad = true; break; } }
                } else {
                    // walk params object: want "name" (string) and "arguments" (object)
                    const uint8_t* pstart = c.p;
                    ++c.p;  // consume '{'
                    c.ws();
                    if (!c.eof() && c.peek() == '}') { ++c.p; }
                    else {
                        while (true) {
                            c.ws();
                            const uint8_t* pk0 = c.p + 1;
                            if (!skip_string(c)) { bad = true; break; }
                            const uint8_t* pk1 = c.p - 1;
                            if (!c.lit(':')) { bad = true; break; }
                            c.ws();
                            int pklen = (int)(pk1 - pk0);
                            if (key_is(pk0, pklen, "name")) {
                                const uint8_t* v0 = c.p;
                                if (c.eof() || c.peek() != '"' || !skip_string(c)) { bad = true; break; }
                                // fast path only for escape-free names
                                bool has_esc = false;
                                for (const uint8_t* q = v0 + 1; q < c.p - 1; ++q)
                                    if (*q == '\\') { has_esc = true; break; }
                                if (has_esc) odd = true;
                                name_beg[r] = (int32_t)(v0 + 1 - base);
                                name_end[r] = (int32_t)(c.p - 1 - base);
                            } else if (key_is(pk0, pklen, "arguments")) {
                                c.ws();
                                const uint8_t* v0 = c.p;
                                if (c.eof()) { bad = true; break; }
                                if (c.peek() == '{') {
                                    if (!skip_value(c)) { bad = true; break; }
                                    args_beg[r] = (int32_t)(v0 - base);
                                    args_end[r] = (int32_t)(c.p - base);
                                } else {
                                    odd = true;
                                    if (!skip_value(c)) { bad = true; break; }
                                }
                            } else {
                                if (!skip_value(c)) { bad = true; break; }
                            }
                            c.ws();
                            if (!c.eof() && c.peek() == ',') { ++c.p; continue; }
                            if (!c.eof() && c.peek() == '}') { ++c.p; break; }
                            bad = true; break;
                        }
                        if (bad) break;
                    }
                    (void)pstart;
                }
            } else {
                if (!skip_value(c)) { bad = true; break; }
            }
            c.ws();
            if (!c.eof() && c.peek() == ',') { ++c.p; continue; }
            if (!c.eof() && c.peek() == '}') { ++c.p; break; }
            bad = true;
            break;
        }
        if (bad) { kind[r] = ENV_PARSE_ERR; continue; }
        c.ws();
        if (!c.eof()) { kind[r] = ENV_PARSE_ERR; continue; }  // trailing garbage
        if (!saw_jsonrpc || !saw_method) { kind[r] = ENV_PARSE_ERR; continue; }
        if (odd) { kind[r] = ENV_NEEDS_PY; continue; }
        if (is_tools_call) {
            if (name_beg[r] < 0) { kind[r] = ENV_NEEDS_PY; continue; }  // missing name → exact Python error
            if (args_beg[r] < 0) {
                // absent arguments == {} — point at a shared empty-object span (host fills)
                kind[r] = ENV_TOOLS_CALL;
            } else {
                kind[r] = ENV_TOOLS_CALL;
            }
        } else {
            kind[r] = ENV_OTHER;
        }
    }
}

// Rows are independent — chunk across threads for large batches.
extern "C" int forge_parse_envelopes(
    const uint8_t* data, const int64_t* offsets, int n,
    int32_t* kind,
    int32_t* id_beg, int32_t* id_end,        // -1,-1 = absent (notification)
    int32_t* name_beg, int32_t* name_end,    // tool name string contents
    int32_t* args_beg, int32_t* args_end)    // raw arguments object span
{
    int nthreads = n >= 2048 ? 8 : (n >= 512 ? 4 : 1);
    if (nthreads == 1) {
        parse_rows(data, offsets, 0, n, kind, id_beg, id_end, name_beg, name_end, args_beg, args_end);
        return 0;
    }
    int chunk = (n + nthreads - 1) / nthreads;
    run_parallel(nthreads, [&](int t) {
        int r0 = t * chunk, r1 = r0 + chunk < n ? r0 + chunk : n;
        if (r0 >= r1) return;
        parse_rows(data, offsets, r0, r1,
                   kind, id_beg, id_end, name_beg, name_end, args_beg, args_end);
    });
    return 0;
}
