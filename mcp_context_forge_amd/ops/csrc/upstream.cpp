// Native in-process MCP upstream — the fast_time_server analog.
//
// The reference's benchmark upstream is a native Go binary
// (docker-compose.yml:1485 fast_time_server, targeted by tests/hey/
// payload2.json `convert_time`). This is the same thing as a C++ batch
// call: for each request it parses the raw `arguments` span, runs the tool
// handler (convert_time / get_system_time / echo), and serializes a full
// MCP tool-result JSON into one output blob. The gateway splices these
// bytes into JSON-RPC responses without re-parsing.
//
// Handler kinds: 0 = convert_time, 1 = get_system_time, 2 = echo.

#include <stdint.h>
#include <stdlib.h>
#include <string.h>
#include <functional>
#include <string>
#include <thread>
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

// extract the raw span of `"key": <value>` inside a flat JSON object.
// Returns value span [vb, ve) or false.
bool find_key(const uint8_t* b, const uint8_t* e, const char* key,
              const uint8_t** vb, const uint8_t** ve) {
    size_t klen = strlen(key);
    const uint8_t* p = b;
    if (p >= e || *p != '{') return false;
    ++p;
    bool in_str = false, esc = false;
    int depth = 0;
    // scan top-level keys
    while (p < e) {
        // skip ws
        while (p < e && (*p == ' ' || *p == '\t' || *p == '\n' || *p == '\r')) ++p;
        if (p < e && *p == '}') return false;
        if (p >= e || *p != '"') return false;
        const uint8_t* k0 = ++p;
        while (p < e && *p != '"') { if (*p == '\\') ++p; ++p; }
        const uint8_t* k1 = p;
        if (p >= e) return false;
        ++p;  // closing quote
        while (p < e && (*p == ' ' || *p == '\t' || *p == '\n' || *p == '\r')) ++p;
        if (p >= e || *p != ':') return false;
        ++p;
        while (p < e && (*p == ' ' || *p == '\t' || *p == '\n' || *p == '\r')) ++p;
        const uint8_t* v0 = p;
        // skip value
        if (p < e && (*p == '{' || *p == '[')) {
            depth = 0; in_str = false; esc = false;
            while (p < e) {
                uint8_t x = *p++;
                if (in_str) { if (esc) esc = false; else if (x == '\\') esc = true; else if (x == '"') in_str = false; }
                else if (x == '"') in_str = true;
                else if (x == '{' || x == '[') ++depth;
                else if (x == '}' || x == ']') { if (--depth == 0) break; }
            }
        } else if (p < e && *p == '"') {
            ++p;
            while (p < e && *p != '"') { if (*p == '\\') ++p; ++p; }
            if (p < e) ++p;
        } else {
            while (p < e && *p != ',' && *p != '}') ++p;
        }
        const uint8_t* v1 = p;
        if ((size_t)(k1 - k0) == klen && memcmp(k0, key, klen) == 0) {
            *vb = v0;
            *ve = v1;
            return true;
        }
        while (p < e && (*p == ' ' || *p == '\t' || *p == '\n' || *p == '\r')) ++p;
        if (p < e && *p == ',') { ++p; continue; }
        return false;
    }
    return false;
}

void append_json_escaped(std::string& out, const uint8_t* b, const uint8_t* e) {
    for (const uint8_t* p = b; p < e; ++p) {
        uint8_t c = *p;
        switch (c) {
            case '"': out += "\\\""; break;
            case '\\': out += "\\\\"; break;
            case '\n': out += "\\n"; break;
            case '\r': out += "\\r"; break;
            case '\t': out += "\\t"; break;
            default:
                if (c < 0x20) {
                    char buf[8];
                    snprintf(buf, sizeof(buf), "\\u%04x", c);
                    out += buf;
                } else {
                    out += (char)c;
                }
        }
    }
}

void append_span(std::string& out, const uint8_t* b, const uint8_t* e) {
    out.append((const char*)b, (size_t)(e - b));
}

// Strict-canonical gate for the raw-span fast path. Handlers copy argument
// VALUE bytes verbatim into results; that is only byte-identical to the
// CPU reference (json.loads -> json.dumps, compact, ensure_ascii) when the
// span already IS that canonical form. Anything else — whitespace outside
// strings, any backslash escape, non-ASCII bytes, non-integer numbers —
// returns false and the row is answered on the exact host path instead
// (parity-fuzz finding: pretty-printed args echoed verbatim diverged).
bool is_canonical_span(const uint8_t* b, const uint8_t* e) {
    const uint8_t* p = b;
    bool in_str = false;
    while (p < e) {
        uint8_t c = *p;
        if (c == '\\') return false;     // escapes decode differently
        if (c >= 0x80) return false;     // ensure_ascii would re-escape
        if (in_str) {
            if (c == '"') in_str = false;
            ++p;
            continue;
        }
        if (c == '"') { in_str = true; ++p; continue; }
        if (c == ' ' || c == '\t' || c == '\n' || c == '\r') return false;
        if (c == '-' || (c >= '0' && c <= '9')) {  // number: integers only
            bool neg = c == '-';
            if (neg) ++p;
            const uint8_t* d0 = p;
            while (p < e && *p >= '0' && *p <= '9') ++p;
            if (p == d0) return false;                       // bare '-'
            if (p < e && (*p == '.' || *p == 'e' || *p == 'E')) return false;
            if (p - d0 > 1 && *d0 == '0') return false;      // leading zeros
            if (neg && p - d0 == 1 && *d0 == '0') return false;  // -0 → python re-dumps as 0
            continue;
        }
        if (c == 't') { if (e - p < 4 || memcmp(p, "true", 4) != 0) return false; p += 4; continue; }
        if (c == 'f') { if (e - p < 5 || memcmp(p, "false", 5) != 0) return false; p += 5; continue; }
        if (c == 'n') { if (e - p < 4 || memcmp(p, "null", 4) != 0) return false; p += 4; continue; }
        if (c == '{' || c == '}' || c == '[' || c == ']' || c == ':' || c == ',') { ++p; continue; }
        return false;
    }
    return !in_str;
}

}  // namespace

// Batch upstream call.
//   data/args spans: raw argument objects (may be -1,-1 = empty)
//   kinds[r]: handler kind (0/1/2); now_iso: host-provided wall clock string
// Output: caller-provided growing buffer protocol — we return the required
// size on first call (out=null), caller allocates and calls again.
static void upstream_rows(
    const uint8_t* data, const int32_t* args_beg, const int32_t* args_end,
    const int32_t* kinds, int r0, int r1, const char* now_iso,
    std::string& buf, int64_t* res_beg, int64_t* res_end)
{
    buf.reserve((size_t)(r1 - r0) * 192);
    static const uint8_t EMPTY[2] = {'{', '}'};
    for (int r = r0; r < r1; ++r) {
        res_beg[r] = (int64_t)buf.size();
        const uint8_t* ab = args_beg[r] >= 0 ? data + args_beg[r] : EMPTY;
        const uint8_t* ae = args_beg[r] >= 0 ? data + args_end[r] : EMPTY + 2;
        int k = kinds[r];
        if (!is_canonical_span(ab, ae)) {
            res_end[r] = res_beg[r];  // empty span = punt to the host path
            continue;
        }
        if (k == 0) {  // convert_time
            const uint8_t *tb, *te, *sb, *se, *gb, *ge;
            bool ht = find_key(ab, ae, "time", &tb, &te);
            bool hs = find_key(ab, ae, "source_timezone", &sb, &se);
            bool hg = find_key(ab, ae, "target_timezone", &gb, &ge);
            buf += "{\"content\":[{\"type\":\"text\",\"text\":\"converted\"}],\"structuredContent\":{\"time\":";
            if (ht) append_span(buf, tb, te); else buf += "\"1970-01-01T00:00:00Z\"";
            buf += ",\"source_timezone\":";
            if (hs) append_span(buf, sb, se); else buf += "\"UTC\"";
            buf += ",\"target_timezone\":";
            if (hg) append_span(buf, gb, ge); else buf += "\"UTC\"";
            buf += ",\"converted\":true},\"isError\":false}";
        } else if (k == 1) {  // get_system_time
            const uint8_t *zb, *ze;
            bool hz = find_key(ab, ae, "timezone", &zb, &ze);
            buf += "{\"content\":[{\"type\":\"text\",\"text\":\"";
            buf += now_iso;
            buf += "\"}],\"structuredContent\":{\"time\":\"";
            buf += now_iso;
            buf += "\",\"timezone\":";
            if (hz) append_span(buf, zb, ze); else buf += "\"UTC\"";
            buf += "},\"isError\":false}";
        } else {  // echo
            buf += "{\"content\":[{\"type\":\"text\",\"text\":\"";
            append_json_escaped(buf, ab, ae);
            buf += "\"}],\"structuredContent\":";
            append_span(buf, ab, ae);
            buf += ",\"isError\":false}";
        }
        res_end[r] = (int64_t)buf.size();
    }
}

// Batch entry: rows are independent, so chunk them across threads (each
// with a private buffer), then prefix-sum the chunk sizes, rebase the
// per-row spans, and copy chunks into `out`. Same grow-retry contract as
// before: returns the needed size; copies only when it fits out_cap.
extern "C" int64_t forge_upstream_call_batch(
    const uint8_t* data, const int32_t* args_beg, const int32_t* args_end,
    const int32_t* kinds, int n, const char* now_iso,
    uint8_t* out, int64_t out_cap, int64_t* res_beg, int64_t* res_end)
{
    int nthreads = n >= 2048 ? 8 : (n >= 256 ? 4 : 1);
    if (nthreads == 1) {
        std::string buf;
        upstream_rows(data, args_beg, args_end, kinds, 0, n, now_iso, buf, res_beg, res_end);
        if (out != nullptr && (int64_t)buf.size() <= out_cap)
            memcpy(out, buf.data(), buf.size());
        return (int64_t)buf.size();
    }
    std::vector<std::string> bufs(nthreads);
    int chunk = (n + nthreads - 1) / nthreads;
    run_parallel(nthreads, [&](int t) {
        int r0 = t * chunk, r1 = r0 + chunk < n ? r0 + chunk : n;
        if (r0 >= r1) return;
        upstream_rows(data, args_beg, args_end, kinds, r0, r1, now_iso,
                      bufs[(size_t)t], res_beg, res_end);
    });
    int64_t total = 0;
    std::vector<int64_t> base(nthreads, 0);
    for (int t = 0; t < nthreads; ++t) { base[t] = total; total += (int64_t)bufs[t].size(); }
    if (out == nullptr || total > out_cap) return total;
    for (int t = 0; t < nthreads; ++t) {
        if (!bufs[t].empty()) memcpy(out + base[t], bufs[t].data(), bufs[t].size());
        int r0 = t * chunk, r1 = r0 + chunk < n ? r0 + chunk : n;
        for (int r = r0; r < r1; ++r) { res_beg[r] += base[t]; res_end[r] += base[t]; }
    }
    return total;
}
