// Native rewrite pass: the flagged-row exact semantics in C++.
//
// Reference behavior being replicated (the CPU chain order, plugins/
// builtin.py): argument_normalizer@15 (whitespace collapse + strip; NFC is
// identity on ASCII) → pii_filter@30 (six fixed patterns, re.sub global
// replace, patterns applied sequentially on the already-substituted text)
// over every STRING value of the parsed JSON arguments, then canonical
// re-serialization (json.dumps sort_keys=True, separators=(",",":"),
// ensure_ascii=True).
//
// ENVELOPE: a row is handled natively ONLY when byte-exact equivalence
// with the Python path is provable:
//   * pure-ASCII span (NFC identity; ensure_ascii re-escaping identity)
//   * JSON parses, depth <= 32, numbers are plain integers (float repr
//     round-trips are Python-specific)
//   * no regex_filter work requested (user-configured Python regexes)
//   * string escapes limited to \" \\ \/ \b \f \n \r \t and \u00XX (ASCII)
// Anything else returns PUNT and the row takes the existing Python path.
// tests/test_rewrite_native.py cross-validates against the Python
// implementation over randomized corpora; the GPU parity fuzzer covers
// the full pipeline.

#ifndef _GNU_SOURCE
#define _GNU_SOURCE
#endif
#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

#include <algorithm>
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

constexpr int MAX_DEPTH = 32;

inline bool is_word(uint8_t c) {  // Python re \w (ASCII)
    return (c >= 'A' && c <= 'Z') || (c >= 'a' && c <= 'z') || (c >= '0' && c <= '9') || c == '_';
}
inline bool is_digit(uint8_t c) { return c >= '0' && c <= '9'; }
inline bool is_alpha(uint8_t c) { return (c >= 'A' && c <= 'Z') || (c >= 'a' && c <= 'z'); }

// ---------------------------------------------------------------- PII

// word boundary \b at position i of s (between i-1 and i)
inline bool wb(const std::string& s, size_t i) {
    bool l = i > 0 && is_word((uint8_t)s[i - 1]);
    bool r = i < s.size() && is_word((uint8_t)s[i]);
    return l != r;
}

inline bool klass_email_local(uint8_t c) {
    return is_word(c) || c == '.' || c == '%' || c == '+' || c == '-';
}
inline bool klass_email_domain(uint8_t c) {
    return is_alpha(c) || is_digit(c) || c == '.' || c == '-';
}
inline bool is_sep_sd(uint8_t c) { return c == ' ' || c == '-'; }          // [ \-]
inline bool is_sep_pd(uint8_t c) { return c == ' ' || c == '.' || c == '-'; }  // [ .\-]

// Each matcher: try a match STARTING at i; on success set len. Mirrors the
// exact python regex (incl. \b positions).
// ssn: \b\d{3}-\d{2}-\d{4}\b
bool m_ssn(const std::string& s, size_t i, size_t& len) {
    if (!wb(s, i)) return false;
    if (i + 11 > s.size()) return false;
    const char* p = s.data() + i;
    for (int k = 0; k < 3; ++k) if (!is_digit(p[k])) return false;
    if (p[3] != '-') return false;
    for (int k = 4; k < 6; ++k) if (!is_digit(p[k])) return false;
    if (p[6] != '-') return false;
    for (int k = 7; k < 11; ++k) if (!is_digit(p[k])) return false;
    if (!wb(s, i + 11)) return false;
    len = 11;
    return true;
}

// email: [\w.%+\-]+@[A-Za-z0-9.\-]+\.[A-Za-z]{2,}   (no \b)
bool m_email(const std::string& s, size_t i, size_t& len) {
    size_t j = i;
    while (j < s.size() && klass_email_local((uint8_t)s[j])) ++j;
    if (j == i || j >= s.size() || s[j] != '@') return false;
    size_t d0 = j + 1, d = d0;
    while (d < s.size() && klass_email_domain((uint8_t)s[d])) ++d;
    if (d == d0) return false;
    // backtrack: rightmost '.' inside [d0, d) with >=2 alphas after it
    // (alphas counted within the maximal domain run — greedy {2,} then
    // nothing follows, so the match ends at the end of the alpha run)
    for (size_t dot = d; dot-- > d0 + 1;) {  // dot position candidate; needs chars before it
        if (s[dot] != '.') continue;
        size_t a = dot + 1, e = a;
        while (e < d && is_alpha((uint8_t)s[e])) ++e;
        if (e - a >= 2) {
            len = e - i;
            return true;
        }
    }
    return false;
}

// credit_card: \b\d{4}[ \-]\d{4}[ \-]\d{4}[ \-]\d{4}\b
bool m_cc(const std::string& s, size_t i, size_t& len) {
    if (!wb(s, i)) return false;
    if (i + 19 > s.size()) return false;
    const char* p = s.data() + i;
    for (int g = 0; g < 4; ++g) {
        for (int k = 0; k < 4; ++k)
            if (!is_digit(p[g * 5 + k])) return false;
        if (g < 3 && !is_sep_sd((uint8_t)p[g * 5 + 4])) return false;
    }
    if (!wb(s, i + 19)) return false;
    len = 19;
    return true;
}

// phone: \(?\b\d{3}\)?[ .\-]\d{3}[ .\-]\d{4}\b
bool m_phone(const std::string& s, size_t i, size_t& len) {
    size_t j = i;
    if (j < s.size() && s[j] == '(') ++j;          // \(? greedy
    if (!wb(s, j)) {
        // backtrack \(? to zero-width only if '(' consumed
        if (j == i) return false;
        j = i;
        if (!wb(s, j)) return false;
    }
    if (j + 3 > s.size()) return false;
    for (int k = 0; k < 3; ++k) if (!is_digit((uint8_t)s[j + k])) return false;
    size_t k2 = j + 3;
    if (k2 < s.size() && s[k2] == ')') ++k2;       // \)? greedy (no backtrack needed:
    if (k2 >= s.size() || !is_sep_pd((uint8_t)s[k2])) {
        // backtrack \)? — try without consuming ')'
        if (k2 > j + 3) {
            k2 = j + 3;
            if (k2 >= s.size() || !is_sep_pd((uint8_t)s[k2])) return false;
        } else {
            return false;
        }
    }
    ++k2;
    if (k2 + 3 > s.size()) return false;
    for (int k = 0; k < 3; ++k) if (!is_digit((uint8_t)s[k2 + k])) return false;
    k2 += 3;
    if (k2 >= s.size() || !is_sep_pd((uint8_t)s[k2])) return false;
    ++k2;
    if (k2 + 4 > s.size()) return false;
    for (int k = 0; k < 4; ++k) if (!is_digit((uint8_t)s[k2 + k])) return false;
    k2 += 4;
    if (!wb(s, k2)) return false;
    len = k2 - i;
    return true;
}

// ipv4: \b\d{1,3}\.\d{1,3}\.\d{1,3}\.\d{1,3}\b  (greedy {1,3} with backtrack)
bool m_ipv4(const std::string& s, size_t i, size_t& len) {
    if (!wb(s, i)) return false;
    size_t j = i;
    // octets 1..3: digits{1..3} then '.'; greedy semantics: take max digits
    // (<=3) such that the next char is '.'; since digits can't be '.', the
    // run length is fixed by the input — just bound it
    for (int oct = 0; oct < 3; ++oct) {
        size_t d = 0;
        while (j + d < s.size() && is_digit((uint8_t)s[j + d]) && d < 3) ++d;
        if (d == 0) return false;
        if (j + d >= s.size() || s[j + d] != '.') return false;
        // python would fail if a 4th digit precedes '.', because {1,3} can't
        // cover it and '.' won't match a digit — detect: next char after the
        // run of <=3 must be '.', but if the digit run continues past 3 the
        // char at j+3 is a digit, not '.', and backtracking to shorter runs
        // still faces a digit → no match at this i
        j += d + 1;
    }
    size_t d = 0;
    while (j + d < s.size() && is_digit((uint8_t)s[j + d]) && d < 3) ++d;
    if (d == 0) return false;
    // \b after: if a 4th digit follows, {1,3} backtracks: \b between digit
    // and digit fails for d=3,2,1 → no match
    if (j + d < s.size() && is_digit((uint8_t)s[j + d])) return false;
    j += d;
    if (!wb(s, j)) return false;
    len = j - i;
    return true;
}

// aws_key: \bAKIA[0-9A-Z]{16}\b
bool m_aws(const std::string& s, size_t i, size_t& len) {
    if (!wb(s, i)) return false;
    if (i + 20 > s.size()) return false;
    if (memcmp(s.data() + i, "AKIA", 4) != 0) return false;
    for (int k = 4; k < 20; ++k) {
        uint8_t c = (uint8_t)s[i + k];
        if (!((c >= '0' && c <= '9') || (c >= 'A' && c <= 'Z'))) return false;
    }
    if (!wb(s, i + 20)) return false;
    len = 20;
    return true;
}

typedef bool (*Matcher)(const std::string&, size_t, size_t&);
struct PiiDef { const char* name; const char* repl; Matcher fn; };
const PiiDef PII[6] = {
    {"ssn", "[SSN_REDACTED]", m_ssn},
    {"email", "[EMAIL_REDACTED]", m_email},
    {"credit_card", "[CREDIT_CARD_REDACTED]", m_cc},
    {"phone", "[PHONE_REDACTED]", m_phone},
    {"ipv4", "[IPV4_REDACTED]", m_ipv4},
    {"aws_key", "[AWS_KEY_REDACTED]", m_aws},
};

// re.sub semantics: left-to-right non-overlapping global replace
bool pii_sub_one(std::string& s, const PiiDef& p) {
    std::string out;
    bool any = false;
    size_t i = 0;
    const bool email = p.fn == m_email;
    while (i < s.size()) {
        size_t len = 0;
        if (p.fn(s, i, len)) {
            out += p.repl;
            i += len;
            any = true;
        } else if (email && klass_email_local((uint8_t)s[i])) {
            // every start inside one local-part run shares the same '@'
            // position and domain, so they all fail identically — skip the
            // run. Kills the O(L²) cost on long word runs (python re pays
            // it; outputs are unchanged, only the scan order collapses).
            size_t j = i + 1;
            while (j < s.size() && klass_email_local((uint8_t)s[j])) ++j;
            out.append(s, i, j - i);
            i = j;
        } else {
            out += s[i++];
        }
    }
    if (any) s.swap(out);
    return any;
}

// patterns applied sequentially over the already-substituted text, in the
// fixed order, gated by active_mask (config categories) and want_mask
// (the GPU scan's per-row accept bits — mask_text_subset semantics)
uint32_t pii_mask_text(std::string& s, uint32_t active_mask, uint32_t want_mask) {
    uint32_t found = 0;
    for (int k = 0; k < 6; ++k) {
        if (!((active_mask >> k) & 1) || !((want_mask >> k) & 1)) continue;
        if (pii_sub_one(s, PII[k])) found |= 1u << k;
    }
    return found;
}

// ---------------------------------------------------------------- normalize

// argument_normalizer.norm for ASCII: collapse [ \t\f\v]+ → " ", then strip()
// (python str.strip removes all whitespace incl. \n)
void norm_ascii(std::string& s, bool collapse, bool strip) {
    if (collapse) {
        std::string out;
        out.reserve(s.size());
        for (size_t i = 0; i < s.size();) {
            char c = s[i];
            if (c == ' ' || c == '\t' || c == '\f' || c == '\v') {
                out += ' ';
                while (i < s.size() && (s[i] == ' ' || s[i] == '\t' || s[i] == '\f' || s[i] == '\v')) ++i;
            } else {
                out += c;
                ++i;
            }
        }
        s.swap(out);
    }
    if (strip) {
        size_t a = 0, b = s.size();
        auto is_ws = [](char c) {
            return c == ' ' || c == '\t' || c == '\n' || c == '\r' || c == '\f' || c == '\v';
        };
        while (a < b && is_ws(s[a])) ++a;
        while (b > a && is_ws(s[b - 1])) --b;
        s = s.substr(a, b - a);
    }
}

// ---------------------------------------------------------------- JSON

struct Val;
struct Member { std::string key; size_t vidx; };
struct Val {
    // FLT is never produced by the parser (floats punt) — only by the TOON
    // meta builder, with str holding the exact Python repr text
    enum Kind { OBJ, ARR, STR, INT, TRUE_, FALSE_, NULL_, FLT } kind;
    std::string str;                 // STR: decoded text; INT: literal digits
    std::vector<Member> members;     // OBJ
    std::vector<size_t> items;       // ARR
};

struct Parser {
    const uint8_t* p;
    const uint8_t* e;
    std::vector<Val>& pool;
    bool ok = true;

    void skip_ws() {
        while (p < e && (*p == ' ' || *p == '\t' || *p == '\n' || *p == '\r')) ++p;
    }

    bool decode_string(std::string& out) {
        // at opening quote
        if (p >= e || *p != '"') return false;
        ++p;
        while (p < e) {
            uint8_t c = *p;
            if (c == '"') { ++p; return true; }
            if (c >= 0x80) return false;   // non-ASCII → punt
            if (c == '\\') {
                if (p + 1 >= e) return false;
                uint8_t n = p[1];
                p += 2;
                switch (n) {
                    case '"': out += '"'; break;
                    case '\\': out += '\\'; break;
                    case '/': out += '/'; break;
                    case 'b': out += '\b'; break;
                    case 'f': out += '\f'; break;
                    case 'n': out += '\n'; break;
                    case 'r': out += '\r'; break;
                    case 't': out += '\t'; break;
                    case 'u': {
                        if (p + 4 > e) return false;
                        unsigned v = 0;
                        for (int k = 0; k < 4; ++k) {
                            uint8_t h = p[k];
                            v <<= 4;
                            if (h >= '0' && h <= '9') v |= h - '0';
                            else if (h >= 'a' && h <= 'f') v |= h - 'a' + 10;
                            else if (h >= 'A' && h <= 'F') v |= h - 'A' + 10;
                            else return false;
                        }
                        if (v > 0x7F) return false;  // decodes to non-ASCII → punt
                        out += (char)v;
                        p += 4;
                        break;
                    }
                    default: return false;
                }
            } else {
                out += (char)c;
                ++p;
            }
        }
        return false;
    }

    size_t parse_value(int depth) {
        if (depth > MAX_DEPTH) { ok = false; return 0; }
        skip_ws();
        if (p >= e) { ok = false; return 0; }
        size_t idx = pool.size();
        pool.emplace_back();
        uint8_t c = *p;
        if (c == '{') {
            pool[idx].kind = Val::OBJ;
            ++p;
            skip_ws();
            if (p < e && *p == '}') { ++p; return idx; }
            while (ok) {
                skip_ws();
                std::string key;
                if (!decode_string(key)) { ok = false; return 0; }
                skip_ws();
                if (p >= e || *p != ':') { ok = false; return 0; }
                ++p;
                size_t v = parse_value(depth + 1);
                if (!ok) return 0;
                // json.loads: last duplicate key wins
                bool dup = false;
                for (auto& m : pool[idx].members)
                    if (m.key == key) { m.vidx = v; dup = true; break; }
                if (!dup) pool[idx].members.push_back({std::move(key), v});
                skip_ws();
                if (p < e && *p == ',') { ++p; continue; }
                if (p < e && *p == '}') { ++p; return idx; }
                ok = false;
                return 0;
            }
            return 0;
        }
        if (c == '[') {
            pool[idx].kind = Val::ARR;
            ++p;
            skip_ws();
            if (p < e && *p == ']') { ++p; return idx; }
            while (ok) {
                size_t v = parse_value(depth + 1);
                if (!ok) return 0;
                pool[idx].items.push_back(v);
                skip_ws();
                if (p < e && *p == ',') { ++p; continue; }
                if (p < e && *p == ']') { ++p; return idx; }
                ok = false;
                return 0;
            }
            return 0;
        }
        if (c == '"') {
            pool[idx].kind = Val::STR;
            if (!decode_string(pool[idx].str)) { ok = false; return 0; }
            return idx;
        }
        if (c == 't') {
            if (e - p < 4 || memcmp(p, "true", 4)) { ok = false; return 0; }
            pool[idx].kind = Val::TRUE_;
            p += 4;
            return idx;
        }
        if (c == 'f') {
            if (e - p < 5 || memcmp(p, "false", 5)) { ok = false; return 0; }
            pool[idx].kind = Val::FALSE_;
            p += 5;
            return idx;
        }
        if (c == 'n') {
            if (e - p < 4 || memcmp(p, "null", 4)) { ok = false; return 0; }
            pool[idx].kind = Val::NULL_;
            p += 4;
            return idx;
        }
        if (c == '-' || is_digit(c)) {
            pool[idx].kind = Val::INT;
            const uint8_t* s0 = p;
            if (*p == '-') ++p;
            if (p >= e || !is_digit(*p)) { ok = false; return 0; }
            while (p < e && is_digit(*p)) ++p;
            if (p < e && (*p == '.' || *p == 'e' || *p == 'E')) { ok = false; return 0; }  // float → punt
            // leading zeros: json.loads rejects 0123 → punt (stay exact)
            size_t dn = (size_t)(p - s0) - (s0[0] == '-' ? 1 : 0);
            const uint8_t* dp = s0 + (s0[0] == '-' ? 1 : 0);
            if (dn > 1 && dp[0] == '0') { ok = false; return 0; }
            // int canonical form == literal digits (python int() round-trip)
            // except "-0" → "0"
            if (dn == 1 && dp[0] == '0' && s0[0] == '-')
                pool[idx].str = "0";
            else
                pool[idx].str.assign((const char*)s0, (size_t)(p - s0));
            return idx;
        }
        ok = false;
        return 0;
    }
};

void append_escaped(std::string& out, const std::string& s) {
    // json.dumps ensure_ascii on ASCII input: escape " \ and control chars
    for (char ch : s) {
        uint8_t c = (uint8_t)ch;
        switch (c) {
            case '"': out += "\\\""; break;
            case '\\': out += "\\\\"; break;
            case '\b': out += "\\b"; break;
            case '\f': out += "\\f"; break;
            case '\n': out += "\\n"; break;
            case '\r': out += "\\r"; break;
            case '\t': out += "\\t"; break;
            default:
                if (c < 0x20) {
                    char b[8];
                    snprintf(b, sizeof(b), "\\u%04x", c);
                    out += b;
                } else {
                    out += (char)c;
                }
        }
    }
}

// sorted=true → json.dumps(..., sort_keys=True) (the _text_of/scan form);
// sorted=false → json.dumps(...) wire/insertion order (the dispatch form,
// matching the CPU chain's serialization of the parsed dict)
void serialize(const std::vector<Val>& pool, size_t idx, std::string& out, bool sorted) {
    const Val& v = pool[idx];
    switch (v.kind) {
        case Val::OBJ: {
            std::vector<const Member*> ms;
            ms.reserve(v.members.size());
            for (auto& m : v.members) ms.push_back(&m);
            if (sorted)  // ASCII keys sort identically by bytes
                std::sort(ms.begin(), ms.end(),
                          [](const Member* a, const Member* b) { return a->key < b->key; });
            out += '{';
            for (size_t k = 0; k < ms.size(); ++k) {
                if (k) out += ',';
                out += '"';
                append_escaped(out, ms[k]->key);
                out += "\":";
                serialize(pool, ms[k]->vidx, out, sorted);
            }
            out += '}';
            break;
        }
        case Val::ARR:
            out += '[';
            for (size_t k = 0; k < v.items.size(); ++k) {
                if (k) out += ',';
                serialize(pool, v.items[k], out, sorted);
            }
            out += ']';
            break;
        case Val::STR:
            out += '"';
            append_escaped(out, v.str);
            out += '"';
            break;
        case Val::INT: out += v.str; break;
        case Val::FLT: out += v.str; break;
        case Val::TRUE_: out += "true"; break;
        case Val::FALSE_: out += "false"; break;
        case Val::NULL_: out += "null"; break;
    }
}

// ---------------------------------------------------------------- TOON
// Exact port of plugins/toon.py encode(): unquoted simple tokens,
// `field[N]: a,b,c` primitive arrays, columnar `[N]{f1,f2}:` blocks for
// uniform object arrays, `- ` items for mixed arrays, 2-space indents.

inline bool toon_simple(const std::string& s) {  // _SIMPLE: ^[A-Za-z0-9_.@+\-]+\Z
    if (s.empty()) return false;
    for (char ch : s) {
        uint8_t c = (uint8_t)ch;
        if (!(is_alpha(c) || is_digit(c) || c == '_' || c == '.' || c == '@' || c == '+' || c == '-'))
            return false;
    }
    return true;
}

inline bool toon_numeric(const std::string& s) {  // _NUMERIC: ^-?\d+(\.\d+)?([eE][+-]?\d+)?\Z
    size_t i = 0, n = s.size();
    if (i < n && s[i] == '-') ++i;
    size_t d0 = i;
    while (i < n && is_digit((uint8_t)s[i])) ++i;
    if (i == d0) return false;
    if (i < n && s[i] == '.') {
        ++i;
        size_t f0 = i;
        while (i < n && is_digit((uint8_t)s[i])) ++i;
        if (i == f0) return false;
    }
    if (i < n && (s[i] == 'e' || s[i] == 'E')) {
        ++i;
        if (i < n && (s[i] == '+' || s[i] == '-')) ++i;
        size_t e0 = i;
        while (i < n && is_digit((uint8_t)s[i])) ++i;
        if (i == e0) return false;
    }
    return i == n;
}

// _scalar for a python str: quote (json.dumps) when ambiguous
void toon_scalar_str(const std::string& s, std::string& out) {
    if (s.empty() || !toon_simple(s) || s == "null" || s == "true" || s == "false" ||
        toon_numeric(s)) {
        out += '"';
        append_escaped(out, s);
        out += '"';
    } else {
        out += s;
    }
}

inline bool toon_is_scalar(const Val& v) { return v.kind != Val::OBJ && v.kind != Val::ARR; }

void toon_scalar(const std::vector<Val>& pool, size_t vi, std::string& out) {
    const Val& v = pool[vi];
    switch (v.kind) {
        case Val::NULL_: out += "null"; return;
        case Val::TRUE_: out += "true"; return;
        case Val::FALSE_: out += "false"; return;
        case Val::INT: out += v.str; return;   // json.dumps(int) == literal digits
        case Val::FLT: out += v.str; return;
        case Val::STR: toon_scalar_str(v.str, out); return;
        default: return;  // containers handled by the caller
    }
}

// fields if arr is a non-empty list of flat dicts with identical scalar
// keys, all _SIMPLE (they ride unquoted inside the `{a,b}` header)
bool toon_uniform(const std::vector<Val>& pool, const Val& arr, std::vector<std::string>& keys) {
    if (arr.items.empty()) return false;
    for (size_t it : arr.items) {
        const Val& x = pool[it];
        if (x.kind != Val::OBJ || x.members.empty()) return false;
    }
    keys.clear();
    for (auto& m : pool[arr.items[0]].members) keys.push_back(m.key);
    std::sort(keys.begin(), keys.end());
    for (auto& k : keys)
        if (!toon_simple(k)) return false;
    std::vector<std::string> ks;
    for (size_t it : arr.items) {
        const Val& x = pool[it];
        ks.clear();
        for (auto& m : x.members) {
            ks.push_back(m.key);
            if (!toon_is_scalar(pool[m.vidx])) return false;
        }
        std::sort(ks.begin(), ks.end());
        if (ks != keys) return false;
    }
    return true;
}

void toon_encode_value(const std::vector<Val>& pool, const std::string* key, size_t vi,
                       int indent, std::vector<std::string>& out);

void toon_encode_container(const std::vector<Val>& pool, size_t vi, int indent,
                           std::vector<std::string>& out) {
    const Val& v = pool[vi];
    if (v.kind == Val::OBJ) {
        for (auto& m : v.members) toon_encode_value(pool, &m.key, m.vidx, indent, out);
    } else {
        toon_encode_value(pool, nullptr, vi, indent, out);
    }
}

void toon_encode_value(const std::vector<Val>& pool, const std::string* key, size_t vi,
                       int indent, std::vector<std::string>& out) {
    std::string pad((size_t)indent * 2, ' ');
    std::string label;
    if (key) toon_scalar_str(*key, label);
    const Val& v = pool[vi];

    if (toon_is_scalar(v)) {
        std::string line = pad;
        if (key) { line += label; line += ": "; }
        toon_scalar(pool, vi, line);
        out.push_back(std::move(line));
        return;
    }

    if (v.kind == Val::ARR) {
        std::vector<std::string> fields;
        if (toon_uniform(pool, v, fields)) {
            std::string head = pad;
            if (key) head += label;
            head += "[" + std::to_string(v.items.size()) + "]{";
            for (size_t k = 0; k < fields.size(); ++k) {
                if (k) head += ',';
                head += fields[k];
            }
            head += "}:";
            out.push_back(std::move(head));
            for (size_t it : v.items) {
                std::string line = pad + "  ";
                const Val& x = pool[it];
                for (size_t k = 0; k < fields.size(); ++k) {
                    if (k) line += ',';
                    for (auto& m : x.members)
                        if (m.key == fields[k]) { toon_scalar(pool, m.vidx, line); break; }
                }
                out.push_back(std::move(line));
            }
            return;
        }
        bool all_sc = true;
        for (size_t it : v.items)
            if (!toon_is_scalar(pool[it])) { all_sc = false; break; }
        if (all_sc) {  // includes the empty array (trailing space like python)
            std::string line = pad;
            if (key) line += label;
            line += "[" + std::to_string(v.items.size()) + "]: ";
            for (size_t k = 0; k < v.items.size(); ++k) {
                if (k) line += ',';
                toon_scalar(pool, v.items[k], line);
            }
            out.push_back(std::move(line));
            return;
        }
        std::string head = pad;
        if (key) head += label;
        head += "[" + std::to_string(v.items.size()) + "]:";
        out.push_back(std::move(head));
        for (size_t it : v.items) {
            if (toon_is_scalar(pool[it])) {
                std::string line = pad + "  - ";
                toon_scalar(pool, it, line);
                out.push_back(std::move(line));
            } else {
                out.push_back(pad + "  -");
                toon_encode_container(pool, it, indent + 2, out);
            }
        }
        return;
    }

    // OBJ
    if (key) {
        out.push_back(pad + label + ":");
        toon_encode_container(pool, vi, indent + 1, out);
    } else {
        toon_encode_container(pool, vi, indent, out);
    }
}

std::string toon_encode(const std::vector<Val>& pool, size_t root) {
    std::vector<std::string> lines;
    toon_encode_value(pool, nullptr, root, 0, lines);
    std::string s;
    for (size_t i = 0; i < lines.size(); ++i) {
        if (i) s += '\n';
        s += lines[i];
    }
    return s;
}

// repr(round(frac, 4)) for the savings field. glibc %.4f rounds the exact
// binary value to nearest (ties cannot occur: k.00005 decimals have no
// finite binary expansion), matching python round(); stripping trailing
// zeros (keeping one fractional digit) reproduces float.__repr__'s
// shortest round-trip form for 4-decimal values.
std::string py_float4(double frac) {
    char b[64];
    snprintf(b, sizeof(b), "%.4f", frac);
    std::string s(b);
    size_t dot = s.find('.');
    size_t last = s.size();
    while (last - 1 > dot + 1 && s[last - 1] == '0') --last;
    return s.substr(0, last);
}

}  // namespace

// row status
enum : int32_t { RW_DONE = 0, RW_PUNT = 1, RW_BLOCKED = 2, RW_BADJSON = 3, RW_DENY = 4 };

// Rewrite a batch of flagged rows. Per row:
//   do_flags bit0 = apply normalizer, bit1 = apply pii, bit2 = deny check
//   (regex rows must not be passed here — caller punts them)
//   pii_want  = per-row GPU accept bits (mask_text_subset gating)
// The deny check (bit2) replicates DenyFilterPlugin@10 exactly: a
// (case-insensitive) substring search of each deny word over the SORTED
// serialization of the PRE-rewrite decoded payload (_text_of) — this is
// how escape-hidden words that the raw-byte scan cannot see are caught.
// Outputs: status[], found_bits[] (pii categories), deny_hit[] (first
// matching deny-word index, -1 none), canonical rewritten args in the
// arena (grow-retry contract like forge_decide).
static void rewrite_rows_range(
    const uint8_t* blob, const int32_t* args_beg, const int32_t* args_end,
    int r0, int r1,
    const uint8_t* do_flags, const uint32_t* pii_want,
    uint32_t pii_active_mask, int pii_mode,
    int norm_collapse, int norm_strip,
    const uint8_t* deny_blob, const int32_t* deny_off, int n_deny, int deny_ci,
    int32_t* status, uint32_t* found_bits, int32_t* deny_hit,
    std::string& buf,
    int64_t* out_beg, int64_t* out_end,
    int64_t* scan_beg, int64_t* scan_end,
    const uint8_t* harm_blob, const int32_t* harm_off, int n_harm,
    int32_t* harm_out,
    const uint8_t* sk_blob, const int32_t* sk_beg, const int32_t* sk_end,
    const int8_t* sk_type, const uint8_t* sk_req,
    const int32_t* sk_lo, const int32_t* sk_hi, uint8_t* schema_out)
{
    buf.reserve((size_t)(r1 - r0) * 64);
    for (int i = r0; i < r1; ++i) {
        status[i] = RW_PUNT;
        found_bits[i] = 0;
        deny_hit[i] = -1;
        if (harm_out) harm_out[i] = -1;
        if (schema_out) schema_out[i] = 2;
        out_beg[i] = out_end[i] = -1;
        scan_beg[i] = scan_end[i] = -1;
        const uint8_t* b = blob + args_beg[i];
        const uint8_t* e = blob + args_end[i];
        if (e < b) continue;
        // pure-ASCII precheck (NFC identity + ensure_ascii identity)
        bool ascii = true;
        for (const uint8_t* q = b; q < e; ++q)
            if (*q >= 0x80) { ascii = false; break; }
        if (!ascii) continue;  // PUNT
        std::vector<Val> pool;
        pool.reserve(32);
        Parser ps{b, e, pool};
        ps.skip_ws();
        if (ps.p >= ps.e) { status[i] = RW_BADJSON; continue; }
        size_t root = ps.parse_value(0);
        if (ps.ok) {
            ps.skip_ws();
            if (ps.p != ps.e) ps.ok = false;
        }
        if (!ps.ok) continue;  // PUNT (floats, deep nesting, bad escapes…)
        uint8_t fl = do_flags[i];
        if ((fl & 4) && n_deny > 0) {
            // DenyFilterPlugin semantics: substring over the sorted
            // serialization of the PRE-rewrite decoded payload
            std::string hay;
            serialize(pool, root, hay, /*sorted=*/true);
            if (deny_ci)
                for (auto& ch : hay)
                    if (ch >= 'A' && ch <= 'Z') ch += 32;
            int hitw = -1;
            for (int w = 0; w < n_deny && hitw < 0; ++w) {
                size_t wn = (size_t)(deny_off[w + 1] - deny_off[w]);
                if (wn == 0) continue;
                if (memmem(hay.data(), hay.size(), deny_blob + deny_off[w], wn) != nullptr)
                    hitw = w;
            }
            if (hitw >= 0) {
                deny_hit[i] = hitw;
                status[i] = RW_DENY;
                continue;
            }
        }
        // walk strings in insertion order (json.loads dict order == wire
        // order; _walk_strings visits values in that order — ordering only
        // matters for found-category accumulation, which is a set)
        uint32_t found = 0;
        for (auto& v : pool) {
            if (v.kind != Val::STR) continue;
            if (fl & 1) norm_ascii(v.str, norm_collapse != 0, norm_strip != 0);
            if (fl & 2) {
                if (pii_mode == 0) {
                    found |= pii_mask_text(v.str, pii_active_mask, pii_want[i]);
                } else {
                    // block/audit: python computes the substitution then
                    // discards it — found must reflect substitution-order
                    // matching, the text must stay unchanged
                    std::string tmp = v.str;
                    found |= pii_mask_text(tmp, pii_active_mask, pii_want[i]);
                }
            }
        }
        found_bits[i] = found;
        if (found && pii_mode == 1) {
            status[i] = RW_BLOCKED;
            continue;
        }
        std::string wire, sorted_s;
        serialize(pool, root, wire, /*sorted=*/false);
        serialize(pool, root, sorted_s, /*sorted=*/true);
        if ((fl & 16) && schema_out && sk_lo && sk_lo[i] >= 0) {
            uint8_t ok = 1;
            if (pool[root].kind != Val::OBJ) {
                ok = 0;  // non-object arguments → exact python path decides
            } else {
                for (int q = sk_lo[i]; q < sk_hi[i] && ok; ++q) {
                    const char* kb = (const char*)sk_blob + sk_beg[q];
                    size_t kn = (size_t)(sk_end[q] - sk_beg[q]);
                    const Val* v = nullptr;
                    for (auto& mem : pool[root].members)
                        if (mem.key.size() == kn && memcmp(mem.key.data(), kb, kn) == 0) {
                            v = &pool[mem.vidx];
                            break;
                        }
                    if (v == nullptr) {
                        if (sk_req[q]) ok = 0;
                        continue;
                    }
                    switch (sk_type[q]) {
                        case 1: if (v->kind != Val::STR) ok = 0; break;
                        case 2: if (v->kind != Val::INT) ok = 0; break;
                        case 3: if (v->kind != Val::TRUE_ && v->kind != Val::FALSE_) ok = 0; break;
                        case 4: if (v->kind != Val::ARR) ok = 0; break;
                        case 5: if (v->kind != Val::OBJ) ok = 0; break;
                        case 6: if (v->kind != Val::NULL_) ok = 0; break;
                        default: break;
                    }
                }
            }
            schema_out[i] = ok;
        }
        if ((fl & 8) && n_harm > 0 && harm_out) {
            std::string hay2 = sorted_s;
            for (auto& ch : hay2)
                if (ch >= 'A' && ch <= 'Z') ch += 32;
            for (int w = 0; w < n_harm; ++w) {
                size_t wn = (size_t)(harm_off[w + 1] - harm_off[w]);
                if (wn == 0) continue;
                if (memmem(hay2.data(), hay2.size(), harm_blob + harm_off[w], wn) != nullptr) {
                    harm_out[i] = w;
                    break;
                }
            }
        }
        out_beg[i] = (int64_t)buf.size();
        buf += wire;
        out_end[i] = (int64_t)buf.size();
        if (sorted_s == wire) {  // common: already sorted / no objects
            scan_beg[i] = out_beg[i];
            scan_end[i] = out_end[i];
        } else {
            scan_beg[i] = (int64_t)buf.size();
            buf += sorted_s;
            scan_end[i] = (int64_t)buf.size();
        }
        status[i] = RW_DONE;
    }
}

// Rewrite a batch of flagged rows — threaded over row ranges (rows are
// independent; per-thread arenas are stitched and spans rebased).
extern "C" int64_t forge_rewrite_rows(
    const uint8_t* blob, const int32_t* args_beg, const int32_t* args_end, int n,
    const uint8_t* do_flags, const uint32_t* pii_want,
    uint32_t pii_active_mask, int pii_mode /*0 mask, 1 block, 2 audit*/,
    int norm_collapse, int norm_strip,
    const uint8_t* deny_blob, const int32_t* deny_off, int n_deny, int deny_ci,
    int32_t* status, uint32_t* found_bits, int32_t* deny_hit,
    uint8_t* arena, int64_t arena_cap,
    int64_t* out_beg, int64_t* out_end,        // dispatch form (wire key order)
    int64_t* scan_beg, int64_t* scan_end,      // scan form (sorted keys)
    // do_flags bit3: harmful_content@60 over the POST-rewrite sorted text
    // (phrase.lower() in _text_of(args).lower()). REPORTED, not a status —
    // the caller orders it against the moderation verdict exactly as the
    // CPU chain does (moderation first).
    const uint8_t* harm_blob, const int32_t* harm_off, int n_harm,
    int32_t* harm_out,
    // do_flags bit4: FAST-mode schema check on the post-rewrite tree
    // (flat object schemas: required presence + value types ONLY — richer
    // schemas never compile to "fast", gpu/pipeline._compile_tool_schema).
    // Entry table: key spans into sk_blob + type code (0 none, 1 string,
    // 2 integer, 3 boolean, 4 array, 5 object, 6 null) + required flag;
    // per-row entry range [sk_lo, sk_hi). schema_out: 1 pass, 0 FAIL
    // (caller reruns the exact python validator for the error message),
    // 2 not checked.
    const uint8_t* sk_blob, const int32_t* sk_beg, const int32_t* sk_end,
    const int8_t* sk_type, const uint8_t* sk_req,
    const int32_t* sk_lo, const int32_t* sk_hi, uint8_t* schema_out)
{
    int nthreads = n >= 512 ? 8 : (n >= 64 ? 4 : 1);
    std::vector<std::string> bufs((size_t)nthreads);
    int chunk = (n + nthreads - 1) / nthreads;
    auto run = [&](int t) {
        int r0 = t * chunk, r1 = r0 + chunk < n ? r0 + chunk : n;
        if (r0 >= r1) return;
        rewrite_rows_range(blob, args_beg, args_end, r0, r1, do_flags, pii_want,
                           pii_active_mask, pii_mode, norm_collapse, norm_strip,
                           deny_blob, deny_off, n_deny, deny_ci,
                           status, found_bits, deny_hit, bufs[(size_t)t],
                           out_beg, out_end, scan_beg, scan_end,
                           harm_blob, harm_off, n_harm, harm_out,
                           sk_blob, sk_beg, sk_end, sk_type, sk_req,
                           sk_lo, sk_hi, schema_out);
    };
    if (nthreads == 1) {
        run(0);
    } else {
        run_parallel(nthreads, run);
    }
    int64_t total = 0;
    std::vector<int64_t> base((size_t)nthreads, 0);
    for (int t = 0; t < nthreads; ++t) { base[(size_t)t] = total; total += (int64_t)bufs[(size_t)t].size(); }
    if (total > arena_cap) return -total;
    for (int t = 0; t < nthreads; ++t) {
        if (!bufs[(size_t)t].empty()) memcpy(arena + base[(size_t)t], bufs[(size_t)t].data(), bufs[(size_t)t].size());
        int r0 = t * chunk, r1 = r0 + chunk < n ? r0 + chunk : n;
        for (int r = r0; r < r1; ++r)
            if (out_beg[r] >= 0) {
                out_beg[r] += base[(size_t)t];
                out_end[r] += base[(size_t)t];
                scan_beg[r] += base[(size_t)t];
                scan_end[r] += base[(size_t)t];
            }
    }
    return total;
}

// Native result post chain: the _host_post hot path (gpu/pipeline.py) for
// rows inside the same provable-equivalence envelope as the rewrite lane.
// Per row, in the exact CPU-chain order: pii_filter@30 over every string
// of the parsed result → harmful_content@60 (case-insensitive substring
// over the SORTED compact serialization of the post-PII result) →
// toon_encoder@900 (structuredContent compression incl. the _meta.toon
// record) → wire-order re-serialization (json.dumps insertion order).
// Rows with user regexes, output schemas, or guard-length overflow must
// not be passed here (caller punts them); non-ASCII / float / bad-JSON
// results return RW_PUNT and take the Python path.
//   do_flags: bit0 pii, bit1 harm, bit2 toon
//   harm_blob/off: phrases, pre-lowercased ASCII, plugin order
//   is_err: truthiness of result["isError"] on the FINAL result
static void post_rows_range(
    const uint8_t* blob, const int64_t* res_beg, const int64_t* res_end,
    int r0, int r1,
    const uint8_t* do_flags,
    uint32_t pii_active_mask, int pii_mode,
    const uint8_t* harm_blob, const int32_t* harm_off, int n_harm,
    int64_t toon_min_size, double toon_min_savings,
    int32_t* status, uint32_t* found_bits, int32_t* harm_hit, uint8_t* is_err,
    std::string& buf, int64_t* out_beg, int64_t* out_end)
{
    buf.reserve((size_t)(r1 - r0) * 128);
    for (int i = r0; i < r1; ++i) {
        status[i] = RW_PUNT;
        found_bits[i] = 0;
        harm_hit[i] = -1;
        is_err[i] = 0;
        out_beg[i] = out_end[i] = -1;
        const uint8_t* b = blob + res_beg[i];
        const uint8_t* e = blob + res_end[i];
        if (e < b) continue;
        bool ascii = true;
        for (const uint8_t* q = b; q < e; ++q)
            if (*q >= 0x80) { ascii = false; break; }
        if (!ascii) continue;  // PUNT
        std::vector<Val> pool;
        pool.reserve(64);
        Parser ps{b, e, pool};
        ps.skip_ws();
        if (ps.p >= ps.e) { status[i] = RW_BADJSON; continue; }
        size_t root = ps.parse_value(0);
        if (ps.ok) {
            ps.skip_ws();
            if (ps.p != ps.e) ps.ok = false;
        }
        if (!ps.ok) continue;  // PUNT (floats, depth, escapes…)
        uint8_t fl = do_flags[i];

        // --- pii over every string of the result tree ---
        uint32_t found = 0;
        if (fl & 1) {
            for (auto& v : pool) {
                if (v.kind != Val::STR) continue;
                if (pii_mode == 0) {
                    found |= pii_mask_text(v.str, pii_active_mask, 0xFFFFFFFFu);
                } else {
                    std::string tmp = v.str;
                    found |= pii_mask_text(tmp, pii_active_mask, 0xFFFFFFFFu);
                }
            }
        }
        found_bits[i] = found;
        if (found && pii_mode == 1) { status[i] = RW_BLOCKED; continue; }

        // --- harm phrases over sorted serialization of the post-pii tree ---
        if ((fl & 2) && n_harm > 0) {
            std::string hay;
            serialize(pool, root, hay, /*sorted=*/true);
            for (auto& ch : hay)
                if (ch >= 'A' && ch <= 'Z') ch += 32;
            for (int w = 0; w < n_harm; ++w) {
                size_t wn = (size_t)(harm_off[w + 1] - harm_off[w]);
                if (wn == 0) continue;
                if (memmem(hay.data(), hay.size(), harm_blob + harm_off[w], wn) != nullptr) {
                    harm_hit[i] = w;
                    break;
                }
            }
            if (harm_hit[i] >= 0) { status[i] = RW_DENY; continue; }
        }

        // --- toon: structuredContent compression + _meta.toon record ---
        bool punt_row = false;
        if ((fl & 4) && pool[root].kind == Val::OBJ) {
            size_t sc = SIZE_MAX;
            for (auto& m : pool[root].members)
                if (m.key == "structuredContent") { sc = m.vidx; break; }
            if (sc != SIZE_MAX && pool[sc].kind != Val::NULL_) {
                std::string scj;
                serialize(pool, sc, scj, /*sorted=*/false);
                int64_t j = (int64_t)scj.size();
                if (j >= toon_min_size) {
                    std::string enc = toon_encode(pool, sc);
                    int64_t t = (int64_t)enc.size();
                    double frac = j ? 1.0 - (double)t / (double)j : 0.0;
                    if (frac >= toon_min_savings) {
                        // existing non-dict "_meta" → python (setdefault semantics)
                        size_t meta_i = SIZE_MAX;
                        for (auto& m : pool[root].members)
                            if (m.key == "_meta") {
                                meta_i = m.vidx;
                                if (pool[m.vidx].kind != Val::OBJ) punt_row = true;
                                break;
                            }
                        if (!punt_row) {
                            // NOTE: pool grows below — take indices, never refs
                            auto mk = [&pool](Val::Kind k) {
                                size_t ix = pool.size();
                                pool.emplace_back();
                                pool[ix].kind = k;
                                return ix;
                            };
                            size_t s_type = mk(Val::STR); pool[s_type].str = "text";
                            size_t s_text = mk(Val::STR); pool[s_text].str = std::move(enc);
                            size_t item = mk(Val::OBJ);
                            pool[item].members.push_back({"type", s_type});
                            pool[item].members.push_back({"text", s_text});
                            size_t carr = mk(Val::ARR);
                            pool[carr].items.push_back(item);
                            bool had = false;
                            for (auto& m : pool[root].members)
                                if (m.key == "content") { m.vidx = carr; had = true; break; }
                            if (!had) pool[root].members.push_back({"content", carr});
                            size_t jb = mk(Val::INT); pool[jb].str = std::to_string(j);
                            size_t tb = mk(Val::INT); pool[tb].str = std::to_string(t);
                            size_t sv = mk(Val::FLT); pool[sv].str = py_float4(frac);
                            size_t toonobj = mk(Val::OBJ);
                            pool[toonobj].members.push_back({"json_bytes", jb});
                            pool[toonobj].members.push_back({"toon_bytes", tb});
                            pool[toonobj].members.push_back({"savings", sv});
                            if (meta_i == SIZE_MAX) {
                                size_t mo = mk(Val::OBJ);
                                pool[mo].members.push_back({"toon", toonobj});
                                pool[root].members.push_back({"_meta", mo});
                            } else {
                                bool hadt = false;
                                for (auto& m : pool[meta_i].members)
                                    if (m.key == "toon") { m.vidx = toonobj; hadt = true; break; }
                                if (!hadt) pool[meta_i].members.push_back({"toon", toonobj});
                            }
                        }
                    }
                }
            }
        }
        if (punt_row) continue;  // PUNT

        // --- is_err + final wire serialization ---
        if (pool[root].kind == Val::OBJ) {
            for (auto& m : pool[root].members)
                if (m.key == "isError") {
                    const Val& x = pool[m.vidx];
                    bool err = false;
                    switch (x.kind) {
                        case Val::TRUE_: err = true; break;
                        case Val::INT: err = x.str != "0"; break;
                        case Val::STR: err = !x.str.empty(); break;
                        case Val::OBJ: err = !x.members.empty(); break;
                        case Val::ARR: err = !x.items.empty(); break;
                        default: err = false;
                    }
                    is_err[i] = err ? 1 : 0;
                    break;
                }
        }
        std::string wire;
        serialize(pool, root, wire, /*sorted=*/false);
        out_beg[i] = (int64_t)buf.size();
        buf += wire;
        out_end[i] = (int64_t)buf.size();
        status[i] = RW_DONE;
    }
}

extern "C" int64_t forge_post_rows(
    const uint8_t* blob, const int64_t* res_beg, const int64_t* res_end, int n,
    const uint8_t* do_flags,
    uint32_t pii_active_mask, int pii_mode /*0 mask, 1 block, 2 audit*/,
    const uint8_t* harm_blob, const int32_t* harm_off, int n_harm,
    int64_t toon_min_size, double toon_min_savings,
    int32_t* status, uint32_t* found_bits, int32_t* harm_hit, uint8_t* is_err,
    uint8_t* arena, int64_t arena_cap, int64_t* out_beg, int64_t* out_end)
{
    int nthreads = n >= 512 ? 8 : (n >= 64 ? 4 : 1);
    std::vector<std::string> bufs((size_t)nthreads);
    int chunk = (n + nthreads - 1) / nthreads;
    auto run = [&](int t) {
        int r0 = t * chunk, r1 = r0 + chunk < n ? r0 + chunk : n;
        if (r0 >= r1) return;
        post_rows_range(blob, res_beg, res_end, r0, r1, do_flags,
                        pii_active_mask, pii_mode, harm_blob, harm_off, n_harm,
                        toon_min_size, toon_min_savings,
                        status, found_bits, harm_hit, is_err,
                        bufs[(size_t)t], out_beg, out_end);
    };
    if (nthreads == 1) {
        run(0);
    } else {
        run_parallel(nthreads, run);
    }
    int64_t total = 0;
    std::vector<int64_t> base((size_t)nthreads, 0);
    for (int t = 0; t < nthreads; ++t) { base[(size_t)t] = total; total += (int64_t)bufs[(size_t)t].size(); }
    if (total > arena_cap) return -total;
    for (int t = 0; t < nthreads; ++t) {
        if (!bufs[(size_t)t].empty()) memcpy(arena + base[(size_t)t], bufs[(size_t)t].data(), bufs[(size_t)t].size());
        int r0 = t * chunk, r1 = r0 + chunk < n ? r0 + chunk : n;
        for (int r = r0; r < r1; ++r)
            if (out_beg[r] >= 0) {
                out_beg[r] += base[(size_t)t];
                out_end[r] += base[(size_t)t];
            }
    }
    return total;
}
