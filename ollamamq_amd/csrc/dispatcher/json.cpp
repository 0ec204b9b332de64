#include "json.h"

#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <cstring>

namespace omq {

// ------------------------------------------------------------- serialize
static void esc(const std::string& s, std::string& out) {
    out += '"';
    for (unsigned char c : s) {
        switch (c) {
            case '"': out += "\\\""; break;
            case '\\': out += "\\\\"; break;
            case '\n': out += "\\n"; break;
            case '\r': out += "\\r"; break;
            case '\t': out += "\\t"; break;
            default:
                if (c < 0x20) {
                    char buf[8];
                    snprintf(buf, sizeof buf, "\\u%04x", c);
                    out += buf;
                } else {
                    out += (char)c;
                }
        }
    }
    out += '"';
}

static void dump_to(const Json& j, std::string& out) {
    switch (j.type) {
        case Json::Type::Null: out += "null"; break;
        case Json::Type::Bool: out += j.b ? "true" : "false"; break;
        case Json::Type::Num: {
            char buf[40];
            if (std::floor(j.num) == j.num && std::fabs(j.num) < 1e15) {
                snprintf(buf, sizeof buf, "%lld", (long long)j.num);
            } else {
                // shortest representation that round-trips the double
                // (plain %g keeps 6 digits and corrupts seeds/timestamps)
                snprintf(buf, sizeof buf, "%.15g", j.num);
                if (strtod(buf, nullptr) != j.num)
                    snprintf(buf, sizeof buf, "%.17g", j.num);
            }
            out += buf;
            break;
        }
        case Json::Type::Str: esc(j.str, out); break;
        case Json::Type::Arr: {
            out += '[';
            for (size_t i = 0; i < j.arr.size(); i++) {
                if (i) out += ',';
                dump_to(j.arr[i], out);
            }
            out += ']';
            break;
        }
        case Json::Type::Obj: {
            out += '{';
            bool first = true;
            for (const auto& [k, v] : j.obj) {
                if (!first) out += ',';
                first = false;
                esc(k, out);
                out += ':';
                dump_to(v, out);
            }
            out += '}';
            break;
        }
    }
}

std::string Json::dump() const {
    std::string out;
    dump_to(*this, out);
    return out;
}

// ----------------------------------------------------------------- parse
namespace {
struct Parser {
    const char* p;
    const char* end;
    int depth = 0;

    void ws() {
        while (p < end && (*p == ' ' || *p == '\t' || *p == '\n' || *p == '\r'))
            p++;
    }
    bool lit(const char* s, size_t n) {
        if ((size_t)(end - p) < n || memcmp(p, s, n) != 0) return false;
        p += n;
        return true;
    }
    bool parse_string(std::string& out) {
        if (p >= end || *p != '"') return false;
        p++;
        out.clear();
        while (p < end && *p != '"') {
            if (*p == '\\') {
                p++;
                if (p >= end) return false;
                switch (*p) {
                    case '"': out += '"'; break;
                    case '\\': out += '\\'; break;
                    case '/': out += '/'; break;
                    case 'b': out += '\b'; break;
                    case 'f': out += '\f'; break;
                    case 'n': out += '\n'; break;
                    case 'r': out += '\r'; break;
                    case 't': out += '\t'; break;
                    case 'u': {
                        auto hex4 = [](const char* q, unsigned* v) -> bool {
                            unsigned x = 0;
                            for (int i = 0; i < 4; i++) {
                                char c = q[i];
                                x <<= 4;
                                if (c >= '0' && c <= '9') x |= c - '0';
                                else if (c >= 'a' && c <= 'f')
                                    x |= c - 'a' + 10;
                                else if (c >= 'A' && c <= 'F')
                                    x |= c - 'A' + 10;
                                else return false;
                            }
                            *v = x;
                            return true;
                        };
                        if (end - p < 5) return false;
                        unsigned cp;
                        if (!hex4(p + 1, &cp)) return false;
                        p += 4;
                        if (cp >= 0xD800 && cp <= 0xDBFF) {
                            // UTF-16 surrogate pair (how JSON escapes
                            // astral chars, e.g. emoji from Python's
                            // json.dumps); unpaired halves fold to U+FFFD
                            unsigned lo;
                            if (end - p >= 7 && p[1] == '\\' &&
                                p[2] == 'u' && hex4(p + 3, &lo) &&
                                lo >= 0xDC00 && lo <= 0xDFFF) {
                                cp = 0x10000 + ((cp - 0xD800) << 10) +
                                     (lo - 0xDC00);
                                p += 6;
                            } else {
                                cp = 0xFFFD;
                            }
                        } else if (cp >= 0xDC00 && cp <= 0xDFFF) {
                            cp = 0xFFFD;  // lone low surrogate
                        }
                        if (cp < 0x80) out += (char)cp;
                        else if (cp < 0x800) {
                            out += (char)(0xC0 | (cp >> 6));
                            out += (char)(0x80 | (cp & 0x3F));
                        } else if (cp < 0x10000) {
                            out += (char)(0xE0 | (cp >> 12));
                            out += (char)(0x80 | ((cp >> 6) & 0x3F));
                            out += (char)(0x80 | (cp & 0x3F));
                        } else {
                            out += (char)(0xF0 | (cp >> 18));
                            out += (char)(0x80 | ((cp >> 12) & 0x3F));
                            out += (char)(0x80 | ((cp >> 6) & 0x3F));
                            out += (char)(0x80 | (cp & 0x3F));
                        }
                        break;
                    }
                    default: return false;
                }
                p++;
            } else {
                out += *p++;
            }
        }
        if (p >= end) return false;
        p++;  // closing quote
        return true;
    }
    bool value(Json& out) {
        if (++depth > 256) return false;
        ws();
        if (p >= end) return false;
        bool ok = false;
        if (*p == '{') {
            p++;
            out = Json::object();
            ws();
            if (p < end && *p == '}') { p++; ok = true; }
            else {
                while (true) {
                    std::string key;
                    ws();
                    if (!parse_string(key)) break;
                    ws();
                    if (p >= end || *p != ':') break;
                    p++;
                    Json v;
                    if (!value(v)) break;
                    out.obj.emplace_back(std::move(key), std::move(v));
                    ws();
                    if (p < end && *p == ',') { p++; continue; }
                    if (p < end && *p == '}') { p++; ok = true; }
                    break;
                }
            }
        } else if (*p == '[') {
            p++;
            out = Json::array();
            ws();
            if (p < end && *p == ']') { p++; ok = true; }
            else {
                while (true) {
                    Json v;
                    if (!value(v)) break;
                    out.arr.push_back(std::move(v));
                    ws();
                    if (p < end && *p == ',') { p++; continue; }
                    if (p < end && *p == ']') { p++; ok = true; }
                    break;
                }
            }
        } else if (*p == '"') {
            out = Json::string("");
            ok = parse_string(out.str);
        } else if (lit("true", 4)) {
            out = Json::boolean(true);
            ok = true;
        } else if (lit("false", 5)) {
            out = Json::boolean(false);
            ok = true;
        } else if (lit("null", 4)) {
            out = Json::null();
            ok = true;
        } else {
            char* endp = nullptr;
            double d = strtod(p, &endp);
            if (endp != p && endp <= end) {
                out = Json::number(d);
                p = endp;
                ok = true;
            }
        }
        depth--;
        return ok;
    }
};
}  // namespace

std::optional<Json> Json::parse(const std::string& text) {
    Parser ps{text.data(), text.data() + text.size()};
    Json out;
    if (!ps.value(out)) return std::nullopt;
    ps.ws();
    if (ps.p != ps.end) return std::nullopt;  // trailing garbage
    return out;
}

}  // namespace omq
