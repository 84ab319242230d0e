#include "json.hpp"

#include <cmath>
#include <cstdio>
#include <cstring>

namespace bamd {

namespace {

struct Parser {
    const char* p;
    const char* end;
    int depth = 0;

    bool eof() const { return p >= end; }
    void skipWs() {
        while (p < end &&
               (*p == ' ' || *p == '\t' || *p == '\n' || *p == '\r'))
            ++p;
    }
    bool lit(const char* s, size_t n) {
        if ((size_t)(end - p) < n || memcmp(p, s, n) != 0) return false;
        p += n;
        return true;
    }

    bool parseValue(Json& out) {
        if (++depth > 128) return false;
        skipWs();
        if (eof()) return false;
        bool ok;
        switch (*p) {
        case '{': ok = parseObject(out); break;
        case '[': ok = parseArray(out); break;
        case '"': {
            std::string s;
            ok = parseString(s);
            if (ok) out = Json(std::move(s));
            break;
        }
        case 't': ok = lit("true", 4); if (ok) out = Json(true); break;
        case 'f': ok = lit("false", 5); if (ok) out = Json(false); break;
        case 'n': ok = lit("null", 4); if (ok) out = Json(nullptr); break;
        default: ok = parseNumber(out); break;
        }
        --depth;
        return ok;
    }

    bool parseObject(Json& out) {
        ++p;  // '{'
        JsonObject obj;
        skipWs();
        if (!eof() && *p == '}') {
            ++p;
            out = Json(std::move(obj));
            return true;
        }
        while (true) {
            skipWs();
            if (eof() || *p != '"') return false;
            std::string key;
            if (!parseString(key)) return false;
            skipWs();
            if (eof() || *p != ':') return false;
            ++p;
            Json v;
            if (!parseValue(v)) return false;
            obj[std::move(key)] = std::move(v);
            skipWs();
            if (eof()) return false;
            if (*p == ',') { ++p; continue; }
            if (*p == '}') { ++p; out = Json(std::move(obj)); return true; }
            return false;
        }
    }

    bool parseArray(Json& out) {
        ++p;  // '['
        JsonArray arr;
        skipWs();
        if (!eof() && *p == ']') {
            ++p;
            out = Json(std::move(arr));
            return true;
        }
        while (true) {
            Json v;
            if (!parseValue(v)) return false;
            arr.push_back(std::move(v));
            skipWs();
            if (eof()) return false;
            if (*p == ',') { ++p; continue; }
            if (*p == ']') { ++p; out = Json(std::move(arr)); return true; }
            return false;
        }
    }

    static void appendUtf8(std::string& s, uint32_t cp) {
        if (cp < 0x80) {
            s.push_back((char)cp);
        } else if (cp < 0x800) {
            s.push_back((char)(0xC0 | (cp >> 6)));
            s.push_back((char)(0x80 | (cp & 0x3F)));
        } else if (cp < 0x10000) {
            s.push_back((char)(0xE0 | (cp >> 12)));
            s.push_back((char)(0x80 | ((cp >> 6) & 0x3F)));
            s.push_back((char)(0x80 | (cp & 0x3F)));
        } else {
            s.push_back((char)(0xF0 | (cp >> 18)));
            s.push_back((char)(0x80 | ((cp >> 12) & 0x3F)));
            s.push_back((char)(0x80 | ((cp >> 6) & 0x3F)));
            s.push_back((char)(0x80 | (cp & 0x3F)));
        }
    }

    bool hex4(uint32_t& out) {
        if (end - p < 4) return false;
        out = 0;
        for (int i = 0; i < 4; ++i) {
            char c = *p++;
            out <<= 4;
            if (c >= '0' && c <= '9') out |= (uint32_t)(c - '0');
            else if (c >= 'a' && c <= 'f') out |= (uint32_t)(c - 'a' + 10);
            else if (c >= 'A' && c <= 'F') out |= (uint32_t)(c - 'A' + 10);
            else return false;
        }
        return true;
    }

    bool parseString(std::string& out) {
        ++p;  // '"'
        while (p < end) {
            unsigned char c = (unsigned char)*p;
            if (c == '"') { ++p; return true; }
            if (c == '\\') {
                ++p;
                if (eof()) return false;
                char e = *p++;
                switch (e) {
                case '"': out.push_back('"'); break;
                case '\\': out.push_back('\\'); break;
                case '/': out.push_back('/'); break;
                case 'b': out.push_back('\b'); break;
                case 'f': out.push_back('\f'); break;
                case 'n': out.push_back('\n'); break;
                case 'r': out.push_back('\r'); break;
                case 't': out.push_back('\t'); break;
                case 'u': {
                    uint32_t cp;
                    if (!hex4(cp)) return false;
                    if (cp >= 0xD800 && cp <= 0xDBFF) {
                        // surrogate pair
                        if (end - p >= 6 && p[0] == '\\' && p[1] == 'u') {
                            p += 2;
                            uint32_t lo;
                            if (!hex4(lo)) return false;
                            if (lo >= 0xDC00 && lo <= 0xDFFF) {
                                cp = 0x10000 + ((cp - 0xD800) << 10) +
                                     (lo - 0xDC00);
                            } else {
                                appendUtf8(out, 0xFFFD);
                                cp = 0xFFFD;
                                appendUtf8(out, cp);
                                break;
                            }
                        } else {
                            cp = 0xFFFD;
                        }
                    }
                    appendUtf8(out, cp);
                    break;
                }
                default: return false;
                }
            } else if (c < 0x20) {
                return false;
            } else {
                out.push_back((char)c);
                ++p;
            }
        }
        return false;
    }

    bool parseNumber(Json& out) {
        const char* start = p;
        if (p < end && *p == '-') ++p;
        while (p < end && *p >= '0' && *p <= '9') ++p;
        bool isInt = true;
        if (p < end && *p == '.') {
            isInt = false;
            ++p;
            while (p < end && *p >= '0' && *p <= '9') ++p;
        }
        if (p < end && (*p == 'e' || *p == 'E')) {
            isInt = false;
            ++p;
            if (p < end && (*p == '+' || *p == '-')) ++p;
            while (p < end && *p >= '0' && *p <= '9') ++p;
        }
        if (p == start || (p == start + 1 && *start == '-')) return false;
        std::string tok(start, (size_t)(p - start));
        if (isInt) {
            errno = 0;
            char* endp = nullptr;
            long long v = strtoll(tok.c_str(), &endp, 10);
            if (errno == 0 && endp && *endp == '\0') {
                out = Json((int64_t)v);
                return true;
            }
            // fall through to double on overflow
        }
        char* endp = nullptr;
        double d = strtod(tok.c_str(), &endp);
        if (!endp || *endp != '\0') return false;
        out = Json(d);
        return true;
    }
};

}  // namespace

std::optional<Json> Json::parse(std::string_view text) {
    Parser ps{text.data(), text.data() + text.size()};
    Json v;
    if (!ps.parseValue(v)) return std::nullopt;
    ps.skipWs();
    if (!ps.eof()) return std::nullopt;
    return v;
}

void jsonEscape(std::string_view in, std::string& out) {
    out.push_back('"');
    for (unsigned char c : in) {
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
                char buf[8];
                snprintf(buf, sizeof(buf), "\\u%04x", c);
                out += buf;
            } else {
                out.push_back((char)c);
            }
        }
    }
    out.push_back('"');
}

void Json::dumpTo(std::string& out) const {
    switch (type_) {
    case Type::Null: out += "null"; break;
    case Type::Bool: out += bool_ ? "true" : "false"; break;
    case Type::Int: {
        char buf[24];
        snprintf(buf, sizeof(buf), "%lld", (long long)int_);
        out += buf;
        break;
    }
    case Type::Double: {
        if (std::isfinite(dbl_)) {
            char buf[32];
            snprintf(buf, sizeof(buf), "%.17g", dbl_);
            out += buf;
        } else {
            out += "null";
        }
        break;
    }
    case Type::String: jsonEscape(str_, out); break;
    case Type::Array: {
        out.push_back('[');
        bool first = true;
        for (const auto& v : *arr_) {
            if (!first) out.push_back(',');
            first = false;
            v.dumpTo(out);
        }
        out.push_back(']');
        break;
    }
    case Type::Object: {
        out.push_back('{');
        bool first = true;
        for (const auto& [k, v] : *obj_) {
            if (!first) out.push_back(',');
            first = false;
            jsonEscape(k, out);
            out.push_back(':');
            v.dumpTo(out);
        }
        out.push_back('}');
        break;
    }
    }
}

std::string Json::dump() const {
    std::string out;
    out.reserve(64);
    dumpTo(out);
    return out;
}

}  // namespace bamd
