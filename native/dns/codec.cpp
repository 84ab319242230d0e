#include "codec.hpp"

#include <arpa/inet.h>

#include <cstdio>
#include <cstring>
#include <map>

namespace bamd::dns {

const char* typeName(uint16_t t) {
    switch (t) {
    case TYPE_A: return "A";
    case TYPE_NS: return "NS";
    case TYPE_CNAME: return "CNAME";
    case TYPE_SOA: return "SOA";
    case TYPE_PTR: return "PTR";
    case TYPE_MX: return "MX";
    case TYPE_TXT: return "TXT";
    case TYPE_AAAA: return "AAAA";
    case TYPE_SRV: return "SRV";
    case TYPE_OPT: return "OPT";
    case TYPE_ANY: return "ANY";
    default: {
        static thread_local char buf[16];
        snprintf(buf, sizeof(buf), "TYPE%u", t);
        return buf;
    }
    }
}

uint16_t typeFromName(const std::string& n) {
    if (n == "A") return TYPE_A;
    if (n == "NS") return TYPE_NS;
    if (n == "CNAME") return TYPE_CNAME;
    if (n == "SOA") return TYPE_SOA;
    if (n == "PTR") return TYPE_PTR;
    if (n == "MX") return TYPE_MX;
    if (n == "TXT") return TYPE_TXT;
    if (n == "AAAA") return TYPE_AAAA;
    if (n == "SRV") return TYPE_SRV;
    if (n == "OPT") return TYPE_OPT;
    if (n == "ANY") return TYPE_ANY;
    return 0;
}

const char* rcodeName(uint8_t rc) {
    switch (rc) {
    case RCODE_NOERROR: return "NOERROR";
    case RCODE_FORMERR: return "FORMERR";
    case RCODE_SERVFAIL: return "SERVFAIL";
    case RCODE_NXDOMAIN: return "NXDOMAIN";
    case RCODE_NOTIMP: return "NOTIMP";
    case RCODE_REFUSED: return "REFUSED";
    default: {
        static thread_local char buf[16];
        snprintf(buf, sizeof(buf), "RCODE%u", rc);
        return buf;
    }
    }
}

void toLowerAscii(std::string& s) {
    for (char& c : s)
        if (c >= 'A' && c <= 'Z') c += 32;
}

Record Record::A(std::string name, const std::string& ipv4, uint32_t ttl) {
    Record r;
    r.name = std::move(name);
    r.type = TYPE_A;
    r.ttl = ttl;
    inet_pton(AF_INET, ipv4.c_str(), r.a.data());
    return r;
}

Record Record::AAAA(std::string name, const std::string& ipv6, uint32_t ttl) {
    Record r;
    r.name = std::move(name);
    r.type = TYPE_AAAA;
    r.ttl = ttl;
    inet_pton(AF_INET6, ipv6.c_str(), r.aaaa.data());
    return r;
}

Record Record::SRV(std::string name, std::string tgt, uint16_t port,
                   uint32_t ttl, uint16_t prio, uint16_t weight) {
    Record r;
    r.name = std::move(name);
    r.type = TYPE_SRV;
    r.ttl = ttl;
    r.target = std::move(tgt);
    r.port = port;
    r.priority = prio;
    r.weight = weight;
    return r;
}

Record Record::PTR(std::string name, std::string tgt, uint32_t ttl) {
    Record r;
    r.name = std::move(name);
    r.type = TYPE_PTR;
    r.ttl = ttl;
    r.target = std::move(tgt);
    return r;
}

Record Record::CNAME(std::string name, std::string tgt, uint32_t ttl) {
    Record r;
    r.name = std::move(name);
    r.type = TYPE_CNAME;
    r.ttl = ttl;
    r.target = std::move(tgt);
    return r;
}

Record Record::TXT(std::string name, std::string text, uint32_t ttl) {
    Record r;
    r.name = std::move(name);
    r.type = TYPE_TXT;
    r.ttl = ttl;
    r.target = std::move(text);
    return r;
}

Record Record::SOA(std::string name, SoaData soa, uint32_t ttl) {
    Record r;
    r.name = std::move(name);
    r.type = TYPE_SOA;
    r.ttl = ttl;
    r.soa = std::move(soa);
    return r;
}

Record Record::OPT(uint16_t udpSize) {
    Record r;
    r.name.clear();
    r.type = TYPE_OPT;
    r.rclass = udpSize;
    r.ttl = 0;
    return r;
}

std::string Record::addrString() const {
    char buf[INET6_ADDRSTRLEN] = {0};
    if (type == TYPE_A)
        inet_ntop(AF_INET, a.data(), buf, sizeof(buf));
    else if (type == TYPE_AAAA)
        inet_ntop(AF_INET6, aaaa.data(), buf, sizeof(buf));
    return buf;
}

const Record* Message::edns() const {
    for (const auto& r : additionals)
        if (r.type == TYPE_OPT) return &r;
    return nullptr;
}

/* ---------------- encoding ---------------- */

namespace {

struct Encoder {
    std::vector<uint8_t>& buf;
    bool ok = true;  // false: a name had a >63-byte label (unencodable)
    explicit Encoder(std::vector<uint8_t>& b) : buf(b) { buf.clear(); }
    // Compression table: (suffix view, wire offset). Views reference
    // the record name strings, which outlive the encode; linear scan
    // beats a map for the few dozen names in a response, and avoids
    // all substring allocations on the hot path.
    static constexpr size_t kMaxOffsets = 64;
    std::pair<std::string_view, uint16_t> offsets[kMaxOffsets];
    size_t nOffsets = 0;

    const uint16_t* findOffset(std::string_view sv) const {
        for (size_t i = 0; i < nOffsets; ++i)
            if (offsets[i].first == sv) return &offsets[i].second;
        return nullptr;
    }
    void rememberOffset(std::string_view sv, uint16_t off) {
        if (nOffsets < kMaxOffsets) offsets[nOffsets++] = {sv, off};
    }

    void u8(uint8_t v) { buf.push_back(v); }
    void u16(uint16_t v) {
        buf.push_back((uint8_t)(v >> 8));
        buf.push_back((uint8_t)v);
    }
    void u32(uint32_t v) {
        buf.push_back((uint8_t)(v >> 24));
        buf.push_back((uint8_t)(v >> 16));
        buf.push_back((uint8_t)(v >> 8));
        buf.push_back((uint8_t)v);
    }
    void raw(const void* p, size_t n) {
        const uint8_t* b = (const uint8_t*)p;
        buf.insert(buf.end(), b, b + n);
    }

    /* Encode a name with compression. `name` presentation form, no
     * trailing dot. NOTE: the view must stay valid for the whole
     * encode (record name strings do). */
    void encodeName(std::string_view rest, bool compress = true) {
        while (!rest.empty()) {
            if (compress) {
                const uint16_t* off = findOffset(rest);
                if (off != nullptr) {
                    u16((uint16_t)(0xC000 | *off));
                    return;
                }
            }
            if (buf.size() < 0x4000)
                rememberOffset(rest, (uint16_t)buf.size());
            size_t dot = rest.find('.');
            std::string_view label =
                dot == std::string_view::npos ? rest
                                              : rest.substr(0, dot);
            rest = dot == std::string_view::npos
                       ? std::string_view()
                       : rest.substr(dot + 1);
            if (label.size() > 63) {
                /* DNS labels cap at 63 bytes; an over-long label from
                 * store/config data must fail the encode rather than be
                 * silently altered on the wire. */
                ok = false;
                u8(0);
                return;
            }
            u8((uint8_t)label.size());
            raw(label.data(), label.size());
        }
        u8(0);
    }

    void encodeRecord(const Record& r) {
        encodeName(r.name);
        u16(r.type);
        u16(r.rclass);
        u32(r.ttl);
        size_t lenPos = buf.size();
        u16(0);  // placeholder
        switch (r.type) {
        case TYPE_A:
            raw(r.a.data(), 4);
            break;
        case TYPE_AAAA:
            raw(r.aaaa.data(), 16);
            break;
        case TYPE_NS:
        case TYPE_CNAME:
        case TYPE_PTR:
            encodeName(r.target);
            break;
        case TYPE_TXT: {
            // split into 255-byte character-strings
            size_t off = 0;
            const std::string& t = r.target;
            do {
                size_t n = t.size() - off;
                if (n > 255) n = 255;
                u8((uint8_t)n);
                raw(t.data() + off, n);
                off += n;
            } while (off < t.size());
            break;
        }
        case TYPE_SRV:
            u16(r.priority);
            u16(r.weight);
            u16(r.port);
            // RFC 2782: SRV target must not be compressed.
            encodeName(r.target, false);
            break;
        case TYPE_SOA:
            encodeName(r.soa.mname);
            encodeName(r.soa.rname);
            u32(r.soa.serial);
            u32(r.soa.refresh);
            u32(r.soa.retry);
            u32(r.soa.expire);
            u32(r.soa.minimum);
            break;
        default:
            raw(r.rdataRaw.data(), r.rdataRaw.size());
            break;
        }
        size_t rdlen = buf.size() - lenPos - 2;
        buf[lenPos] = (uint8_t)(rdlen >> 8);
        buf[lenPos + 1] = (uint8_t)rdlen;
    }
};

uint16_t flagsWord(const Header& h) {
    uint16_t f = 0;
    if (h.qr) f |= 0x8000;
    f |= (uint16_t)((h.opcode & 0xF) << 11);
    if (h.aa) f |= 0x0400;
    if (h.tc) f |= 0x0200;
    if (h.rd) f |= 0x0100;
    if (h.ra) f |= 0x0080;
    f |= (h.rcode & 0xF);
    return f;
}

bool encodeImpl(const Message& m, bool truncated,
                std::vector<uint8_t>& out) {
    Encoder e(out);
    e.buf.reserve(512);
    Header h = m.header;
    if (truncated) h.tc = true;
    e.u16(h.id);
    e.u16(flagsWord(h));
    e.u16((uint16_t)m.questions.size());
    if (truncated) {
        e.u16(0);
        e.u16(0);
        // keep OPT in additionals even when truncating
        uint16_t nopt = 0;
        for (const auto& r : m.additionals)
            if (r.type == TYPE_OPT) nopt++;
        e.u16(nopt);
    } else {
        e.u16((uint16_t)m.answers.size());
        e.u16((uint16_t)m.authorities.size());
        e.u16((uint16_t)m.additionals.size());
    }
    for (const auto& q : m.questions) {
        e.encodeName(q.name);
        e.u16(q.qtype);
        e.u16(q.qclass);
    }
    if (!truncated) {
        for (const auto& r : m.answers) e.encodeRecord(r);
        for (const auto& r : m.authorities) e.encodeRecord(r);
        for (const auto& r : m.additionals) e.encodeRecord(r);
    } else {
        for (const auto& r : m.additionals)
            if (r.type == TYPE_OPT) e.encodeRecord(r);
    }
    return e.ok;
}

}  // namespace

void Message::encodeInto(std::vector<uint8_t>& out, size_t maxSize) const {
    if (!encodeImpl(*this, false, out)) {
        /* A record carried a >63-byte label: answer SERVFAIL instead of
         * emitting a silently altered name. Client-decoded questions
         * can never hit this (wire labels cap at 63), but re-check and
         * drop them too if a synthesized question is itself bad. */
        Message fail;
        fail.header = header;
        fail.header.qr = true;
        fail.header.rcode = RCODE_SERVFAIL;
        fail.questions = questions;
        if (!encodeImpl(fail, false, out)) {
            fail.questions.clear();
            encodeImpl(fail, false, out);
        }
        return;
    }
    if (maxSize > 0 && out.size() > maxSize)
        encodeImpl(*this, true, out);
}

std::vector<uint8_t> Message::encode(size_t maxSize) const {
    std::vector<uint8_t> out;
    encodeInto(out, maxSize);
    return out;
}

/* ---------------- decoding ---------------- */

namespace {

struct Decoder {
    const uint8_t* data;
    size_t len;
    size_t pos = 0;

    bool need(size_t n) const { return pos + n <= len; }
    bool u8(uint8_t& v) {
        if (!need(1)) return false;
        v = data[pos++];
        return true;
    }
    bool u16(uint16_t& v) {
        if (!need(2)) return false;
        v = (uint16_t)((data[pos] << 8) | data[pos + 1]);
        pos += 2;
        return true;
    }
    bool u32(uint32_t& v) {
        if (!need(4)) return false;
        v = ((uint32_t)data[pos] << 24) | ((uint32_t)data[pos + 1] << 16) |
            ((uint32_t)data[pos + 2] << 8) | (uint32_t)data[pos + 3];
        pos += 4;
        return true;
    }

    /* Decode a possibly-compressed name starting at pos. */
    bool name(std::string& out) { return nameAt(pos, out, 0, &pos); }

    bool nameAt(size_t at, std::string& out, int depth, size_t* endPos) {
        if (depth > 16) return false;
        size_t p = at;
        bool jumped = false;
        size_t afterFirstJump = 0;
        while (true) {
            if (p >= len) return false;
            uint8_t l = data[p];
            if ((l & 0xC0) == 0xC0) {
                if (p + 1 >= len) return false;
                size_t target = ((size_t)(l & 0x3F) << 8) | data[p + 1];
                if (!jumped) {
                    afterFirstJump = p + 2;
                    jumped = true;
                }
                if (target >= p) return false;  // forward pointers invalid
                p = target;
                ++depth;
                if (depth > 16) return false;
                continue;
            }
            if ((l & 0xC0) != 0) return false;  // reserved label types
            ++p;
            if (l == 0) break;
            if (p + l > len) return false;
            if (!out.empty()) out.push_back('.');
            out.append((const char*)data + p, l);
            p += l;
            if (out.size() > 255) return false;
        }
        *endPos = jumped ? afterFirstJump : p;
        return true;
    }

    bool record(Record& r) {
        if (!name(r.name)) return false;
        uint16_t rdlen;
        if (!u16(r.type) || !u16(r.rclass) || !u32(r.ttl) || !u16(rdlen))
            return false;
        if (!need(rdlen)) return false;
        size_t rdEnd = pos + rdlen;
        switch (r.type) {
        case TYPE_A:
            if (rdlen != 4) return false;
            memcpy(r.a.data(), data + pos, 4);
            pos = rdEnd;
            break;
        case TYPE_AAAA:
            if (rdlen != 16) return false;
            memcpy(r.aaaa.data(), data + pos, 16);
            pos = rdEnd;
            break;
        case TYPE_NS:
        case TYPE_CNAME:
        case TYPE_PTR:
            if (!name(r.target)) return false;
            pos = rdEnd;
            break;
        case TYPE_TXT: {
            while (pos < rdEnd) {
                uint8_t n;
                if (!u8(n)) return false;
                if (pos + n > rdEnd) return false;
                r.target.append((const char*)data + pos, n);
                pos += n;
            }
            pos = rdEnd;
            break;
        }
        case TYPE_SRV:
            if (!u16(r.priority) || !u16(r.weight) || !u16(r.port))
                return false;
            if (!name(r.target)) return false;
            pos = rdEnd;
            break;
        case TYPE_SOA:
            if (!name(r.soa.mname) || !name(r.soa.rname)) return false;
            if (!u32(r.soa.serial) || !u32(r.soa.refresh) ||
                !u32(r.soa.retry) || !u32(r.soa.expire) ||
                !u32(r.soa.minimum))
                return false;
            pos = rdEnd;
            break;
        default:
            r.rdataRaw.assign(data + pos, data + rdEnd);
            pos = rdEnd;
            break;
        }
        return true;
    }
};

}  // namespace

std::optional<Message> Message::decode(const uint8_t* data, size_t len) {
    Decoder d{data, len};
    Message m;
    uint16_t flags, qd, an, ns, ar;
    if (!d.u16(m.header.id) || !d.u16(flags) || !d.u16(qd) || !d.u16(an) ||
        !d.u16(ns) || !d.u16(ar))
        return std::nullopt;
    m.header.qr = (flags & 0x8000) != 0;
    m.header.opcode = (uint8_t)((flags >> 11) & 0xF);
    m.header.aa = (flags & 0x0400) != 0;
    m.header.tc = (flags & 0x0200) != 0;
    m.header.rd = (flags & 0x0100) != 0;
    m.header.ra = (flags & 0x0080) != 0;
    m.header.rcode = (uint8_t)(flags & 0xF);
    if (qd > 32 || an > 4096 || ns > 4096 || ar > 4096) return std::nullopt;
    for (int i = 0; i < qd; ++i) {
        Question q;
        if (!d.name(q.name) || !d.u16(q.qtype) || !d.u16(q.qclass))
            return std::nullopt;
        m.questions.push_back(std::move(q));
    }
    auto section = [&](int count, std::vector<Record>& out) {
        for (int i = 0; i < count; ++i) {
            Record r;
            if (!d.record(r)) return false;
            out.push_back(std::move(r));
        }
        return true;
    };
    if (!section(an, m.answers) || !section(ns, m.authorities) ||
        !section(ar, m.additionals))
        return std::nullopt;
    return m;
}

}  // namespace bamd::dns
