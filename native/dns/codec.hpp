/*
 * binder-amd: DNS wire codec (the in-repo replacement for the reference's
 * `mname` npm dependency — SURVEY.md §2.2; behavior observed at
 * /root/reference/lib/server.js call sites).
 *
 * Covers: header flags, questions, A/AAAA/NS/CNAME/SOA/PTR/TXT/SRV/OPT
 * records, name compression on encode, pointer-following on decode, EDNS
 * payload-size negotiation, and TC-bit truncation for UDP.
 */
#pragma once

#include <array>
#include <cstdint>
#include <optional>
#include <string>
#include <vector>

namespace bamd::dns {

/* RR types (RFC 1035/2782/6891). */
enum : uint16_t {
    TYPE_A = 1,
    TYPE_NS = 2,
    TYPE_CNAME = 5,
    TYPE_SOA = 6,
    TYPE_PTR = 12,
    TYPE_MX = 15,
    TYPE_TXT = 16,
    TYPE_AAAA = 28,
    TYPE_SRV = 33,
    TYPE_OPT = 41,
    TYPE_ANY = 255,
};

enum : uint16_t { CLASS_IN = 1 };

/*
 * Rcodes. The reference's rcode policy matrix (REFUSED for misses, SERVFAIL
 * when ZK is unavailable, NXDOMAIN for wrong SRV proto, NOTIMP for
 * unsupported qtypes) is implemented in the resolution engine; the codec
 * just carries the value (lib/server.js:227-246, 186-192, 343, 501-505).
 */
enum Rcode : uint8_t {
    RCODE_NOERROR = 0,
    RCODE_FORMERR = 1,
    RCODE_SERVFAIL = 2,
    RCODE_NXDOMAIN = 3,
    RCODE_NOTIMP = 4,
    RCODE_REFUSED = 5,
};

const char* typeName(uint16_t t);          // "A", "SRV", ... or "TYPE%u"
uint16_t typeFromName(const std::string&); // 0 if unknown
const char* rcodeName(uint8_t rc);

struct Question {
    std::string name;  // presentation form, no trailing dot; "" = root
    uint16_t qtype = TYPE_A;
    uint16_t qclass = CLASS_IN;
};

struct SoaData {
    std::string mname;
    std::string rname;
    uint32_t serial = 0;
    uint32_t refresh = 0;
    uint32_t retry = 0;
    uint32_t expire = 0;
    uint32_t minimum = 0;
};

struct Record {
    std::string name;
    uint16_t type = TYPE_A;
    uint16_t rclass = CLASS_IN;  // for OPT: requestor UDP payload size
    uint32_t ttl = 0;            // for OPT: ext-rcode/version/DO bits

    /* rdata, interpreted per type: */
    std::array<uint8_t, 4> a{};      // A
    std::array<uint8_t, 16> aaaa{};  // AAAA
    std::string target;              // CNAME/PTR/NS target; TXT text
    uint16_t priority = 0, weight = 0, port = 0;  // SRV (target above)
    SoaData soa;                     // SOA
    std::vector<uint8_t> rdataRaw;   // OPT options / unknown types

    static Record A(std::string name, const std::string& ipv4, uint32_t ttl);
    static Record AAAA(std::string name, const std::string& ipv6,
                       uint32_t ttl);
    static Record SRV(std::string name, std::string tgt, uint16_t port,
                      uint32_t ttl, uint16_t prio = 0, uint16_t weight = 10);
    static Record PTR(std::string name, std::string tgt, uint32_t ttl);
    static Record CNAME(std::string name, std::string tgt, uint32_t ttl);
    static Record TXT(std::string name, std::string text, uint32_t ttl);
    static Record SOA(std::string name, SoaData soa, uint32_t ttl);
    static Record OPT(uint16_t udpSize);

    /* Presentation of the A/AAAA address ("1.2.3.4"). */
    std::string addrString() const;
};

struct Header {
    uint16_t id = 0;
    bool qr = false;  // response
    uint8_t opcode = 0;
    bool aa = false;
    bool tc = false;
    bool rd = false;
    bool ra = false;
    uint8_t rcode = RCODE_NOERROR;
};

struct Message {
    Header header;
    std::vector<Question> questions;
    std::vector<Record> answers;
    std::vector<Record> authorities;
    std::vector<Record> additionals;

    /* First OPT record in additionals, if any (EDNS). */
    const Record* edns() const;

    /*
     * Encode. maxSize 0 = unlimited (TCP). On overflow, the message is
     * re-encoded with all RR sections dropped and TC set (client retries
     * over TCP).
     */
    std::vector<uint8_t> encode(size_t maxSize = 0) const;
    /* Same, reusing the caller's buffer (cleared first) — the hot-path
     * variant: no allocation once the buffer has warmed up. */
    void encodeInto(std::vector<uint8_t>& out, size_t maxSize = 0) const;

    /* Decode; nullopt on malformed wire data. */
    static std::optional<Message> decode(const uint8_t* data, size_t len);
};

/* Lowercase ASCII in place (DNS names are case-insensitive). */
void toLowerAscii(std::string& s);

}  // namespace bamd::dns
