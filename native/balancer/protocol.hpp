/*
 * binder-amd: balancer <-> backend framed protocol ("bsock1").
 *
 * The reference's mname-balancer submodule (not vendored in the snapshot;
 * SURVEY.md §2 row 11) multiplexes :53 traffic to per-process UNIX
 * sockets, conveying the original client address/port/family so backends
 * log real client info (lib/server.js:486-487). The wire format is not
 * public, so bsock1 is this project's own framing with the same
 * capability:
 *
 *   frame  := magic(0xB5) type(u8) payload_len(u32le) payload
 *   QUERY  (1): req_id(u32le) family(u8: 4|6) proto(u8: 0 udp|1 tcp)
 *               src_port(u16le) src_addr(16B) dns_message
 *   REPLY  (2): req_id(u32le) dns_message
 *   PING   (3): empty          (balancer -> backend health probe)
 *   PONG   (4): empty
 *
 * One persistent stream connection per backend, queries multiplexed by
 * req_id. Backend presence = socket exists in the socket directory and
 * accepts connections (main.js:181-199: unlink on SIGTERM is the drain
 * signal).
 */
#pragma once

#include <cstdint>
#include <cstring>
#include <string>

namespace bamd::bsock {

constexpr uint8_t kMagic = 0xB5;
enum : uint8_t {
    FRAME_QUERY = 1,
    FRAME_REPLY = 2,
    FRAME_PING = 3,
    FRAME_PONG = 4,
};
constexpr size_t kHeaderLen = 6;
constexpr size_t kQueryHeadLen = 4 + 1 + 1 + 2 + 16;
constexpr uint32_t kMaxPayload = 1 << 20;

inline void putU32(std::string& out, uint32_t v) {
    char b[4] = {(char)(v), (char)(v >> 8), (char)(v >> 16),
                 (char)(v >> 24)};
    out.append(b, 4);
}
inline uint32_t getU32(const uint8_t* p) {
    return (uint32_t)p[0] | ((uint32_t)p[1] << 8) | ((uint32_t)p[2] << 16) |
           ((uint32_t)p[3] << 24);
}
inline void putU16(std::string& out, uint16_t v) {
    char b[2] = {(char)(v), (char)(v >> 8)};
    out.append(b, 2);
}
inline uint16_t getU16(const uint8_t* p) {
    return (uint16_t)((uint16_t)p[0] | ((uint16_t)p[1] << 8));
}

inline void appendFrame(std::string& out, uint8_t type,
                        const std::string& payload) {
    out.push_back((char)kMagic);
    out.push_back((char)type);
    putU32(out, (uint32_t)payload.size());
    out += payload;
}

struct QueryFrame {
    uint32_t reqId;
    uint8_t family;  // 4 or 6
    uint8_t proto;   // 0 udp, 1 tcp
    uint16_t srcPort;
    uint8_t srcAddr[16];
    const uint8_t* dns;
    size_t dnsLen;
};

/* Parse one QUERY payload; returns false if malformed. */
inline bool parseQuery(const uint8_t* p, size_t len, QueryFrame& out) {
    if (len < kQueryHeadLen) return false;
    out.reqId = getU32(p);
    out.family = p[4];
    out.proto = p[5];
    out.srcPort = getU16(p + 6);
    memcpy(out.srcAddr, p + 8, 16);
    out.dns = p + kQueryHeadLen;
    out.dnsLen = len - kQueryHeadLen;
    return true;
}

}  // namespace bamd::bsock
