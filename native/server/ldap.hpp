/*
 * binder-amd: minimal LDAPv3 client (the `ufds` npm dependency's
 * capability surface as used by the reference: simple bind + subtree
 * search, recursion.js:129-148, 210-219 — its listResolvers(region)
 * is `search base "region=<r>, o=smartdc" filter
 * (objectclass=resolver)` per the doc comment recursion.js:17-19).
 *
 * BLOCKING client with socket timeouts — the recursion module runs it
 * on a short-lived helper thread every 5 minutes and posts results
 * back to the event loop (EventLoop::postFromThread); it must never be
 * called from the serving thread. Supports ldap:// (plain) and
 * ldaps:// (OpenSSL, no certificate verification — matching internal
 * UFDS deployments with private CAs).
 */
#pragma once

#include <cstdint>
#include <map>
#include <string>
#include <vector>

namespace bamd::ldap {

struct Options {
    std::string host;
    uint16_t port = 389;
    bool tls = false;           // ldaps://
    /* Certificate verification is ON by default (the reference trusts
     * the ldapjs/ufds stack's default posture; an internal-CA deploy
     * sets caFile, and tlsVerify=false is an explicit opt-out). */
    bool tlsVerify = true;
    std::string caFile;         // PEM bundle; empty = system CA paths
    std::string bindDn;         // empty = anonymous
    std::string bindPassword;
    int timeoutMs = 10000;
};

using Entry = std::map<std::string, std::vector<std::string>>;

class Client {
  public:
    explicit Client(Options opts) : opts_(std::move(opts)) {}
    ~Client();

    /* Connect + (optional TLS) + bind. Returns false and sets error()
     * on failure. */
    bool connect();

    /* Subtree search with an equality filter; entries get a "dn" key
     * plus each returned attribute. */
    bool search(const std::string& base, const std::string& attr,
                const std::string& value, std::vector<Entry>& out);

    void close();
    bool isConnected() const { return fd_ >= 0; }
    const Options& options() const { return opts_; }
    const std::string& error() const { return err_; }

  private:
    bool sendMessage(const std::string& payload);
    bool readMessage(std::string& out);
    ssize_t rawRead(void* buf, size_t n);
    ssize_t rawWrite(const void* buf, size_t n);

    Options opts_;
    int fd_ = -1;
    void* ssl_ = nullptr;      // SSL*
    void* sslCtx_ = nullptr;   // SSL_CTX*
    int nextId_ = 1;
    std::string err_;
    std::string rbuf_;
};

/* ---- BER helpers (exposed for tests) ---- */
std::string berTLV(uint8_t tag, const std::string& content);
std::string berInt(int64_t v);
std::string berEnum(int64_t v);
std::string berBool(bool b);
std::string berOctet(const std::string& s, uint8_t tag = 0x04);

struct BerReader {
    const uint8_t* p;
    size_t len;
    size_t pos = 0;
    bool ok = true;

    bool readTL(uint8_t& tag, size_t& vlen);
    bool readTLV(uint8_t& tag, BerReader& inner);
    int64_t readInt();                   // INTEGER or ENUMERATED
    std::string readOctet(uint8_t expectTag = 0x04);
    bool atEnd() const { return pos >= len; }
};

}  // namespace bamd::ldap
