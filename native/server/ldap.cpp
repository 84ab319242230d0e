#include "ldap.hpp"

#include <arpa/inet.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <openssl/err.h>
#include <openssl/ssl.h>
#include <sys/socket.h>
#include <sys/time.h>
#include <unistd.h>

#include <cstring>

namespace bamd::ldap {

/* ---------------- BER ---------------- */

std::string berTLV(uint8_t tag, const std::string& content) {
    std::string out;
    out.push_back((char)tag);
    size_t n = content.size();
    if (n < 128) {
        out.push_back((char)n);
    } else {
        uint8_t lenBytes[4];
        int nb = 0;
        size_t v = n;
        while (v > 0) {
            lenBytes[nb++] = (uint8_t)(v & 0xFF);
            v >>= 8;
        }
        out.push_back((char)(0x80 | nb));
        for (int i = nb - 1; i >= 0; --i) out.push_back((char)lenBytes[i]);
    }
    out += content;
    return out;
}

static std::string berIntBody(int64_t v) {
    /* minimal two's-complement encoding */
    std::string body;
    bool more = true;
    uint64_t u = (uint64_t)v;
    char bytes[9];
    int n = 0;
    while (more) {
        bytes[n++] = (char)(u & 0xFF);
        int8_t top = (int8_t)(u & 0xFF);
        u = (uint64_t)((int64_t)u >> 8);
        more = !(((int64_t)u == 0 && top >= 0) ||
                 ((int64_t)u == -1 && top < 0));
        if (n >= 8) break;
    }
    for (int i = n - 1; i >= 0; --i) body.push_back(bytes[i]);
    return body;
}

std::string berInt(int64_t v) { return berTLV(0x02, berIntBody(v)); }
std::string berEnum(int64_t v) { return berTLV(0x0A, berIntBody(v)); }
std::string berBool(bool b) {
    return berTLV(0x01, std::string(1, b ? (char)0xFF : (char)0x00));
}
std::string berOctet(const std::string& s, uint8_t tag) {
    return berTLV(tag, s);
}

bool BerReader::readTL(uint8_t& tag, size_t& vlen) {
    if (pos + 2 > len) {
        ok = false;
        return false;
    }
    tag = p[pos++];
    uint8_t l = p[pos++];
    if (l < 128) {
        vlen = l;
    } else {
        int nb = l & 0x7F;
        if (nb == 0 || nb > 4 || pos + (size_t)nb > len) {
            ok = false;
            return false;
        }
        vlen = 0;
        for (int i = 0; i < nb; ++i) vlen = (vlen << 8) | p[pos++];
    }
    if (pos + vlen > len) {
        ok = false;
        return false;
    }
    return true;
}

bool BerReader::readTLV(uint8_t& tag, BerReader& inner) {
    size_t vlen;
    if (!readTL(tag, vlen)) return false;
    inner = BerReader{p + pos, vlen};
    pos += vlen;
    return true;
}

int64_t BerReader::readInt() {
    uint8_t tag;
    size_t vlen;
    if (!readTL(tag, vlen)) return 0;
    if (tag != 0x02 && tag != 0x0A) {
        pos += vlen;
        return 0;
    }
    int64_t v = (vlen > 0 && (p[pos] & 0x80)) ? -1 : 0;
    for (size_t i = 0; i < vlen; ++i) v = (v << 8) | p[pos + i];
    pos += vlen;
    return v;
}

std::string BerReader::readOctet(uint8_t expectTag) {
    uint8_t tag;
    size_t vlen;
    if (!readTL(tag, vlen)) return "";
    std::string out((const char*)p + pos, vlen);
    pos += vlen;
    (void)expectTag;
    return out;
}

/* ---------------- client ---------------- */

Client::~Client() { close(); }

void Client::close() {
    if (ssl_ != nullptr) {
        SSL_shutdown((SSL*)ssl_);
        SSL_free((SSL*)ssl_);
        ssl_ = nullptr;
    }
    if (sslCtx_ != nullptr) {
        SSL_CTX_free((SSL_CTX*)sslCtx_);
        sslCtx_ = nullptr;
    }
    if (fd_ >= 0) {
        ::close(fd_);
        fd_ = -1;
    }
}

ssize_t Client::rawRead(void* buf, size_t n) {
    if (ssl_ != nullptr) return SSL_read((SSL*)ssl_, buf, (int)n);
    return ::recv(fd_, buf, n, 0);
}

ssize_t Client::rawWrite(const void* buf, size_t n) {
    if (ssl_ != nullptr) return SSL_write((SSL*)ssl_, buf, (int)n);
    return ::send(fd_, buf, n, 0);
}

bool Client::connect() {
    fd_ = socket(AF_INET, SOCK_STREAM | SOCK_CLOEXEC, 0);
    if (fd_ < 0) {
        err_ = "socket failed";
        return false;
    }
    struct timeval tv {opts_.timeoutMs / 1000,
                       (opts_.timeoutMs % 1000) * 1000};
    setsockopt(fd_, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));
    setsockopt(fd_, SOL_SOCKET, SO_SNDTIMEO, &tv, sizeof(tv));
    int one = 1;
    setsockopt(fd_, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
    struct sockaddr_in sa {};
    sa.sin_family = AF_INET;
    sa.sin_port = htons(opts_.port);
    if (inet_pton(AF_INET, opts_.host.c_str(), &sa.sin_addr) != 1) {
        struct addrinfo hints {};
        hints.ai_family = AF_INET;
        hints.ai_socktype = SOCK_STREAM;
        struct addrinfo* res = nullptr;
        int rc = getaddrinfo(opts_.host.c_str(), nullptr, &hints, &res);
        if (rc != 0 || res == nullptr) {
            err_ = "cannot resolve host: " + opts_.host;
            if (res) freeaddrinfo(res);
            close();
            return false;
        }
        sa.sin_addr = ((struct sockaddr_in*)res->ai_addr)->sin_addr;
        freeaddrinfo(res);
    }
    if (::connect(fd_, (struct sockaddr*)&sa, sizeof(sa)) != 0) {
        err_ = std::string("connect: ") + strerror(errno);
        close();
        return false;
    }

    if (opts_.tls) {
        SSL_CTX* ctx = SSL_CTX_new(TLS_client_method());
        if (ctx == nullptr) {
            err_ = "SSL_CTX_new failed";
            close();
            return false;
        }
        if (opts_.tlsVerify) {
            bool caOk;
            if (!opts_.caFile.empty())
                caOk = SSL_CTX_load_verify_locations(
                           ctx, opts_.caFile.c_str(), nullptr) == 1;
            else
                caOk = SSL_CTX_set_default_verify_paths(ctx) == 1;
            if (!caOk) {
                err_ = "cannot load CA certificates";
                SSL_CTX_free(ctx);
                close();
                return false;
            }
            SSL_CTX_set_verify(ctx, SSL_VERIFY_PEER, nullptr);
        } else {
            SSL_CTX_set_verify(ctx, SSL_VERIFY_NONE, nullptr);
        }
        sslCtx_ = ctx;
        SSL* ssl = SSL_new(ctx);
        if (opts_.tlsVerify) {
            /* Bind the verified chain to the peer we dialed: IP SAN
             * check when host is an address, DNS SAN/CN otherwise. */
            X509_VERIFY_PARAM* vp = SSL_get0_param(ssl);
            struct in_addr ia;
            if (inet_pton(AF_INET, opts_.host.c_str(), &ia) == 1)
                X509_VERIFY_PARAM_set1_ip_asc(vp, opts_.host.c_str());
            else
                X509_VERIFY_PARAM_set1_host(vp, opts_.host.c_str(),
                                            opts_.host.size());
        }
        SSL_set_fd(ssl, fd_);
        ssl_ = ssl;
        if (SSL_connect(ssl) != 1) {
            long vr = SSL_get_verify_result(ssl);
            err_ = "TLS handshake failed";
            if (vr != X509_V_OK) {
                err_ += ": ";
                err_ += X509_verify_cert_error_string(vr);
            }
            close();
            return false;
        }
    }

    /* BindRequest (simple auth) */
    std::string bind = berInt(3) + berOctet(opts_.bindDn) +
                       berOctet(opts_.bindPassword, 0x80);
    std::string msg =
        berTLV(0x30, berInt(nextId_++) + berTLV(0x60, bind));
    if (!sendMessage(msg)) return false;

    std::string resp;
    if (!readMessage(resp)) return false;
    BerReader r{(const uint8_t*)resp.data(), resp.size()};
    uint8_t tag;
    BerReader env{nullptr, 0};
    if (!r.readTLV(tag, env) || tag != 0x30) {
        err_ = "malformed bind response";
        return false;
    }
    env.readInt();  // messageID
    BerReader op{nullptr, 0};
    if (!env.readTLV(tag, op) || tag != 0x61) {
        err_ = "unexpected bind response op";
        return false;
    }
    int64_t rc = op.readInt();
    if (rc != 0) {
        err_ = "bind failed, resultCode " + std::to_string(rc);
        return false;
    }
    return true;
}

bool Client::sendMessage(const std::string& payload) {
    size_t off = 0;
    while (off < payload.size()) {
        ssize_t nw = rawWrite(payload.data() + off, payload.size() - off);
        if (nw <= 0) {
            err_ = "write failed";
            return false;
        }
        off += (size_t)nw;
    }
    return true;
}

bool Client::readMessage(std::string& out) {
    /* read one complete top-level BER TLV */
    while (true) {
        if (rbuf_.size() >= 2) {
            BerReader probe{(const uint8_t*)rbuf_.data(), rbuf_.size()};
            uint8_t tag;
            size_t vlen;
            size_t save = probe.pos;
            if (probe.readTL(tag, vlen)) {
                size_t total = probe.pos - save + vlen;
                if (rbuf_.size() >= total) {
                    out = rbuf_.substr(0, total);
                    rbuf_.erase(0, total);
                    return true;
                }
            } else if (rbuf_.size() > 6) {
                /* header present but length incomplete? keep reading
                 * unless clearly oversized */
                if (rbuf_.size() > (1u << 24)) {
                    err_ = "oversized LDAP message";
                    return false;
                }
            }
        }
        char buf[8192];
        ssize_t nr = rawRead(buf, sizeof(buf));
        if (nr <= 0) {
            err_ = "read failed/timeout";
            return false;
        }
        rbuf_.append(buf, (size_t)nr);
    }
}

bool Client::search(const std::string& base, const std::string& attr,
                    const std::string& value, std::vector<Entry>& out) {
    int msgId = nextId_++;
    std::string filter = berTLV(0xA3, berOctet(attr) + berOctet(value));
    std::string req = berOctet(base) + berEnum(2) /* wholeSubtree */ +
                      berEnum(0) /* neverDeref */ + berInt(0) +
                      berInt(0) + berBool(false) + filter +
                      berTLV(0x30, "") /* all attributes */;
    std::string msg =
        berTLV(0x30, berInt(msgId) + berTLV(0x63, req));
    if (!sendMessage(msg)) return false;

    while (true) {
        std::string resp;
        if (!readMessage(resp)) return false;
        BerReader r{(const uint8_t*)resp.data(), resp.size()};
        uint8_t tag;
        BerReader env{nullptr, 0};
        if (!r.readTLV(tag, env) || tag != 0x30) {
            err_ = "malformed search response";
            return false;
        }
        env.readInt();  // messageID
        BerReader op{nullptr, 0};
        if (!env.readTLV(tag, op)) {
            err_ = "malformed search op";
            return false;
        }
        if (tag == 0x64) {  // SearchResultEntry
            Entry e;
            e["dn"].push_back(op.readOctet());
            BerReader attrs{nullptr, 0};
            uint8_t t2;
            if (op.readTLV(t2, attrs) && t2 == 0x30) {
                while (!attrs.atEnd() && attrs.ok) {
                    BerReader one{nullptr, 0};
                    if (!attrs.readTLV(t2, one) || t2 != 0x30) break;
                    std::string name = one.readOctet();
                    BerReader vals{nullptr, 0};
                    if (one.readTLV(t2, vals) && t2 == 0x31) {
                        while (!vals.atEnd() && vals.ok)
                            e[name].push_back(vals.readOctet());
                    }
                }
            }
            out.push_back(std::move(e));
        } else if (tag == 0x65) {  // SearchResultDone
            int64_t rc = op.readInt();
            if (rc != 0) {
                err_ = "search failed, resultCode " + std::to_string(rc);
                return false;
            }
            return true;
        } else if (tag == 0x73) {
            /* SearchResultReference: ignore */
        } else {
            /* unsolicited/unknown: ignore */
        }
    }
}

}  // namespace bamd::ldap
