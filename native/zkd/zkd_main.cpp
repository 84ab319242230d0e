/*
 * zkd: native single-node ZooKeeper-compatible registry.
 *
 * The reference ships a real ZooKeeper inside its image
 * (/root/reference/Makefile:74-77) as binder's backing store; this is
 * the from-scratch native equivalent for single-node deployments and
 * development, speaking the same client wire protocol (jute) the
 * in-repo client (native/zk/client.cpp) and any ZooKeeper client use:
 *
 *   - session handshake with resume + timeout-based expiry (expired
 *     sessions' ephemeral znodes are reaped, firing watches — the
 *     mechanism by which dead registrars vanish from discovery);
 *   - persistent / ephemeral / sequential creates, delete, setData
 *     with version checks, getData / getChildren(2) / exists with
 *     one-shot watches, ping, close;
 *   - a durable transaction log in the real FileTxnLog v2 format
 *     (magic ZKLG, adler32 checksum, 0x42 end-of-record) that
 *     bin/zklogcat decodes; replay on restart truncates a torn tail
 *     at the first corrupt entry (like ZooKeeper) and compacts when
 *     history dominates state;
 *   - the ruok/stat/srvr four-letter-word diagnostics real ZK tools
 *     expect.
 *
 * Single-threaded epoll loop; the protocol details are pinned by
 * tests/test_zk_golden.py (spec-derived byte vectors run against this
 * server too) and tests/test_zkd_native.py.
 */
#include <arpa/inet.h>
#include <fcntl.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/epoll.h>
#include <signal.h>
#include <sys/signalfd.h>
#include <sys/socket.h>
#include <sys/stat.h>
#include <unistd.h>

#include <cstdio>
#include <algorithm>
#include <cstring>
#include <functional>
#include <map>
#include <memory>
#include <set>
#include <string>
#include <vector>

#include "../common/log.hpp"
#include "../common/loop.hpp"
#include "../zk/jute.hpp"

using namespace bamd;
using namespace bamd::zk;

namespace {

constexpr int32_t kLogMagic = 0x5A4B4C47;  // "ZKLG"
constexpr int32_t ZUNIMPLEMENTED = -6;

uint32_t adler32(const uint8_t* data, size_t len) {
    uint32_t a = 1, b = 0;
    for (size_t i = 0; i < len; ++i) {
        a = (a + data[i]) % 65521u;
        b = (b + a) % 65521u;
    }
    return (b << 16) | a;
}

std::string parentOf(const std::string& path) {
    if (path == "/") return "";
    size_t idx = path.rfind('/');
    return idx == 0 ? "/" : path.substr(0, idx);
}

std::string baseOf(const std::string& path) {
    return path.substr(path.rfind('/') + 1);
}

int64_t wallMillis() {
    struct timespec ts;
    clock_gettime(CLOCK_REALTIME, &ts);
    return (int64_t)ts.tv_sec * 1000 + ts.tv_nsec / 1000000;
}

struct Node {
    std::string data;
    std::set<std::string> children;
    int32_t version = 0;
    int32_t cversion = 0;
    int64_t ephemeralOwner = 0;
    int64_t czxid = 0, mzxid = 0, ctime = 0, mtime = 0, pzxid = 0;
};

struct Session {
    int64_t id;
    int32_t timeoutMs;
    int64_t lastSeenMs;
};

struct Conn {
    int fd = -1;
    std::string in, out;
    size_t outOff = 0;
    bool handshaken = false;
    bool writeBlocked = false;
    bool closed = false;
    int64_t sessionId = 0;
    std::set<std::string> dataWatches, childWatches, existsWatches;
};

class Zkd {
  public:
    Zkd(EventLoop* loop, Logger log, std::string host, uint16_t port,
        std::string dataDir, int32_t sessionTimeoutMs)
        : loop_(loop), log_(std::move(log)), host_(std::move(host)),
          port_(port), dataDir_(std::move(dataDir)),
          sessionTimeoutMs_(sessionTimeoutMs) {
        nodes_["/"] = Node{};
    }

    bool start();
    uint16_t boundPort() const { return port_; }
    size_t nodeCount() const { return nodes_.size() - 1; }

  private:
    /* ---- txn log ---- */
    bool openLog();
    void appendTxn(int32_t type, const std::string& body,
                   int64_t clientId);
    long replay(const std::string& path);  // -1: no/invalid header
    void maybeCompact();

    /* ---- tree (mirrors the semantics pinned by the golden tests) */
    int32_t doCreate(const std::string& path, const std::string& data,
                     int64_t ephemeralOwner, int64_t txnSession,
                     bool log);
    int32_t doSet(const std::string& path, const std::string& data,
                  int32_t version, int64_t txnSession);
    int32_t doDelete(const std::string& path, int32_t version,
                     int64_t txnSession);
    void fire(const std::string& path, int32_t ev, bool child,
              bool existsOnly);
    void writeStat(Writer& w, const Node& n);
    void reapEphemerals(int64_t sessionId);

    /* ---- sessions ---- */
    void sweepSessions();

    /* ---- IO ---- */
    void onAccept();
    void onConn(const std::shared_ptr<Conn>& c, uint32_t ev);
    void handlePacket(const std::shared_ptr<Conn>& c,
                      const uint8_t* p, size_t n);
    void handshake(const std::shared_ptr<Conn>& c, Reader& r);
    void op(const std::shared_ptr<Conn>& c, Reader& r);
    void reply(Conn* c, int32_t xid, int32_t err,
               const std::string& body = "");
    void sendRaw(Conn* c, const std::string& payload);
    void flushConn(Conn* c);
    void closeConn(Conn* c);
    bool fourLetter(const std::shared_ptr<Conn>& c);

    EventLoop* loop_;
    Logger log_;
    std::string host_;
    uint16_t port_;
    std::string dataDir_;
    int32_t sessionTimeoutMs_;

    int listenFd_ = -1;
    std::map<std::string, Node> nodes_;
    std::map<int64_t, Session> sessions_;
    std::map<int, std::shared_ptr<Conn>> conns_;
    int64_t nextSession_ = 0x200000001;
    int64_t zxid_ = 1;
    long replayedEntries_ = 0;
    FILE* txnlog_ = nullptr;
    std::string logPath_;
    uint64_t opsServed_ = 0, watchesFired_ = 0, sessionsMade_ = 0;
};

bool Zkd::openLog() {
    if (dataDir_.empty()) return true;
    ::mkdir(dataDir_.c_str(), 0755);
    logPath_ = dataDir_ + "/log.1";
    struct stat st;
    bool fresh = ::stat(logPath_.c_str(), &st) != 0;
    if (!fresh) {
        long good = replay(logPath_);
        if (good < 0) {
            /* header never landed: start a fresh log */
            FILE* f = fopen(logPath_.c_str(), "wb");
            if (f == nullptr) return false;
            Writer h;
            h.i32(kLogMagic);
            h.i32(2);
            h.i64(0);
            fwrite(h.buf.data(), 1, h.buf.size(), f);
            fclose(f);
        } else if (good < (long)st.st_size) {
            /* torn tail: truncate to the last verified entry so new
             * appends stay reachable to the next replay */
            if (truncate(logPath_.c_str(), good) != 0)
                log_.warn({{"path", Json(logPath_)}},
                          "could not truncate torn txn log tail");
        }
        maybeCompact();
    }
    txnlog_ = fopen(logPath_.c_str(), "ab");
    if (txnlog_ == nullptr) return false;
    if (fresh) {
        Writer h;
        h.i32(kLogMagic);
        h.i32(2);
        h.i64(0);
        fwrite(h.buf.data(), 1, h.buf.size(), txnlog_);
        fflush(txnlog_);
    }
    return true;
}

void Zkd::appendTxn(int32_t type, const std::string& body,
                    int64_t clientId) {
    if (txnlog_ == nullptr) return;
    Writer t;
    t.i64(clientId);
    t.i32(0);  // cxid
    t.i64(zxid_);
    t.i64(wallMillis());  // real FileTxnLog carries wall-clock time
    t.i32(type);
    t.buf += body;
    uint32_t crc = adler32((const uint8_t*)t.buf.data(), t.buf.size());
    Writer rec;
    rec.i64((int64_t)crc);
    rec.i32((int32_t)t.buf.size());
    rec.buf += t.buf;
    rec.buf.push_back(0x42);
    fwrite(rec.buf.data(), 1, rec.buf.size(), txnlog_);
    fflush(txnlog_);
}

long Zkd::replay(const std::string& path) {
    FILE* f = fopen(path.c_str(), "rb");
    if (f == nullptr) return -1;
    std::string data;
    char buf[65536];
    size_t nr;
    while ((nr = fread(buf, 1, sizeof(buf), f)) > 0) data.append(buf, nr);
    fclose(f);
    replayedEntries_ = 0;
    if (data.size() < 16) return -1;
    Reader hr{(const uint8_t*)data.data(), data.size()};
    if (hr.i32() != kLogMagic) return -1;
    size_t off = 16;
    while (off + 12 <= data.size()) {
        Reader r{(const uint8_t*)data.data() + off, data.size() - off};
        int64_t crc = r.i64();
        int32_t tlen = r.i32();
        if (crc == 0 || tlen <= 0 ||
            off + 12 + (size_t)tlen + 1 > data.size())
            break;
        const uint8_t* txn = (const uint8_t*)data.data() + off + 12;
        if ((uint32_t)crc != adler32(txn, (size_t)tlen) ||
            (uint8_t)data[off + 12 + (size_t)tlen] != 0x42)
            break;
        off += 12 + (size_t)tlen + 1;
        replayedEntries_++;
        Reader tr{txn, (size_t)tlen};
        tr.i64();  // clientId
        tr.i32();  // cxid
        int64_t zx = tr.i64();
        tr.i64();  // time
        int32_t type = tr.i32();
        if (zx > zxid_) zxid_ = zx;
        if (type == OP_CREATE) {
            std::string p = tr.str();
            std::string d = tr.str();
            int32_t nacl = tr.i32();
            for (int32_t i = 0; i < nacl && tr.ok; ++i) {
                tr.i32();
                tr.str();
                tr.str();
            }
            bool ephemeral = tr.boolean();
            /* ephemeral creates are skipped: their sessions are gone,
             * which is what a real restart + expiry yields */
            if (ephemeral || !tr.ok) continue;
            auto pit = nodes_.find(parentOf(p));
            if (pit == nodes_.end() || nodes_.count(p)) continue;
            Node n;
            n.data = d;
            n.czxid = n.mzxid = zx;
            nodes_[p] = std::move(n);
            pit->second.children.insert(baseOf(p));
        } else if (type == OP_SETDATA) {
            std::string p = tr.str();
            std::string d = tr.str();
            auto it = nodes_.find(p);
            if (it != nodes_.end()) {
                it->second.data = d;
                it->second.version++;
                it->second.mzxid = zx;
            }
        } else if (type == OP_DELETE) {
            std::string p = tr.str();
            auto it = nodes_.find(p);
            if (it != nodes_.end() && it->second.children.empty()) {
                nodes_.erase(it);
                auto pit = nodes_.find(parentOf(p));
                if (pit != nodes_.end())
                    pit->second.children.erase(baseOf(p));
            }
        }
    }
    return (long)off;
}

void Zkd::maybeCompact() {
    long live = (long)nodes_.size() - 1;
    if (replayedEntries_ <= 2 * live + 64) return;
    std::string tmp = logPath_ + ".compact";
    FILE* f = fopen(tmp.c_str(), "wb");
    if (f == nullptr) return;
    Writer h;
    h.i32(kLogMagic);
    h.i32(2);
    h.i64(0);
    fwrite(h.buf.data(), 1, h.buf.size(), f);
    /* parents before children: sort by depth then path */
    std::vector<const std::string*> paths;
    for (const auto& [p, n] : nodes_)
        if (p != "/") paths.push_back(&p);
    std::sort(paths.begin(), paths.end(),
              [](const std::string* a, const std::string* b) {
                  long da = std::count(a->begin(), a->end(), '/');
                  long db = std::count(b->begin(), b->end(), '/');
                  return da != db ? da < db : *a < *b;
              });
    int64_t now = wallMillis();
    for (const std::string* p : paths) {
        Writer t;
        t.i64(0);
        t.i32(0);
        t.i64(zxid_);
        t.i64(now);
        t.i32(OP_CREATE);
        t.str(*p);
        t.buffer(nodes_[*p].data);
        t.i32(0);  // empty ACL vector (replay skips what it reads)
        t.boolean(false);
        uint32_t crc =
            adler32((const uint8_t*)t.buf.data(), t.buf.size());
        Writer rec;
        rec.i64((int64_t)crc);
        rec.i32((int32_t)t.buf.size());
        rec.buf += t.buf;
        rec.buf.push_back(0x42);
        fwrite(rec.buf.data(), 1, rec.buf.size(), f);
    }
    fclose(f);
    rename(tmp.c_str(), logPath_.c_str());
}

void Zkd::writeStat(Writer& w, const Node& n) {
    w.i64(n.czxid);
    w.i64(n.mzxid);
    w.i64(n.ctime);
    w.i64(n.mtime);
    w.i32(n.version);
    w.i32(n.cversion);
    w.i32(0);  // aversion
    w.i64(n.ephemeralOwner);
    w.i32((int32_t)n.data.size());
    w.i32((int32_t)n.children.size());
    w.i64(n.pzxid);
}

void Zkd::fire(const std::string& path, int32_t ev, bool child,
               bool existsOnly) {
    Writer w;
    w.i32(XID_NOTIFICATION);
    w.i64(-1);
    w.i32(ZOK);
    w.i32(ev);
    w.i32(STATE_SYNC_CONNECTED);
    w.str(path);
    /* collect first: sendRaw -> flushConn may closeConn, which
     * erases from conns_ and would invalidate this iteration */
    std::vector<std::shared_ptr<Conn>> fired;
    for (auto& [fd, c] : conns_) {
        if (c->closed || !c->handshaken) continue;
        bool hit = false;
        if (child) {
            hit = c->childWatches.erase(path) > 0;
        } else {
            if (!existsOnly) hit = c->dataWatches.erase(path) > 0;
            if (c->existsWatches.erase(path) > 0) hit = true;
        }
        if (hit) fired.push_back(c);
    }
    for (auto& c : fired) {
        if (c->closed) continue;
        sendRaw(c.get(), w.buf);
        watchesFired_++;
    }
}

int32_t Zkd::doCreate(const std::string& path, const std::string& data,
                      int64_t ephemeralOwner, int64_t txnSession,
                      bool logTxn) {
    if (nodes_.count(path)) return ZNODEEXISTS;
    auto pit = nodes_.find(parentOf(path));
    if (pit == nodes_.end()) return ZNONODE;
    if (pit->second.ephemeralOwner) return ZNOCHILDRENFOREPHEMERALS;
    zxid_++;
    if (logTxn) {
        Writer b;
        b.str(path);
        b.buffer(data);
        writeOpenAcl(b);
        b.boolean(ephemeralOwner != 0);
        b.i32(pit->second.cversion + 1);
        appendTxn(OP_CREATE, b.buf,
                  txnSession ? txnSession : ephemeralOwner);
    }
    Node n;
    n.data = data;
    n.ephemeralOwner = ephemeralOwner;
    n.czxid = n.mzxid = zxid_;
    n.ctime = n.mtime = wallMillis();
    nodes_[path] = std::move(n);
    pit = nodes_.find(parentOf(path));  // map may have rehashed
    pit->second.children.insert(baseOf(path));
    pit->second.cversion++;
    pit->second.pzxid = zxid_;
    fire(path, EV_NODE_CREATED, false, true);
    fire(parentOf(path), EV_NODE_CHILDREN_CHANGED, true, false);
    return ZOK;
}

int32_t Zkd::doSet(const std::string& path, const std::string& data,
                   int32_t version, int64_t txnSession) {
    auto it = nodes_.find(path);
    if (it == nodes_.end()) return ZNONODE;
    if (version != -1 && version != it->second.version)
        return ZBADVERSION;
    zxid_++;
    it->second.data = data;
    it->second.version++;
    it->second.mzxid = zxid_;
    it->second.mtime = wallMillis();
    Writer b;
    b.str(path);
    b.buffer(data);
    b.i32(it->second.version);
    appendTxn(OP_SETDATA, b.buf, txnSession);
    fire(path, EV_NODE_DATA_CHANGED, false, false);
    return ZOK;
}

int32_t Zkd::doDelete(const std::string& path, int32_t version,
                      int64_t txnSession) {
    auto it = nodes_.find(path);
    if (it == nodes_.end()) return ZNONODE;
    if (!it->second.children.empty()) return ZNOTEMPTY;
    if (version != -1 && version != it->second.version)
        return ZBADVERSION;
    zxid_++;
    Writer b;
    b.str(path);
    appendTxn(OP_DELETE, b.buf, txnSession);
    nodes_.erase(it);
    std::string parent = parentOf(path);
    auto pit = nodes_.find(parent);
    if (pit != nodes_.end()) {
        pit->second.children.erase(baseOf(path));
        pit->second.cversion++;
        pit->second.pzxid = zxid_;
        fire(parent, EV_NODE_CHILDREN_CHANGED, true, false);
    }
    fire(path, EV_NODE_DELETED, false, false);
    return ZOK;
}

void Zkd::reapEphemerals(int64_t sessionId) {
    if (sessionId == 0) return;
    std::vector<std::string> doomed;
    for (const auto& [p, n] : nodes_)
        if (n.ephemeralOwner == sessionId) doomed.push_back(p);
    std::sort(doomed.begin(), doomed.end(),
              [](const std::string& a, const std::string& b) {
                  return a.size() > b.size();
              });
    for (const auto& p : doomed) doDelete(p, -1, sessionId);
}

void Zkd::sweepSessions() {
    int64_t now = monotonicMillis();
    /* Real-ZK semantics: liveness comes from PACKETS (pings), not
     * from the TCP connection existing — a silent client's session
     * expires even while its socket stays open. lastSeenMs is
     * refreshed in op()/handshake() only. */
    std::vector<int64_t> dead;
    for (auto& [sid, s] : sessions_)
        if (now - s.lastSeenMs > s.timeoutMs) dead.push_back(sid);
    for (int64_t sid : dead) {
        log_.info({{"session", Json((int64_t)sid)}},
                  "session expired; reaping ephemerals");
        sessions_.erase(sid);
        reapEphemerals(sid);
        /* drop the session's connections: their next op would act on
         * an expired session */
        std::vector<Conn*> doomed;
        for (auto& [fd, c] : conns_)
            if (!c->closed && c->sessionId == sid)
                doomed.push_back(c.get());
        for (Conn* c : doomed) closeConn(c);
    }
}

bool Zkd::start() {
    if (!openLog()) {
        log_.error({{"dir", Json(dataDir_)}}, "cannot open txn log");
        return false;
    }
    listenFd_ =
        socket(AF_INET, SOCK_STREAM | SOCK_NONBLOCK | SOCK_CLOEXEC, 0);
    int one = 1;
    setsockopt(listenFd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    struct sockaddr_in sa {};
    sa.sin_family = AF_INET;
    sa.sin_port = htons(port_);
    inet_pton(AF_INET, host_.c_str(), &sa.sin_addr);
    if (bind(listenFd_, (struct sockaddr*)&sa, sizeof(sa)) != 0 ||
        listen(listenFd_, 64) != 0) {
        log_.error({{"port", Json((int)port_)}}, "zkd bind failed");
        return false;
    }
    socklen_t slen = sizeof(sa);
    getsockname(listenFd_, (struct sockaddr*)&sa, &slen);
    port_ = ntohs(sa.sin_port);
    loop_->addFd(listenFd_, EPOLLIN, [this](uint32_t) { onAccept(); });

    auto tick = std::make_shared<std::function<void()>>();
    *tick = [this, tick]() {
        sweepSessions();
        loop_->addTimer(1000, *tick);
    };
    loop_->addTimer(1000, *tick);
    return true;
}

void Zkd::onAccept() {
    while (true) {
        int fd = accept4(listenFd_, nullptr, nullptr,
                         SOCK_NONBLOCK | SOCK_CLOEXEC);
        if (fd < 0) return;
        int one = 1;
        setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
        auto c = std::make_shared<Conn>();
        c->fd = fd;
        conns_[fd] = c;
        loop_->addFd(fd, EPOLLIN,
                     [this, c](uint32_t ev) { onConn(c, ev); });
    }
}

void Zkd::closeConn(Conn* c) {
    if (c->closed) return;
    c->closed = true;
    loop_->delFd(c->fd);
    close(c->fd);
    conns_.erase(c->fd);
}

void Zkd::sendRaw(Conn* c, const std::string& payload) {
    Writer w;
    w.i32((int32_t)payload.size());
    c->out += w.buf;
    c->out += payload;
    flushConn(c);
}

void Zkd::reply(Conn* c, int32_t xid, int32_t err,
                const std::string& body) {
    Writer w;
    w.i32(xid);
    w.i64(zxid_);
    w.i32(err);
    w.buf += body;
    sendRaw(c, w.buf);
}

void Zkd::flushConn(Conn* c) {
    while (c->outOff < c->out.size() && !c->closed) {
        ssize_t nw = write(c->fd, c->out.data() + c->outOff,
                           c->out.size() - c->outOff);
        if (nw > 0) {
            c->outOff += (size_t)nw;
            continue;
        }
        if (nw < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
            if (!c->writeBlocked) {
                c->writeBlocked = true;
                loop_->modFd(c->fd, EPOLLIN | EPOLLOUT);
            }
            return;
        }
        closeConn(c);
        return;
    }
    if (c->outOff >= c->out.size()) {
        c->out.clear();
        c->outOff = 0;
    }
    if (c->writeBlocked && !c->closed) {
        c->writeBlocked = false;
        loop_->modFd(c->fd, EPOLLIN);
    }
}

bool Zkd::fourLetter(const std::shared_ptr<Conn>& c) {
    if (c->handshaken || c->in.size() < 4) return false;
    std::string cmd = c->in.substr(0, 4);
    std::string resp;
    if (cmd == "ruok") {
        resp = "imok";
    } else if (cmd == "stat" || cmd == "srvr") {
        char buf[512];
        snprintf(buf, sizeof(buf),
                 "zkd (binder-amd) single-node\n"
                 "Znodes: %zu\nSessions: %zu\nOps: %llu\n"
                 "Watches fired: %llu\nZxid: 0x%llx\nMode: standalone\n",
                 nodes_.size() - 1, sessions_.size(),
                 (unsigned long long)opsServed_,
                 (unsigned long long)watchesFired_,
                 (unsigned long long)zxid_);
        resp = buf;
    } else {
        return false;
    }
    c->out += resp;
    flushConn(c.get());
    closeConn(c.get());
    return true;
}

void Zkd::onConn(const std::shared_ptr<Conn>& c, uint32_t ev) {
    if (c->closed) return;
    if (ev & (EPOLLHUP | EPOLLERR)) {
        closeConn(c.get());
        return;
    }
    if (ev & EPOLLOUT) flushConn(c.get());
    if (c->closed || !(ev & EPOLLIN)) return;

    char buf[65536];
    while (true) {
        ssize_t nr = read(c->fd, buf, sizeof(buf));
        if (nr > 0) {
            c->in.append(buf, (size_t)nr);
            continue;
        }
        if (nr < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) break;
        closeConn(c.get());
        return;
    }

    if (!c->handshaken && fourLetter(c)) return;

    while (c->in.size() >= 4 && !c->closed) {
        Reader lr{(const uint8_t*)c->in.data(), 4};
        int32_t plen = lr.i32();
        if (plen < 0 || plen > (64 << 20)) {
            closeConn(c.get());
            return;
        }
        if (c->in.size() < 4 + (size_t)plen) break;
        handlePacket(c, (const uint8_t*)c->in.data() + 4, (size_t)plen);
        c->in.erase(0, 4 + (size_t)plen);
    }
}

void Zkd::handlePacket(const std::shared_ptr<Conn>& c, const uint8_t* p,
                       size_t n) {
    Reader r{p, n};
    if (!c->handshaken)
        handshake(c, r);
    else
        op(c, r);
}

void Zkd::handshake(const std::shared_ptr<Conn>& c, Reader& r) {
    r.i32();  // protocolVersion
    r.i64();  // lastZxidSeen
    int32_t timeout = r.i32();
    int64_t sid = r.i64();
    r.str();  // passwd
    /* optional readOnly byte (3.4.6+) ignored: 3.4.0-era clients
     * omit it (pinned by test_zk_golden) */
    if (sid != 0 && sessions_.count(sid) == 0) {
        /* expired: sessionId 0 / timeOut 0 */
        Writer w;
        w.i32(0);
        w.i32(0);
        w.i64(0);
        w.buffer(std::string(16, '\0'));
        w.boolean(false);
        sendRaw(c.get(), w.buf);
        return;
    }
    int32_t neg = std::min(timeout > 0 ? timeout : sessionTimeoutMs_,
                           sessionTimeoutMs_);
    if (sid == 0) {
        sid = nextSession_++;
        sessionsMade_++;
        zxid_++;
        Writer b;
        b.i32(neg);
        appendTxn(-10 /* createSession */, b.buf, sid);
    }
    sessions_[sid] = Session{sid, neg, monotonicMillis()};
    c->sessionId = sid;
    c->handshaken = true;
    Writer w;
    w.i32(0);
    w.i32(neg);
    w.i64(sid);
    w.buffer(std::string(16, '\x01'));
    w.boolean(false);
    sendRaw(c.get(), w.buf);
}

void Zkd::op(const std::shared_ptr<Conn>& c, Reader& r) {
    auto sit = sessions_.find(c->sessionId);
    if (sit != sessions_.end()) sit->second.lastSeenMs = monotonicMillis();

    int32_t xid = r.i32();
    if (xid == XID_PING) {
        reply(c.get(), XID_PING, ZOK);
        return;
    }
    int32_t opcode = r.i32();
    opsServed_++;

    switch (opcode) {
    case OP_CLOSE: {
        zxid_++;
        appendTxn(-11 /* closeSession */, "", c->sessionId);
        sessions_.erase(c->sessionId);
        reapEphemerals(c->sessionId);
        reply(c.get(), xid, ZOK);
        flushConn(c.get());
        closeConn(c.get());
        return;
    }
    case OP_GETDATA: {
        std::string path = r.str();
        bool watch = r.boolean();
        auto it = nodes_.find(path);
        if (it == nodes_.end()) {
            reply(c.get(), xid, ZNONODE);
            return;
        }
        if (watch) c->dataWatches.insert(path);
        Writer b;
        b.buffer(it->second.data);
        writeStat(b, it->second);
        reply(c.get(), xid, ZOK, b.buf);
        return;
    }
    case OP_GETCHILDREN:
    case OP_GETCHILDREN2: {
        std::string path = r.str();
        bool watch = r.boolean();
        auto it = nodes_.find(path);
        if (it == nodes_.end()) {
            reply(c.get(), xid, ZNONODE);
            return;
        }
        if (watch) c->childWatches.insert(path);
        Writer b;
        b.i32((int32_t)it->second.children.size());
        for (const auto& kid : it->second.children) b.str(kid);
        if (opcode == OP_GETCHILDREN2) writeStat(b, it->second);
        reply(c.get(), xid, ZOK, b.buf);
        return;
    }
    case OP_EXISTS: {
        std::string path = r.str();
        bool watch = r.boolean();
        auto it = nodes_.find(path);
        if (watch) {
            c->existsWatches.insert(path);
            if (it != nodes_.end()) c->dataWatches.insert(path);
        }
        if (it == nodes_.end()) {
            reply(c.get(), xid, ZNONODE);
        } else {
            Writer b;
            writeStat(b, it->second);
            reply(c.get(), xid, ZOK, b.buf);
        }
        return;
    }
    case OP_CREATE: {
        std::string path = r.str();
        std::string data = r.str();
        int32_t nacl = r.i32();
        for (int32_t i = 0; i < nacl && r.ok; ++i) {
            r.i32();
            r.str();
            r.str();
        }
        int32_t flags = r.i32();
        if (!r.ok) {
            reply(c.get(), xid, ZMARSHALLINGERROR);
            return;
        }
        if (flags & CREATE_SEQUENTIAL) {
            /* 10-digit zero-padded per-parent counter suffix */
            auto pit = nodes_.find(parentOf(path));
            int32_t seq = pit == nodes_.end() ? 0 : pit->second.cversion;
            char suf[16];
            snprintf(suf, sizeof(suf), "%010d", seq);
            path += suf;
        }
        int64_t owner =
            (flags & CREATE_EPHEMERAL) ? c->sessionId : 0;
        int32_t rc = doCreate(path, data, owner, c->sessionId, true);
        if (rc == ZOK) {
            Writer b;
            b.str(path);
            reply(c.get(), xid, rc, b.buf);
        } else {
            reply(c.get(), xid, rc);
        }
        return;
    }
    case OP_SETDATA: {
        std::string path = r.str();
        std::string data = r.str();
        int32_t version = r.i32();
        int32_t rc = doSet(path, data, version, c->sessionId);
        if (rc == ZOK) {
            Writer b;
            writeStat(b, nodes_[path]);
            reply(c.get(), xid, rc, b.buf);
        } else {
            reply(c.get(), xid, rc);
        }
        return;
    }
    case OP_DELETE: {
        std::string path = r.str();
        int32_t version = r.i32();
        reply(c.get(), xid, doDelete(path, version, c->sessionId));
        return;
    }
    case OP_SYNC: {
        std::string path = r.str();
        Writer b;
        b.str(path);
        reply(c.get(), xid, ZOK, b.buf);
        return;
    }
    default:
        reply(c.get(), xid, ZUNIMPLEMENTED);
    }
}

}  // namespace

int main(int argc, char** argv) {
    signal(SIGPIPE, SIG_IGN);
    const char* lvl = getenv("LOG_LEVEL");
    Logger log("zkd", logLevelFromName(lvl ? lvl : "info",
                                       LogLevel::Info));
    std::string host = "127.0.0.1";
    uint16_t port = 2181;
    std::string dataDir;
    int32_t timeoutMs = 30000;
    int c;
    while ((c = getopt(argc, argv, "hH:p:d:t:")) != -1) {
        switch (c) {
        case 'H': host = optarg; break;
        case 'p': port = (uint16_t)atoi(optarg); break;
        case 'd': dataDir = optarg; break;
        case 't': timeoutMs = atoi(optarg); break;
        case 'h':
        default:
            fprintf(stderr,
                    "usage: zkd [-H host] [-p port] [-d data-dir] "
                    "[-t session-timeout-ms]\n");
            return c == 'h' ? 0 : 1;
        }
    }
    EventLoop loop;
    Zkd zkd(&loop, log, host, port, dataDir, timeoutMs);
    if (!zkd.start()) return 1;

    /* clean shutdown on SIGTERM/SIGINT (flushes the txn log cleanly;
     * also lets coverage/profiling builds write their data) */
    sigset_t mask;
    sigemptyset(&mask);
    sigaddset(&mask, SIGTERM);
    sigaddset(&mask, SIGINT);
    sigprocmask(SIG_BLOCK, &mask, nullptr);
    int sfd = signalfd(-1, &mask, SFD_NONBLOCK | SFD_CLOEXEC);
    loop.addFd(sfd, EPOLLIN, [&loop, sfd](uint32_t) {
        struct signalfd_siginfo si;
        while (read(sfd, &si, sizeof(si)) == sizeof(si)) {
        }
        loop.stop();
    });

    printf("zkd listening on %s:%u, %zu nodes restored\n", host.c_str(),
           (unsigned)zkd.boundPort(), zkd.nodeCount());
    fflush(stdout);
    loop.run();
    return 0;
}
