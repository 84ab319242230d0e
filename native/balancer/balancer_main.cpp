/*
 * binder-balancer: native L4 DNS load balancer (mname-balancer
 * equivalent; SURVEY.md §2 row 11 — the reference submodule is not
 * vendored, so this is a from-scratch design with the same observable
 * capabilities):
 *
 *   - listens on :port UDP + TCP, fans out to backend binderd processes
 *     over UNIX sockets discovered in a socket directory (one socket per
 *     backend; presence = registration, unlink = drain);
 *   - per-remote-IP affinity: each client IP is pinned to a backend
 *     (bin/balstat shows the reference tracks backend_t/remote_t AVLs);
 *   - per-backend health (be_ok): connect failures / PING timeouts mark
 *     a backend down and its remotes are reassigned;
 *   - preserves original client address/port/family across the hop
 *     (bsock1 framing, see protocol.hpp);
 *   - a stats UNIX socket replaces mdb introspection: connecting dumps
 *     one JSON object (backends, remotes, counters) and closes —
 *     consumed by `binder-amd balstat`.
 *
 * Flags: -p port (default 53), -H host, -s socket-dir (default
 * /var/run/binder/sockets), -S stats-socket path, -r rescan interval ms.
 */
#include <arpa/inet.h>
#include <dirent.h>
#include <netinet/in.h>
#include <netinet/udp.h>
#include <signal.h>
#include <sys/epoll.h>
#include <sys/signalfd.h>
#include <sys/socket.h>
#include <sys/stat.h>
#include <sys/un.h>
#include <unistd.h>

#include <algorithm>
#include <cstdlib>
#include <cstring>
#include <map>
#include <unordered_map>
#include <chrono>
#include <memory>
#include <unordered_map>
#include <mutex>
#include <thread>
#include <set>
#include <string>
#include <vector>

#include "../common/json.hpp"
#include "../common/log.hpp"
#include "../common/loop.hpp"
#include "protocol.hpp"

using namespace bamd;

namespace {

/* Remote-assignment counts shared across SO_REUSEPORT workers, keyed by
 * backend socket path: without this each worker balances only its own
 * remotes and the aggregate backend load skews badly. Assignment is the
 * cold path, so a mutex is fine. */
struct GlobalPins {
    std::mutex m;
    std::map<std::string, int> counts;
    void add(const std::string& path, int d) {
        std::lock_guard<std::mutex> g(m);
        counts[path] += d;
        if (counts[path] < 0) counts[path] = 0;
    }
    int get(const std::string& path) {
        std::lock_guard<std::mutex> g(m);
        auto it = counts.find(path);
        return it == counts.end() ? 0 : it->second;
    }
};
GlobalPins g_pins;

/* Per-worker stats snapshots, refreshed each sweep; worker 0 serves
 * the merged view on the stats socket (an mdb-free balstat needs a
 * whole-process picture, not one shard's). */
struct StatsHub {
    std::mutex m;
    std::map<int, std::string> byWorker;  // worker id -> JSON fragment
    void publish(int worker, std::string json) {
        std::lock_guard<std::mutex> g(m);
        byWorker[worker] = std::move(json);
    }
    std::vector<std::string> all() {
        std::lock_guard<std::mutex> g(m);
        std::vector<std::string> out;
        for (auto& [w, j] : byWorker) out.push_back(j);
        return out;
    }
};
StatsHub g_stats;

/*
 * Shared socket-directory scanner: opendir/stat on the serving loop
 * measured as rare ~10 ms stalls on the bench boxes' filesystem
 * (profiles stall1.jsonl: stalls vanish when the tick slows), and
 * every worker scanning the same directory each tick multiplies the
 * exposure. One helper thread scans; workers read the snapshot.
 */
struct DirScanner {
    std::mutex m;
    std::set<std::string> paths;
    std::atomic<uint64_t> gen{0};
    std::thread thread;
    std::atomic<bool> stop{false};

    void start(std::string dir, int intervalMs) {
        thread = std::thread([this, dir = std::move(dir),
                              intervalMs]() {
            while (!stop.load()) {
                std::set<std::string> seen;
                DIR* d = opendir(dir.c_str());
                if (d != nullptr) {
                    struct dirent* ent;
                    while ((ent = readdir(d)) != nullptr) {
                        std::string name = ent->d_name;
                        if (name == "." || name == "..") continue;
                        std::string full = dir + "/" + name;
                        struct stat st;
                        if (stat(full.c_str(), &st) == 0 &&
                            S_ISSOCK(st.st_mode))
                            seen.insert(full);
                    }
                    closedir(d);
                }
                {
                    std::lock_guard<std::mutex> g(m);
                    if (paths != seen) {
                        paths = std::move(seen);
                    }
                }
                gen.fetch_add(1);
                for (int i = 0; i < intervalMs / 10 && !stop.load();
                     ++i)
                    std::this_thread::sleep_for(
                        std::chrono::milliseconds(10));
            }
        });
    }
    std::set<std::string> snapshot() {
        std::lock_guard<std::mutex> g(m);
        return paths;
    }
    void shutdown() {
        stop.store(true);
        if (thread.joinable()) thread.join();
    }
};
DirScanner g_scanner;

/*
 * In-flight request table: a flat power-of-two ring indexed by
 * reqId & (kPendingSlots-1). reqIds are allocated sequentially per
 * backend, so consecutive inserts hit consecutive slots with zero
 * hashing and zero allocation (gprof showed ~25% of balancer user CPU
 * in the previous unordered_map's node insert/erase at saturation).
 * A slot is only ever re-hit kPendingSlots ids later (~0.1 s at
 * saturation): colliding with a still-armed entry means that reply is
 * long past any client deadline, so overwriting it is equivalent to
 * the TTL expiry that would reap it anyway. Late replies whose slot
 * was reused fail the reqId check and are dropped.
 */
struct PendingSlot {
    uint32_t reqId = 0;  /* 0 = empty (ids start at 1) */
    uint32_t expiresAtMs;  /* truncated monotonicMillis */
    int tcpFd;
    uint16_t srcPort;  /* host order */
    uint8_t family;    /* 4 or 6 */
    bool tcp;
    uint8_t addr[16];
};
static_assert(sizeof(PendingSlot) <= 32, "keep the ring compact");

/* Ring size > kMaxPending by default; -q overrides (power of two;
 * tiny values let tests exercise the overwrite accounting). */
size_t g_pendingSlots = 32768;

struct Backend {
    int id;
    std::string path;
    int fd = -1;
    bool ok = false;
    std::string in, out;
    bool writeBlocked = false;
    uint32_t nextReq = 1;
    std::unique_ptr<PendingSlot[]> pending;  /* lazily sized ring */
    size_t pendingCount = 0;
    int64_t lastPongAt = 0;
    int64_t pingSentAt = 0;
    uint64_t queries = 0;
    uint64_t replies = 0;
    /* live pending-ring slots lost to an overwrite (ring collision
     * with >kPendingSlots in flight): the reply for the overwritten
     * request will be silently dropped, so overloads must be visible
     * on the stats socket, not silent */
    uint64_t overwrites = 0;
    size_t outOff = 0;  /* consumed prefix of `out` (flush cursor) */
    size_t sweepCursor = 0;  /* incremental pending-expiry scan */
    size_t remotes = 0;

    PendingSlot* slotFor(uint32_t reqId) {
        if (!pending) {
            pending = std::make_unique<PendingSlot[]>(g_pendingSlots);
            for (size_t i = 0; i < g_pendingSlots; ++i)
                pending[i].reqId = 0;
        }
        return &pending[reqId & (g_pendingSlots - 1)];
    }
    void clearPending() {
        if (pending)
            for (size_t i = 0; i < g_pendingSlots; ++i)
                pending[i].reqId = 0;
        pendingCount = 0;
    }
};

struct TcpClient {
    int fd;
    std::string in, out;
    bool writeBlocked = false;
    bool closed = false;
    struct sockaddr_storage src;
    socklen_t srcLen;
    int backendId = -1;
    int64_t lastActivityMs = 0;
};

class Balancer {
  public:
    Balancer(EventLoop* loop, Logger log, std::string host, uint16_t port,
             std::string sockDir, std::string statsPath, int rescanMs,
             bool reusePort = false, int workerId = 0, int nWorkers = 1)
        : loop_(loop), log_(std::move(log)), host_(std::move(host)),
          port_(port), sockDir_(std::move(sockDir)),
          statsPath_(std::move(statsPath)), rescanMs_(rescanMs),
          reusePort_(reusePort), workerId_(workerId),
          nWorkers_(nWorkers) {}

    bool start();
    void stop();

  private:
    void rescan();
    void connectBackend(Backend* be);
    void backendDown(Backend* be);
    void onBackendEvent(std::shared_ptr<Backend> be, uint32_t ev);
    void backendFlush(Backend* be);
    Backend* pickBackend(const std::string& remoteIp);
    Backend* pickBackendFast(const struct sockaddr_storage& ss);
    Backend* chooseLeastLoaded();
    void onUdpReadable();
    void handleUdpQuery(const uint8_t* dns, size_t dnsLen,
                        const struct sockaddr_storage& src,
                        int64_t expiry, std::set<Backend*>& touched);
    void onTcpAccept();
    void onTcpClient(std::shared_ptr<TcpClient> c, uint32_t ev);
    void tcpClientFlush(TcpClient* c);
    void onStatsAccept();
    void sweep();
    Json snapshot() const;

    EventLoop* loop_;
    Logger log_;
    std::string host_;
    uint16_t port_;
    std::string sockDir_;
    std::string statsPath_;
    int rescanMs_;
    bool reusePort_ = false;
    int workerId_ = 0;
    int nWorkers_ = 1;

    int udpFd_ = -1, tcpFd_ = -1, statsFd_ = -1;
    std::map<std::string, std::shared_ptr<Backend>> backends_;  // by path
    std::map<int, std::shared_ptr<Backend>> backendsById_;
    std::map<std::string, int> remotes_;  // remote ip -> backend id (stats)
    /* hot-path affinity: FNV of raw addr bytes -> backend id (avoids
     * inet_ntop + string alloc per packet; a hash collision only means
     * two IPs share a pin, which is harmless) */
    std::unordered_map<uint64_t, int> remotesFast_;
    std::map<int, std::shared_ptr<TcpClient>> tcpClients_;
    int nextBackendId_ = 1;
    int salvageDepth_ = 0;  /* bounds backendDown->salvage recursion */
    bool replyGso_ = true;  /* UDP_SEGMENT reply runs (auto-fallback) */
    uint64_t udpQueries_ = 0, udpReplies_ = 0, drops_ = 0;

    static constexpr size_t kMaxRemotes = 262144;
    static constexpr size_t kMaxPending = 16384;
    static constexpr int64_t kReplyTtlMs = 3000;
    static constexpr int64_t kPingIntervalMs = 2000;
    static constexpr int64_t kPingTimeoutMs = 6000;

    /* per-instance batch I/O arenas — workers are threads, so these
     * must NOT be static (a shared-static race here collapsed
     * multi-worker throughput). RX buffers are GRO-sized: with UDP_GRO
     * on the ingress socket one "message" may be a coalesced
     * super-packet of up to 64 KB of equal-size datagrams. */
    static constexpr int kBatch = 128;
    static constexpr size_t kRxBufSz = 65536;
    std::vector<uint8_t> rxArena_ =
        std::vector<uint8_t>((size_t)kBatch * kRxBufSz);
    struct mmsghdr rxHdrs_[kBatch];
    struct iovec rxIovs_[kBatch];
    struct sockaddr_storage rxAddrs_[kBatch];
    char rxCtrl_[kBatch][CMSG_SPACE(sizeof(uint16_t))];
    struct mmsghdr replyHdrs_[kBatch];
    struct iovec replyIovs_[kBatch];
    struct sockaddr_storage replyAddrs_[kBatch];
};

bool Balancer::start() {
    /* UDP */
    bool v6 = host_.empty() || host_.find(':') != std::string::npos;
    udpFd_ = socket(v6 ? AF_INET6 : AF_INET,
                    SOCK_DGRAM | SOCK_NONBLOCK | SOCK_CLOEXEC, 0);
    int one = 1;
    setsockopt(udpFd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    if (reusePort_)
        setsockopt(udpFd_, SOL_SOCKET, SO_REUSEPORT, &one, sizeof(one));
    int sz = 8 << 20;
    setsockopt(udpFd_, SOL_SOCKET, SO_RCVBUF, &sz, sizeof(sz));
    setsockopt(udpFd_, SOL_SOCKET, SO_SNDBUF, &sz, sizeof(sz));
    /* Accept GRO-coalesced batches: a GSO-batching client (dnsblast,
     * or any sender using UDP_SEGMENT) then costs the kernel one
     * loopback traversal per BURST instead of per query. Harmless if
     * unsupported or no sender uses it. */
    setsockopt(udpFd_, SOL_UDP, UDP_GRO, &one, sizeof(one));
    tcpFd_ = socket(v6 ? AF_INET6 : AF_INET,
                    SOCK_STREAM | SOCK_NONBLOCK | SOCK_CLOEXEC, 0);
    setsockopt(tcpFd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    if (reusePort_)
        setsockopt(tcpFd_, SOL_SOCKET, SO_REUSEPORT, &one, sizeof(one));

    auto doBind = [&](int fd) {
        if (v6) {
            int zero = 0;
            setsockopt(fd, IPPROTO_IPV6, IPV6_V6ONLY,
                       host_.empty() ? &zero : &one, sizeof(int));
            struct sockaddr_in6 sa {};
            sa.sin6_family = AF_INET6;
            sa.sin6_port = htons(port_);
            if (host_.empty())
                sa.sin6_addr = in6addr_any;
            else
                inet_pton(AF_INET6, host_.c_str(), &sa.sin6_addr);
            return bind(fd, (struct sockaddr*)&sa, sizeof(sa)) == 0;
        }
        struct sockaddr_in sa {};
        sa.sin_family = AF_INET;
        sa.sin_port = htons(port_);
        inet_pton(AF_INET, host_.c_str(), &sa.sin_addr);
        return bind(fd, (struct sockaddr*)&sa, sizeof(sa)) == 0;
    };
    if (!doBind(udpFd_) || !doBind(tcpFd_) || listen(tcpFd_, 512) != 0) {
        log_.error({{"port", Json((int)port_)}}, "balancer bind failed");
        return false;
    }
    loop_->addFd(udpFd_, EPOLLIN, [this](uint32_t) { onUdpReadable(); });
    loop_->addFd(tcpFd_, EPOLLIN, [this](uint32_t) { onTcpAccept(); });

    if (!statsPath_.empty()) {
        statsFd_ = socket(AF_UNIX, SOCK_STREAM | SOCK_NONBLOCK |
                          SOCK_CLOEXEC, 0);
        struct sockaddr_un sa {};
        sa.sun_family = AF_UNIX;
        snprintf(sa.sun_path, sizeof(sa.sun_path), "%s",
                 statsPath_.c_str());
        unlink(sa.sun_path);
        if (bind(statsFd_, (struct sockaddr*)&sa, sizeof(sa)) == 0 &&
            listen(statsFd_, 8) == 0) {
            loop_->addFd(statsFd_, EPOLLIN,
                         [this](uint32_t) { onStatsAccept(); });
        } else {
            log_.warn({{"path", Json(statsPath_)}},
                      "could not bind stats socket");
        }
    }

    rescan();
    /* periodic rescans + sweeps, staggered across workers so the
     * ticks never stall every shard in the same instant */
    auto schedule = std::make_shared<std::function<void()>>();
    *schedule = [this, schedule]() {
        rescan();
        sweep();
        loop_->addTimer(rescanMs_, *schedule);
    };
    int64_t offset =
        nWorkers_ > 0 ? (rescanMs_ * workerId_) / nWorkers_ : 0;
    loop_->addTimer(rescanMs_ + offset, *schedule);
    log_.info({{"port", Json((int)port_)}, {"dir", Json(sockDir_)}},
              "balancer started");
    return true;
}

void Balancer::stop() {
    for (int* fd : {&udpFd_, &tcpFd_, &statsFd_}) {
        if (*fd >= 0) {
            loop_->delFd(*fd);
            close(*fd);
            *fd = -1;
        }
    }
    if (!statsPath_.empty()) unlink(statsPath_.c_str());
}

void Balancer::rescan() {
    /* no filesystem access on the serving loop: the shared scanner
     * thread maintains the socket set */
    std::set<std::string> seen = g_scanner.snapshot();
    {
        for (const std::string& full : seen) {
            if (backends_.count(full) == 0) {
                auto be = std::make_shared<Backend>();
                be->id = nextBackendId_++;
                be->path = full;
                backends_[full] = be;
                backendsById_[be->id] = be;
                log_.info({{"path", Json(full)},
                           {"id", Json((int64_t)be->id)}},
                          "backend discovered");
                connectBackend(be.get());
            } else {
                Backend* be = backends_[full].get();
                if (be->fd < 0) connectBackend(be);  // retry
            }
        }
    }
    /* removed sockets => drain (main.js:181-193 unlink-on-SIGTERM) */
    std::vector<std::string> gone;
    for (auto& [path, be] : backends_)
        if (seen.count(path) == 0) gone.push_back(path);
    for (const auto& path : gone) {
        auto be = backends_[path];
        log_.info({{"path", Json(path)}}, "backend removed (drained)");
        backendDown(be.get());
        backendsById_.erase(be->id);
        backends_.erase(path);
    }
}

void Balancer::connectBackend(Backend* be) {
    int fd = socket(AF_UNIX, SOCK_STREAM | SOCK_NONBLOCK | SOCK_CLOEXEC, 0);
    if (fd < 0) return;
    int sz = 4 << 20;
    setsockopt(fd, SOL_SOCKET, SO_SNDBUF, &sz, sizeof(sz));
    setsockopt(fd, SOL_SOCKET, SO_RCVBUF, &sz, sizeof(sz));
    struct sockaddr_un sa {};
    sa.sun_family = AF_UNIX;
    snprintf(sa.sun_path, sizeof(sa.sun_path), "%s", be->path.c_str());
    int rv = connect(fd, (struct sockaddr*)&sa, sizeof(sa));
    if (rv != 0 && errno != EINPROGRESS) {
        close(fd);
        be->ok = false;
        return;
    }
    be->fd = fd;
    be->ok = true;  // optimistic; PING confirms
    be->lastPongAt = monotonicMillis();
    auto self = backendsById_[be->id];
    loop_->addFd(fd, EPOLLIN, [this, self](uint32_t ev) {
        onBackendEvent(self, ev);
    });
}

void Balancer::backendDown(Backend* be) {
    if (be->fd >= 0) {
        loop_->delFd(be->fd);
        close(be->fd);
        be->fd = -1;
    }
    be->ok = false;
    be->in.clear();
    /* Salvage: frames we queued but never flushed into the dead
     * connection can be re-dispatched to a healthy backend instead of
     * silently dropped — the client's in-flight query then survives a
     * backend crash (a query already inside the dead socket's buffer
     * is genuinely lost; the client retries). */
    std::string unsent = be->out.substr(be->outOff);
    be->out.clear();
    be->outOff = 0;
    be->writeBlocked = false;
    be->clearPending();
    /* unpin remotes so they re-pick a healthy backend */
    for (auto it = remotes_.begin(); it != remotes_.end();) {
        if (it->second == be->id)
            it = remotes_.erase(it);
        else
            ++it;
    }
    for (auto it = remotesFast_.begin(); it != remotesFast_.end();) {
        if (it->second == be->id)
            it = remotesFast_.erase(it);
        else
            ++it;
    }
    g_pins.add(be->path, -(int)be->remotes);
    be->remotes = 0;

    if (!unsent.empty() && salvageDepth_ < 4) {
        salvageDepth_++;
        std::set<Backend*> touched;
        int64_t expiry = monotonicMillis() + kReplyTtlMs;
        size_t off = 0;
        while (unsent.size() - off >= bsock::kHeaderLen) {
            const uint8_t* h = (const uint8_t*)unsent.data() + off;
            uint32_t plen = bsock::getU32(h + 2);
            if (h[0] != bsock::kMagic ||
                plen > bsock::kMaxPayload ||
                unsent.size() - off < bsock::kHeaderLen + plen)
                break;
            const uint8_t* payload = h + bsock::kHeaderLen;
            /* re-dispatch UDP QUERY frames (layout: protocol.hpp);
             * PINGs are dropped, TCP queries have their reply route
             * in the dead backend's pending ring so the client
             * retries those */
            if (h[1] == bsock::FRAME_QUERY &&
                plen >= bsock::kQueryHeadLen && payload[5] == 0) {
                struct sockaddr_storage src {};
                uint16_t port =
                    (uint16_t)(payload[6] | (payload[7] << 8));
                if (payload[4] == 4) {
                    auto* sa = (struct sockaddr_in*)&src;
                    sa->sin_family = AF_INET;
                    sa->sin_port = htons(port);
                    memcpy(&sa->sin_addr, payload + 8, 4);
                } else {
                    auto* sa = (struct sockaddr_in6*)&src;
                    sa->sin6_family = AF_INET6;
                    sa->sin6_port = htons(port);
                    memcpy(&sa->sin6_addr, payload + 8, 16);
                }
                handleUdpQuery(payload + bsock::kQueryHeadLen,
                               plen - bsock::kQueryHeadLen, src,
                               expiry, touched);
            }
            off += bsock::kHeaderLen + plen;
        }
        for (Backend* tb : touched)
            if (tb != be) backendFlush(tb);
        salvageDepth_--;
    }
}

void Balancer::backendFlush(Backend* be) {
    while (be->outOff < be->out.size() && be->fd >= 0) {
        ssize_t nw = write(be->fd, be->out.data() + be->outOff,
                           be->out.size() - be->outOff);
        if (nw > 0) {
            be->outOff += (size_t)nw;
            continue;
        }
        if (nw < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
            /* compact occasionally so a long-blocked backend does not
             * keep a consumed prefix resident (amortized O(1) vs the
             * O(n) memmove-per-write of erase(0, nw)) */
            if (be->outOff > (1u << 20)) {
                be->out.erase(0, be->outOff);
                be->outOff = 0;
            }
            if (!be->writeBlocked) {
                be->writeBlocked = true;
                loop_->modFd(be->fd, EPOLLIN | EPOLLOUT);
            }
            return;
        }
        backendDown(be);
        return;
    }
    if (be->outOff >= be->out.size()) {
        be->out.clear();
        be->outOff = 0;
    }
    if (be->writeBlocked && be->fd >= 0) {
        be->writeBlocked = false;
        loop_->modFd(be->fd, EPOLLIN);
    }
}

void Balancer::onBackendEvent(std::shared_ptr<Backend> be, uint32_t ev) {
    if (be->fd < 0) return;
    if (ev & (EPOLLHUP | EPOLLERR)) {
        log_.warn({{"path", Json(be->path)}}, "backend connection lost");
        backendDown(be.get());
        return;
    }
    if (ev & EPOLLOUT) backendFlush(be.get());
    if (!(ev & EPOLLIN)) return;

    /* bounded batch per event (level-triggered epoll re-fires):
     * draining a many-MB reply backlog in one go stalls the worker's
     * other duties for milliseconds */
    constexpr size_t kMaxReadPerEvent = 256 * 1024;
    size_t got = 0;
    char buf[65536];
    while (be->fd >= 0 && got < kMaxReadPerEvent) {
        ssize_t nr = read(be->fd, buf, sizeof(buf));
        if (nr > 0) {
            be->in.append(buf, (size_t)nr);
            got += (size_t)nr;
            continue;
        }
        if (nr < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) break;
        log_.warn({{"path", Json(be->path)}}, "backend closed connection");
        backendDown(be.get());
        return;
    }

    /* Walk complete frames without per-frame erase; UDP replies are
     * batched (a syscall per reply was the balancer's top cost at
     * high QPS), and — because worker->backend affinity slices give
     * each worker few flows — consecutive same-destination equal-size
     * replies form runs that go out as ONE UDP_SEGMENT super-packet
     * (multi-iov, no copy): the reply hop's kernel cost is per RUN,
     * not per packet. Falls back to sendmmsg when GSO is refused. */
    constexpr int kReplyBatch = kBatch;
    struct mmsghdr* rh = replyHdrs_;
    struct iovec* riov = replyIovs_;
    struct sockaddr_storage* raddr = replyAddrs_;
    int nReply = 0;
    int order[kReplyBatch];
    auto sameDest = [&](int a, int b) {
        if (raddr[a].ss_family != raddr[b].ss_family) return false;
        if (raddr[a].ss_family == AF_INET) {
            auto* x = (const struct sockaddr_in*)&raddr[a];
            auto* y = (const struct sockaddr_in*)&raddr[b];
            return x->sin_port == y->sin_port &&
                   x->sin_addr.s_addr == y->sin_addr.s_addr;
        }
        auto* x = (const struct sockaddr_in6*)&raddr[a];
        auto* y = (const struct sockaddr_in6*)&raddr[b];
        return x->sin6_port == y->sin6_port &&
               memcmp(&x->sin6_addr, &y->sin6_addr, 16) == 0;
    };
    auto flushReplies = [&]() {
        if (nReply == 0) return;
        for (int i = 0; i < nReply; ++i) order[i] = i;
        std::sort(order, order + nReply, [&](int a, int b) {
            int c = memcmp(&raddr[a], &raddr[b], sizeof(raddr[a]));
            if (c != 0) return c < 0;
            return riov[a].iov_len > riov[b].iov_len;
        });
        int i = 0;
        int nSingle = 0;
        static thread_local std::vector<struct iovec> runIovs;
        while (i < nReply) {
            /* run: same destination, equal sizes (a shorter one may
             * close the run — UDP_SEGMENT's trailing segment) */
            int j = i + 1;
            size_t seg = riov[order[i]].iov_len;
            /* one UDP datagram caps at ~64KB: bound the run so large
             * replies can never push the super-packet past it */
            size_t maxRun = seg > 0 ? 60000 / seg : 1;
            if (maxRun > 48) maxRun = 48;
            if (maxRun < 1) maxRun = 1;
            while (j < nReply && (size_t)(j - i) < maxRun &&
                   sameDest(order[i], order[j]) &&
                   (riov[order[j]].iov_len == seg ||
                    (riov[order[j]].iov_len < seg &&
                     (j + 1 == nReply ||
                      !sameDest(order[i], order[j + 1])))))
                ++j;
            if (replyGso_ && j - i >= 2) {
                runIovs.clear();
                for (int k = i; k < j; ++k)
                    runIovs.push_back(riov[order[k]]);
                struct msghdr mh {};
                mh.msg_name = &raddr[order[i]];
                mh.msg_namelen =
                    raddr[order[i]].ss_family == AF_INET
                        ? sizeof(struct sockaddr_in)
                        : sizeof(struct sockaddr_in6);
                mh.msg_iov = runIovs.data();
                mh.msg_iovlen = runIovs.size();
                char cbuf[CMSG_SPACE(sizeof(uint16_t))] = {0};
                mh.msg_control = cbuf;
                mh.msg_controllen = sizeof(cbuf);
                struct cmsghdr* cm = CMSG_FIRSTHDR(&mh);
                cm->cmsg_level = SOL_UDP;
                cm->cmsg_type = UDP_SEGMENT;
                cm->cmsg_len = CMSG_LEN(sizeof(uint16_t));
                uint16_t gso = (uint16_t)seg;
                memcpy(CMSG_DATA(cm), &gso, sizeof(gso));
                if (sendmsg(udpFd_, &mh, 0) < 0) {
                    if (errno == EINVAL || errno == EIO ||
                        errno == ENOTSUP)
                        replyGso_ = false;  /* disable permanently */
                    /* ANY failure: the whole run would be lost —
                     * resend its replies individually */
                    for (int k = i; k < j; ++k) {
                        struct msghdr m1 {};
                        m1.msg_name = mh.msg_name;
                        m1.msg_namelen = mh.msg_namelen;
                        m1.msg_iov = &riov[order[k]];
                        m1.msg_iovlen = 1;
                        ssize_t r1 = sendmsg(udpFd_, &m1, 0);
                        (void)r1;
                    }
                }
            } else {
                /* singles re-packed into a sendmmsg batch */
                for (int k = i; k < j; ++k) {
                    memset(&rh[nSingle], 0, sizeof(rh[nSingle]));
                    rh[nSingle].msg_hdr.msg_iov = &riov[order[k]];
                    rh[nSingle].msg_hdr.msg_iovlen = 1;
                    rh[nSingle].msg_hdr.msg_name = &raddr[order[k]];
                    rh[nSingle].msg_hdr.msg_namelen =
                        raddr[order[k]].ss_family == AF_INET
                            ? sizeof(struct sockaddr_in)
                            : sizeof(struct sockaddr_in6);
                    nSingle++;
                }
            }
            i = j;
        }
        int sent = 0;
        while (sent < nSingle) {
            int rv = sendmmsg(udpFd_, rh + sent, nSingle - sent, 0);
            if (rv <= 0) break;
            sent += rv;
        }
        udpReplies_ += (uint64_t)nReply;
        nReply = 0;
    };

    size_t consumed = 0;
    while (be->in.size() - consumed >= bsock::kHeaderLen) {
        const uint8_t* h = (const uint8_t*)be->in.data() + consumed;
        if (h[0] != bsock::kMagic) {
            backendDown(be.get());
            return;
        }
        uint8_t type = h[1];
        uint32_t plen = bsock::getU32(h + 2);
        if (plen > bsock::kMaxPayload) {
            backendDown(be.get());
            return;
        }
        if (be->in.size() - consumed < bsock::kHeaderLen + plen) break;
        const uint8_t* payload = h + bsock::kHeaderLen;

        if (type == bsock::FRAME_PONG) {
            be->lastPongAt = monotonicMillis();
            be->ok = true;
        } else if (type == bsock::FRAME_REPLY && plen >= 4) {
            uint32_t reqId = bsock::getU32(payload);
            PendingSlot* pr = be->slotFor(reqId);
            if (pr->reqId == reqId) {
                const uint8_t* dns = payload + 4;
                size_t dnsLen = plen - 4;
                if (pr->tcp) {
                    auto cit = tcpClients_.find(pr->tcpFd);
                    if (cit != tcpClients_.end() &&
                        !cit->second->closed) {
                        TcpClient* c = cit->second.get();
                        c->out.push_back((char)(dnsLen >> 8));
                        c->out.push_back((char)dnsLen);
                        c->out.append((const char*)dns, dnsLen);
                        tcpClientFlush(c);
                    }
                } else {
                    if (nReply == kReplyBatch) flushReplies();
                    socklen_t slen;
                    if (pr->family == 4) {
                        auto* sa = (struct sockaddr_in*)&raddr[nReply];
                        /* zero the FULL storage: the reply sorter
                         * memcmps whole sockaddr_storage entries */
                        memset(&raddr[nReply], 0, sizeof(raddr[0]));
                        sa->sin_family = AF_INET;
                        sa->sin_port = htons(pr->srcPort);
                        memcpy(&sa->sin_addr, pr->addr, 4);
                        slen = sizeof(*sa);
                    } else {
                        auto* sa =
                            (struct sockaddr_in6*)&raddr[nReply];
                        memset(&raddr[nReply], 0, sizeof(raddr[0]));
                        sa->sin6_family = AF_INET6;
                        sa->sin6_port = htons(pr->srcPort);
                        memcpy(&sa->sin6_addr, pr->addr, 16);
                        slen = sizeof(*sa);
                    }
                    (void)slen;
                    riov[nReply] = {const_cast<uint8_t*>(dns), dnsLen};
                    nReply++;
                }
                be->replies++;
                pr->reqId = 0;
                be->pendingCount--;
            }
        }
        consumed += bsock::kHeaderLen + plen;
    }
    flushReplies();
    be->in.erase(0, consumed);
}

Backend* Balancer::chooseLeastLoaded() {
    /*
     * Affinity-aware worker->backend mapping: each SO_REUSEPORT worker
     * pins its remotes to a slice of the backends (still falling back
     * to any healthy backend when the slice is down). Concentrating a
     * worker's traffic on few backends makes every backend write/read
     * carry a full batch instead of 1/Nth of one — the chain is
     * syscall-bound under the box's CPU quota, so emptier batches
     * directly cost throughput. Backend load stays balanced because
     * the kernel spreads flows evenly across workers and slices cover
     * all backends.
     */
    Backend* best = nullptr;
    int bestCount = 0;
    bool bestPreferred = false;
    int nb = (int)backendsById_.size();
    int idx = 0;
    for (auto& [id, be] : backendsById_) {
        int i = idx++;
        if (!be->ok || be->fd < 0) continue;
        bool preferred =
            nWorkers_ >= nb ? (workerId_ % (nb > 0 ? nb : 1)) == i
                            : (i % nWorkers_) == workerId_;
        int c = g_pins.get(be->path);
        if (best == nullptr || (preferred && !bestPreferred) ||
            (preferred == bestPreferred && c < bestCount)) {
            best = be.get();
            bestCount = c;
            bestPreferred = preferred;
        }
    }
    return best;
}

Backend* Balancer::pickBackend(const std::string& remoteIp) {
    auto it = remotes_.find(remoteIp);
    if (it != remotes_.end()) {
        auto bit = backendsById_.find(it->second);
        if (bit != backendsById_.end() && bit->second->ok)
            return bit->second.get();
        remotes_.erase(it);
    }
    Backend* best = chooseLeastLoaded();
    if (best != nullptr) {
        remotes_[remoteIp] = best->id;
        best->remotes++;
        g_pins.add(best->path, 1);
    }
    return best;
}

static uint64_t addrKey(const struct sockaddr_storage& ss) {
    const uint8_t* p;
    size_t n;
    if (ss.ss_family == AF_INET) {
        p = (const uint8_t*)&((const struct sockaddr_in*)&ss)->sin_addr;
        n = 4;
    } else {
        p = (const uint8_t*)&((const struct sockaddr_in6*)&ss)->sin6_addr;
        n = 16;
    }
    uint64_t h = 0xcbf29ce484222325ull ^ (uint64_t)ss.ss_family;
    for (size_t i = 0; i < n; ++i) h = (h ^ p[i]) * 0x100000001b3ull;
    return h;
}

Backend* Balancer::pickBackendFast(const struct sockaddr_storage& ss) {
    /* bound the affinity tables: a flood of spoofed source addresses
     * must not grow memory forever. Wiping re-pins live remotes on
     * their next packet (brief affinity reset, logged). */
    if (remotesFast_.size() > kMaxRemotes) {
        log_.warn({{"remotes", Json((int64_t)remotesFast_.size())}},
                  "remote table overflow; resetting affinity pins");
        remotesFast_.clear();
        remotes_.clear();
        for (auto& [id, be] : backendsById_) {
            g_pins.add(be->path, -(int)be->remotes);
            be->remotes = 0;
        }
    }
    uint64_t key = addrKey(ss);
    auto it = remotesFast_.find(key);
    if (it != remotesFast_.end()) {
        auto bit = backendsById_.find(it->second);
        if (bit != backendsById_.end() && bit->second->ok)
            return bit->second.get();
        remotesFast_.erase(it);
    }
    /* cold path: render the IP once for the stats map */
    char ip[48] = "";
    if (ss.ss_family == AF_INET)
        inet_ntop(AF_INET,
                  &((const struct sockaddr_in*)&ss)->sin_addr, ip,
                  sizeof(ip));
    else
        inet_ntop(AF_INET6,
                  &((const struct sockaddr_in6*)&ss)->sin6_addr, ip,
                  sizeof(ip));
    Backend* be = pickBackend(ip);
    if (be != nullptr) remotesFast_[key] = be->id;
    return be;
}

static void ipOf(const struct sockaddr_storage& ss, char* out, size_t n,
                 uint16_t* port, uint8_t* family, uint8_t addr16[16]) {
    memset(addr16, 0, 16);
    if (ss.ss_family == AF_INET) {
        const auto* sa = (const struct sockaddr_in*)&ss;
        inet_ntop(AF_INET, &sa->sin_addr, out, n);
        *port = ntohs(sa->sin_port);
        *family = 4;
        memcpy(addr16, &sa->sin_addr, 4);
    } else {
        const auto* sa = (const struct sockaddr_in6*)&ss;
        inet_ntop(AF_INET6, &sa->sin6_addr, out, n);
        *port = ntohs(sa->sin6_port);
        *family = 6;
        memcpy(addr16, &sa->sin6_addr, 16);
    }
}

void Balancer::onUdpReadable() {
    struct mmsghdr* hdrs = rxHdrs_;
    struct iovec* iovs = rxIovs_;
    struct sockaddr_storage* addrs = rxAddrs_;

    while (true) {
        for (int i = 0; i < kBatch; ++i) {
            iovs[i] = {rxArena_.data() + (size_t)i * kRxBufSz,
                       kRxBufSz};
            memset(&hdrs[i], 0, sizeof(hdrs[i]));
            hdrs[i].msg_hdr.msg_iov = &iovs[i];
            hdrs[i].msg_hdr.msg_iovlen = 1;
            hdrs[i].msg_hdr.msg_name = &addrs[i];
            hdrs[i].msg_hdr.msg_namelen = sizeof(addrs[i]);
            hdrs[i].msg_hdr.msg_control = rxCtrl_[i];
            hdrs[i].msg_hdr.msg_controllen = sizeof(rxCtrl_[i]);
        }
        int n = recvmmsg(udpFd_, hdrs, kBatch, 0, nullptr);
        if (n <= 0) return;
        int64_t expiry = monotonicMillis() + kReplyTtlMs;
        std::set<Backend*> touched;
        for (int i = 0; i < n; ++i) {
            /* GRO: one message may be several equal-size datagrams;
             * the segment size arrives as a UDP_GRO cmsg */
            size_t seg = hdrs[i].msg_len;
            for (struct cmsghdr* cm = CMSG_FIRSTHDR(&hdrs[i].msg_hdr);
                 cm != nullptr;
                 cm = CMSG_NXTHDR(&hdrs[i].msg_hdr, cm)) {
                if (cm->cmsg_level == SOL_UDP &&
                    cm->cmsg_type == UDP_GRO) {
                    uint16_t g;
                    memcpy(&g, CMSG_DATA(cm), sizeof(g));
                    if (g > 0) seg = g;
                }
            }
            const uint8_t* base = rxArena_.data() + (size_t)i * kRxBufSz;
            size_t total = hdrs[i].msg_len;
            for (size_t off = 0; off < total; off += seg) {
                size_t dnsLen = std::min(seg, total - off);
                handleUdpQuery(base + off, dnsLen, addrs[i], expiry,
                               touched);
            }
        }
        for (Backend* be : touched) backendFlush(be);
        if (n < kBatch) return;
    }
}

void Balancer::handleUdpQuery(const uint8_t* dns, size_t dnsLen,
                              const struct sockaddr_storage& src,
                              int64_t expiry,
                              std::set<Backend*>& touched) {
    Backend* be = pickBackendFast(src);
    udpQueries_++;
    if (be == nullptr) {
        drops_++;
        return;
    }
    /* load shedding: a backend this far behind will answer past any
     * client deadline — drop now (clients retry) rather than queue
     * into a multi-ms tail */
    if (be->pendingCount > kMaxPending ||
        be->out.size() - be->outOff > (4u << 20)) {
        drops_++;
        return;
    }
    uint32_t reqId = be->nextReq++;
    PendingSlot* pr = be->slotFor(reqId);
    if (pr->reqId == 0) be->pendingCount++;
    else be->overwrites++;
    pr->reqId = reqId;
    pr->expiresAtMs = (uint32_t)expiry;
    pr->tcp = false;
    pr->tcpFd = -1;
    /* frame header assembled on the stack: one append for the 30-byte
     * head + one for the DNS payload (was 7 string appends per query
     * — 13% of user CPU in _M_append) */
    uint32_t plen = (uint32_t)(bsock::kQueryHeadLen + dnsLen);
    uint8_t head[bsock::kHeaderLen + bsock::kQueryHeadLen];
    head[0] = bsock::kMagic;
    head[1] = bsock::FRAME_QUERY;
    /* bsock1 integers are little-endian (protocol.hpp) */
    head[2] = (uint8_t)plen;
    head[3] = (uint8_t)(plen >> 8);
    head[4] = (uint8_t)(plen >> 16);
    head[5] = (uint8_t)(plen >> 24);
    head[6] = (uint8_t)reqId;
    head[7] = (uint8_t)(reqId >> 8);
    head[8] = (uint8_t)(reqId >> 16);
    head[9] = (uint8_t)(reqId >> 24);
    head[11] = 0;  // udp
    if (src.ss_family == AF_INET) {
        const auto* sa = (const struct sockaddr_in*)&src;
        uint16_t p = ntohs(sa->sin_port);
        head[10] = 4;
        head[12] = (uint8_t)p;
        head[13] = (uint8_t)(p >> 8);
        memcpy(head + 14, &sa->sin_addr, 4);
        memset(head + 18, 0, 12);
        pr->family = 4;
        pr->srcPort = p;
        memcpy(pr->addr, &sa->sin_addr, 4);
    } else {
        const auto* sa = (const struct sockaddr_in6*)&src;
        uint16_t p = ntohs(sa->sin6_port);
        head[10] = 6;
        head[12] = (uint8_t)p;
        head[13] = (uint8_t)(p >> 8);
        memcpy(head + 14, &sa->sin6_addr, 16);
        pr->family = 6;
        pr->srcPort = p;
        memcpy(pr->addr, &sa->sin6_addr, 16);
    }
    std::string& o = be->out;
    o.append((const char*)head, sizeof(head));
    o.append((const char*)dns, dnsLen);
    be->queries++;
    touched.insert(be);
}

void Balancer::onTcpAccept() {
    while (true) {
        struct sockaddr_storage ss;
        socklen_t sl = sizeof(ss);
        int fd = accept4(tcpFd_, (struct sockaddr*)&ss, &sl,
                         SOCK_NONBLOCK | SOCK_CLOEXEC);
        if (fd < 0) return;
        if (tcpClients_.size() >= 4096) {  /* fd-exhaustion guard */
            close(fd);
            continue;
        }
        auto c = std::make_shared<TcpClient>();
        c->fd = fd;
        c->src = ss;
        c->srcLen = sl;
        c->lastActivityMs = monotonicMillis();
        tcpClients_[fd] = c;
        loop_->addFd(fd, EPOLLIN, [this, c](uint32_t ev) {
            onTcpClient(c, ev);
        });
    }
}

void Balancer::tcpClientFlush(TcpClient* c) {
    while (!c->out.empty() && !c->closed) {
        ssize_t nw = write(c->fd, c->out.data(), c->out.size());
        if (nw > 0) {
            c->out.erase(0, (size_t)nw);
            continue;
        }
        if (nw < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
            if (!c->writeBlocked) {
                c->writeBlocked = true;
                loop_->modFd(c->fd, EPOLLIN | EPOLLOUT);
            }
            return;
        }
        c->closed = true;
        loop_->delFd(c->fd);
        close(c->fd);
        tcpClients_.erase(c->fd);
        return;
    }
    if (c->writeBlocked && !c->closed) {
        c->writeBlocked = false;
        loop_->modFd(c->fd, EPOLLIN);
    }
}

void Balancer::onTcpClient(std::shared_ptr<TcpClient> c, uint32_t ev) {
    if (c->closed) return;
    if (ev & (EPOLLHUP | EPOLLERR)) {
        c->closed = true;
        loop_->delFd(c->fd);
        close(c->fd);
        tcpClients_.erase(c->fd);
        return;
    }
    if (ev & EPOLLOUT) tcpClientFlush(c.get());
    if (c->closed || !(ev & EPOLLIN)) return;
    c->lastActivityMs = monotonicMillis();

    char buf[8192];
    while (true) {
        ssize_t nr = read(c->fd, buf, sizeof(buf));
        if (nr > 0) {
            c->in.append(buf, (size_t)nr);
            if (c->in.size() > (1 << 20)) nr = 0;  // runaway
            else continue;
        }
        if (nr < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) break;
        if (nr <= 0) {
            c->closed = true;
            loop_->delFd(c->fd);
            close(c->fd);
            tcpClients_.erase(c->fd);
            return;
        }
    }

    while (c->in.size() >= 2) {
        size_t mlen = ((size_t)(uint8_t)c->in[0] << 8) | (uint8_t)c->in[1];
        if (c->in.size() < 2 + mlen) break;
        char ip[48];
        uint16_t srcPort;
        uint8_t family;
        uint8_t addr16[16];
        ipOf(c->src, ip, sizeof(ip), &srcPort, &family, addr16);
        Backend* be = pickBackend(ip);
        if (be != nullptr) {
            uint32_t reqId = be->nextReq++;
            PendingSlot* pr = be->slotFor(reqId);
            if (pr->reqId == 0) be->pendingCount++;
            else be->overwrites++;
            pr->reqId = reqId;
            pr->tcp = true;
            pr->tcpFd = c->fd;
            pr->expiresAtMs =
                (uint32_t)(monotonicMillis() + kReplyTtlMs);
            std::string payload;
            bsock::putU32(payload, reqId);
            payload.push_back((char)family);
            payload.push_back((char)1);  // tcp
            bsock::putU16(payload, srcPort);
            payload.append((const char*)addr16, 16);
            payload.append(c->in.data() + 2, mlen);
            bsock::appendFrame(be->out, bsock::FRAME_QUERY, payload);
            be->queries++;
            backendFlush(be);
        }
        c->in.erase(0, 2 + mlen);
    }
}

Json Balancer::snapshot() const {
    Json out = Json::object();
    JsonArray bes;
    for (auto& [path, be] : backends_) {
        Json b = Json::object();
        b.set("id", Json((int64_t)be->id));
        b.set("path", Json(be->path));
        b.set("ok", Json(be->ok));
        b.set("remotes", Json((int64_t)be->remotes));
        b.set("queries", Json((int64_t)be->queries));
        b.set("replies", Json((int64_t)be->replies));
        b.set("pending", Json((int64_t)be->pendingCount));
        b.set("overwrites", Json((int64_t)be->overwrites));
        bes.push_back(std::move(b));
    }
    out.set("backends", Json(std::move(bes)));
    JsonArray rms;
    for (auto& [ip, id] : remotes_) {
        Json r = Json::object();
        r.set("addr", Json(ip));
        r.set("backend", Json((int64_t)id));
        rms.push_back(std::move(r));
    }
    out.set("remotes", Json(std::move(rms)));
    out.set("udp_queries", Json((int64_t)udpQueries_));
    out.set("udp_replies", Json((int64_t)udpReplies_));
    out.set("drops", Json((int64_t)drops_));
    return out;
}

void Balancer::onStatsAccept() {
    while (true) {
        /* non-blocking best-effort: a stalled stats reader must not
         * stall the serving loop */
        int fd = accept4(statsFd_, nullptr, nullptr,
                         SOCK_CLOEXEC | SOCK_NONBLOCK);
        if (fd < 0) return;
        /* merge the latest snapshot from every worker (our own fresh;
         * others as of their last sweep tick) */
        g_stats.publish(workerId_, snapshot().dump());
        std::map<std::string, Json> byPath;      // merged backends
        JsonArray remotes;
        int64_t udpQ = 0, udpR = 0, drops = 0;
        int workers = 0;
        for (const std::string& frag : g_stats.all()) {
            auto parsed = Json::parse(frag);
            if (!parsed) continue;
            workers++;
            udpQ += parsed->get("udp_queries").asInt();
            udpR += parsed->get("udp_replies").asInt();
            drops += parsed->get("drops").asInt();
            for (const auto& r : parsed->get("remotes").items())
                remotes.push_back(r);
            for (const auto& b : parsed->get("backends").items()) {
                const std::string& path = b.get("path").asString();
                auto it = byPath.find(path);
                if (it == byPath.end()) {
                    byPath[path] = b;
                } else {
                    Json& m = it->second;
                    m.set("remotes", Json(m.get("remotes").asInt() +
                                          b.get("remotes").asInt()));
                    m.set("queries", Json(m.get("queries").asInt() +
                                          b.get("queries").asInt()));
                    m.set("replies", Json(m.get("replies").asInt() +
                                          b.get("replies").asInt()));
                    m.set("pending", Json(m.get("pending").asInt() +
                                          b.get("pending").asInt()));
                    m.set("overwrites",
                          Json(m.get("overwrites").asInt() +
                               b.get("overwrites").asInt()));
                    m.set("ok", Json(m.get("ok").asBool() ||
                                     b.get("ok").asBool()));
                }
            }
        }
        Json out = Json::object();
        JsonArray bes;
        for (auto& [path, b] : byPath) bes.push_back(std::move(b));
        out.set("backends", Json(std::move(bes)));
        out.set("remotes", Json(std::move(remotes)));
        out.set("udp_queries", Json(udpQ));
        out.set("udp_replies", Json(udpR));
        out.set("drops", Json(drops));
        out.set("workers", Json((int64_t)workers));
        std::string s = out.dump();
        s.push_back('\n');
        ssize_t rv = write(fd, s.data(), s.size());
        (void)rv;
        close(fd);
    }
}

void Balancer::sweep() {
    int64_t now = monotonicMillis();
    /* reap idle TCP clients (60s) */
    std::vector<std::shared_ptr<TcpClient>> idle;
    for (auto& [fd, c] : tcpClients_)
        if (now - c->lastActivityMs > 60000) idle.push_back(c);
    for (auto& c : idle) {
        if (c->closed) continue;
        c->closed = true;
        loop_->delFd(c->fd);
        close(c->fd);
        tcpClients_.erase(c->fd);
    }
    for (auto& [path, be] : backends_) {
        if (be->fd < 0) continue;
        /* Expire stale pendings (uint32 wrap-safe comparison) in a
         * bounded chunk per tick: a full walk of the ring is ~8 MB of
         * cold cache per backend and was measured as a multi-ms stall
         * that put the sweep duration straight into client p99
         * (profiles/ probe data). Correctness doesn't need the full
         * walk — an un-reaped slot is reclaimed on its next ring
         * collision (counted as overwrite) — the chunked scan only
         * keeps pendingCount from drifting for load-shedding. */
        if (be->pending && be->pendingCount > 0) {
            uint32_t now32 = (uint32_t)now;
            size_t chunk = g_pendingSlots / 16 + 1;
            if (chunk > 4096) chunk = 4096;
            for (size_t k = 0; k < chunk; ++k) {
                PendingSlot& s =
                    be->pending[(be->sweepCursor + k) &
                                (g_pendingSlots - 1)];
                if (s.reqId != 0 &&
                    (int32_t)(now32 - s.expiresAtMs) > 0) {
                    s.reqId = 0;
                    be->pendingCount--;
                }
            }
            be->sweepCursor =
                (be->sweepCursor + chunk) & (g_pendingSlots - 1);
        }
        /* health probe */
        if (now - be->pingSentAt >= kPingIntervalMs) {
            bsock::appendFrame(be->out, bsock::FRAME_PING, "");
            be->pingSentAt = now;
            backendFlush(be.get());
        }
        if (now - be->lastPongAt > kPingTimeoutMs) {
            log_.warn({{"path", Json(be->path)}},
                      "backend unresponsive; marking down");
            backendDown(be.get());
        }
    }
    g_stats.publish(workerId_, snapshot().dump());
}

}  // namespace

int main(int argc, char** argv) {
    /* a peer closing mid-write must be an EPIPE errno, not process
     * death */
    signal(SIGPIPE, SIG_IGN);
    const char* lvl = getenv("LOG_LEVEL");
    Logger log("binder-balancer",
               logLevelFromName(lvl ? lvl : "info", LogLevel::Info));
    std::string host;
    uint16_t port = 53;
    std::string dir = "/var/run/binder/sockets";
    std::string stats;
    int rescanMs = 1000;
    int workers = 1;
    int c;
    while ((c = getopt(argc, argv, "hp:H:s:S:r:w:q:")) != -1) {
        switch (c) {
        case 'q': {
            size_t v = (size_t)strtoull(optarg, nullptr, 10);
            size_t p2 = 1;
            while (p2 < v) p2 <<= 1;
            g_pendingSlots = p2 < 8 ? 8 : p2;
            break;
        }
        case 'p': port = (uint16_t)atoi(optarg); break;
        case 'H': host = optarg; break;
        case 's': dir = optarg; break;
        case 'S': stats = optarg; break;
        case 'r': rescanMs = atoi(optarg); break;
        case 'w': workers = atoi(optarg); break;
        case 'h':
        default:
            fprintf(stderr,
                    "usage: binder-balancer [-p port] [-H host] "
                    "[-s socket-dir] [-S stats-socket] [-r rescan-ms] "
                    "[-w workers] [-q ring-slots]\n");
            return c == 'h' ? 0 : 1;
        }
    }
    if (workers < 1) workers = 1;

    sigset_t mask;
    sigemptyset(&mask);
    sigaddset(&mask, SIGTERM);
    sigaddset(&mask, SIGINT);
    sigprocmask(SIG_BLOCK, &mask, nullptr);

    /*
     * -w > 1: SO_REUSEPORT worker shards, each a full independent
     * balancer (own epoll loop, backend connections, affinity map).
     * The kernel hashes client (ip,port) across workers, so per-IP
     * affinity holds within a worker; -w 1 (default) preserves the
     * reference's exact single-process per-IP affinity semantics.
     */
    g_scanner.start(dir, rescanMs < 500 ? rescanMs : 500);
    /* give the first scan a moment so startup discovery is immediate */
    while (g_scanner.gen.load() == 0)
        std::this_thread::sleep_for(std::chrono::milliseconds(2));

    std::vector<std::unique_ptr<EventLoop>> loops;
    std::vector<std::unique_ptr<Balancer>> bals;
    std::vector<std::thread> threads;
    for (int w = 0; w < workers; ++w) {
        loops.emplace_back(std::make_unique<EventLoop>());
        bals.emplace_back(std::make_unique<Balancer>(
            loops[w].get(),
            log.child({{"worker", Json((int64_t)w)}}), host, port, dir,
            w == 0 ? stats : std::string(), rescanMs, workers > 1, w,
            workers));
        if (!bals[w]->start()) return 1;
    }
    for (int w = 0; w < workers; ++w)
        threads.emplace_back([&, w]() { loops[w]->run(); });

    int sfd = signalfd(-1, &mask, SFD_CLOEXEC);
    struct signalfd_siginfo si;
    ssize_t rv = read(sfd, &si, sizeof(si));
    (void)rv;
    for (auto& l : loops) l->stop();
    for (auto& t : threads) t.join();
    for (auto& b : bals) b->stop();
    g_scanner.shutdown();
    return 0;
}
