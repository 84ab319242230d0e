/*
 * binder-amd: the DNS serving core (lib/server.js equivalent) — UDP, TCP
 * and balancer-socket listeners feeding the resolution engine, with the
 * reference's after-hook metrics and per-query log lines
 * (/root/reference/lib/server.js:435-660).
 *
 * Native design notes (not a translation):
 *  - UDP RX/TX uses recvmmsg/sendmmsg batches (up to 64 datagrams per
 *    syscall) — the dominant cost at high QPS is syscalls, not lookups;
 *  - responses are encoded straight into a flat TX arena;
 *  - the balancer hop is a framed protocol over a UNIX stream socket
 *    ("bsock1", see balancer/protocol.hpp) carrying the original client
 *    address, like mname's listenBalancer (server.js:621-631).
 */
#pragma once

#include <netinet/in.h>
#include <sys/socket.h>
#include <sys/un.h>

#include <functional>
#include <random>
#include <map>
#include <memory>
#include <string>
#include <vector>

#include "../common/log.hpp"
#include "../common/loop.hpp"
#include "../engine/engine.hpp"
#include "metrics.hpp"

namespace bamd {

struct ServerOptions {
    std::string name = "binder";
    std::string host;  // bind address; empty = all interfaces
    uint16_t port = 53;
    std::string balancerSocket;  // unix socket path; empty = none
};

struct QueryTimers {
    int64_t parseUs = 0;
    int64_t resolveUs = 0;
    int64_t encodeUs = 0;
};

struct ClientInfo {
    char address[48] = "";  // presentation form
    uint16_t port = 0;
    const char* family = "udp";  // matches query.src.family usage in logs
};

class RecursionIface {
  public:
    virtual ~RecursionIface() = default;
    /* Fill resp (REFUSED if nothing found) and invoke done. May complete
     * synchronously or later from the event loop. */
    virtual void resolve(const dns::Message& query, dns::Message& resp,
                         std::function<void()> done) = 0;
};

class DnsServer {
  public:
    DnsServer(EventLoop* loop, Logger log, ServerOptions opts,
              Engine* engine, Collector* collector);
    ~DnsServer();

    void setRecursion(RecursionIface* r) { recursion_ = r; }

    bool start();  // listen UDP + TCP (+ balancer socket if configured)
    void stop();

    uint16_t boundPort() const { return boundPort_; }

    uint64_t queriesServed() const { return served_; }

  private:
    struct TcpConn {
        int fd;
        std::string in;
        std::string out;
        bool writeBlocked = false;
        ClientInfo ci;
        uint32_t pendingAsync = 0;
        bool closed = false;
        int64_t lastActivityMs = 0;
    };
    struct BalConn {
        int fd;
        std::string in;
        std::string out;
        bool writeBlocked = false;
        bool closed = false;
    };

    bool openUdp();
    bool openTcp();
    bool openBalancer();

    void onUdpReadable();
    void onTcpAccept();
    void onTcpConn(TcpConn* c, uint32_t events);
    void onBalAccept();
    void onBalConn(BalConn* c, uint32_t events);

    /*
     * Decode + resolve + encode. Returns true and fills `out` when the
     * answer is synchronous; returns false when handed to recursion (the
     * asyncReply callback owns delivery) — or when the packet is dropped
     * (out left empty).
     */
    bool process(const uint8_t* data, size_t len, bool udp,
                 const ClientInfo& ci, std::vector<uint8_t>& out,
                 std::function<void(std::vector<uint8_t>)> asyncReply);
    bool fastPath(const uint8_t* data, size_t len, bool udp,
                  const ClientInfo& ci, std::vector<uint8_t>& out);
    std::string_view srvSvc_, srvProto_;   /* scratch, loop thread only */
    std::vector<uint32_t> shuffleIdx_;
    std::mt19937 rng_{std::random_device{}()};
    void initFastMetricSlots();
    /* cached hot-path metric slots (stable map nodes) */
    uint64_t* fpCntA_ = nullptr;
    uint64_t* fpCntSrv_ = nullptr;
    Histogram::Series* fpLatA_ = nullptr;
    Histogram::Series* fpLatSrv_ = nullptr;
    Histogram::Series* fpSizeA_ = nullptr;
    Histogram::Series* fpSizeSrv_ = nullptr;
    std::string logFields_, logScratch_;   /* afterQuery log buffers */

  public:
    /* the store behind the engine (fast-path lookups); set by main */
    void setStore(const Store* s) { engineStore_ = s; }

  private:
    const Store* engineStore_ = nullptr;

    void afterQuery(const dns::Message& query, const dns::Message& resp,
                    const QueryResult& qr, const ClientInfo& ci,
                    size_t bytesSent, int64_t startUs,
                    const QueryTimers& tm);

    void tcpFlush(TcpConn* c);
    void sweepIdleTcp();
    void balFlush(BalConn* c);
    void closeTcp(TcpConn* c);
    void closeBal(BalConn* c);

    EventLoop* loop_;
    Logger log_;
    ServerOptions opts_;
    Engine* engine_;
    RecursionIface* recursion_ = nullptr;

    Counter* reqCounter_ = nullptr;
    Histogram* latHist_ = nullptr;
    Histogram* sizeHist_ = nullptr;

    int udpFd_ = -1;
    int tcpFd_ = -1;
    int balFd_ = -1;
    uint16_t boundPort_ = 0;
    uint64_t served_ = 0;

    /* shared_ptr: async recursion replies may outlive the connection
     * (fds get reused; a closed conn object must stay valid until the
     * last in-flight reply fires and observes `closed`). */
    std::map<int, std::shared_ptr<TcpConn>> tcpConns_;
    std::map<int, std::shared_ptr<BalConn>> balConns_;

    /* UDP batched I/O arenas. */
    static constexpr int kBatch = 64;
    static constexpr size_t kInBuf = 4096;
    std::vector<uint8_t> rxArena_;
    std::vector<struct mmsghdr> rxHdrs_;
    std::vector<struct iovec> rxIovs_;
    std::vector<struct sockaddr_storage> rxAddrs_;
    std::vector<std::vector<uint8_t>> txBufs_;
    std::vector<struct mmsghdr> txHdrs_;
    std::vector<struct iovec> txIovs_;
    std::vector<struct sockaddr_storage> txAddrs_;
};

void fillClientInfo(ClientInfo& ci, const struct sockaddr_storage& ss,
                    const char* family);

}  // namespace bamd
