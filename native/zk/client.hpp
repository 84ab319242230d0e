/*
 * binder-amd: ZooKeeper client (zkstream equivalent).
 *
 * Single-threaded, event-loop driven. Capabilities used by the reference
 * (lib/zk.js:33-47, test/helper.js:98-166): session establishment with
 * timeout, automatic reconnect, a 'session' event on each (re)established
 * session driving full watch re-registration, getData/getChildren with
 * watches, create/setData/delete for tooling and tests, pings to hold
 * the session, and watch-event delivery.
 *
 * Reconnect model: on connection loss we retry with the previous session
 * id; if the server expired us (or we never had one) a fresh session is
 * created. In BOTH cases the 'session' callback fires, and the mirror
 * above re-arms every watch with fresh reads — so missed events during
 * the outage are swallowed by the resync (same recovery shape as
 * lib/zk.js:45-47 + rebind, zk.js:209-223).
 */
#pragma once

#include <functional>
#include <map>
#include <memory>
#include <string>
#include <vector>

#include "../common/log.hpp"
#include "../common/loop.hpp"
#include "jute.hpp"

namespace bamd::zk {

struct ZkConfig {
    std::string host = "127.0.0.1";
    uint16_t port = 2181;
    int sessionTimeoutMs = 30000;  // lib/zk.js:37
    int reconnectDelayMs = 1000;
};

class ZkClient {
  public:
    using SessionCb = std::function<void()>;
    using WatchCb =
        std::function<void(int32_t type, const std::string& path)>;
    using DataCb = std::function<void(int32_t rc, const std::string& data,
                                      const Stat& stat)>;
    using ChildrenCb = std::function<void(
        int32_t rc, const std::vector<std::string>& children)>;
    using StatCb = std::function<void(int32_t rc, const Stat& stat)>;
    using StringCb =
        std::function<void(int32_t rc, const std::string& value)>;
    using VoidCb = std::function<void(int32_t rc)>;

    ZkClient(EventLoop* loop, Logger log, ZkConfig cfg);
    ~ZkClient();

    void start();
    void close();

    bool connected() const { return state_ == State::Connected; }
    int64_t sessionId() const { return sessionId_; }

    /* Fired on every newly usable session (incl. reconnects). */
    void onSession(SessionCb cb) { sessionCb_ = std::move(cb); }
    /* All watch events funnel here (mirror routes by path). */
    void onWatch(WatchCb cb) { watchCb_ = std::move(cb); }

    void getData(const std::string& path, bool watch, DataCb cb);
    void getChildren(const std::string& path, bool watch, ChildrenCb cb);
    void exists(const std::string& path, bool watch, StatCb cb);
    void create(const std::string& path, const std::string& data,
                int32_t flags, StringCb cb);
    void setData(const std::string& path, const std::string& data,
                 int32_t version, VoidCb cb);
    void del(const std::string& path, int32_t version, VoidCb cb);

    /* Stats for metrics/tests. */
    uint64_t reconnects() const { return reconnects_; }
    uint64_t sessionsEstablished() const { return sessions_; }

  private:
    enum class State { Closed, Connecting, Handshaking, Connected };

    struct Pending {
        int32_t xid;
        int32_t op;
        DataCb dataCb;
        ChildrenCb childrenCb;
        StatCb statCb;
        StringCb stringCb;
        VoidCb voidCb;
    };

    void connectStart();
    void onSockEvent(uint32_t events);
    void onReadable();
    void onPacket(const uint8_t* data, size_t len);
    void onConnectResponse(const uint8_t* data, size_t len);
    void sendHandshake();
    void sendPacket(const std::string& body);
    void flush();
    void failAllPending();
    void scheduleReconnect();
    void teardown();
    void armPingTimer();
    void sendPing();
    int32_t nextXid() { return xid_++; }

    EventLoop* loop_;
    Logger log_;
    ZkConfig cfg_;
    State state_ = State::Closed;
    bool closing_ = false;

    int fd_ = -1;
    std::string inBuf_;
    std::string outBuf_;
    bool writeBlocked_ = false;

    int64_t sessionId_ = 0;
    std::string passwd_;
    int negotiatedTimeout_ = 0;
    int64_t lastZxid_ = 0;
    int64_t lastPacketRecvMs_ = 0;

    int32_t xid_ = 1;
    std::map<int32_t, Pending> pending_;  // in-flight, FIFO by xid

    SessionCb sessionCb_;
    WatchCb watchCb_;

    uint64_t reconnectTimer_ = 0;
    uint64_t pingTimer_ = 0;
    uint64_t reconnects_ = 0;
    uint64_t sessions_ = 0;
};

}  // namespace bamd::zk
