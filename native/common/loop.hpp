/*
 * binder-amd: single-threaded epoll event loop.
 *
 * The reference is a single-threaded event-driven server (Node's event loop;
 * SURVEY.md §3.1 "no worker threads"). We keep the same concurrency model —
 * one loop per process, N processes behind the balancer for scale-out — but
 * on raw epoll with edge-level callbacks and a timer heap.
 */
#pragma once

#include <atomic>
#include <cstdint>
#include <functional>
#include <map>
#include <mutex>
#include <queue>
#include <vector>

namespace bamd {

class EventLoop {
  public:
    using FdCallback = std::function<void(uint32_t events)>;
    using TimerCallback = std::function<void()>;

    EventLoop();
    ~EventLoop();
    EventLoop(const EventLoop&) = delete;
    EventLoop& operator=(const EventLoop&) = delete;

    /* events: EPOLLIN / EPOLLOUT bitmask. Callback owns the fd lifecycle. */
    void addFd(int fd, uint32_t events, FdCallback cb);
    void modFd(int fd, uint32_t events);
    void delFd(int fd);

    /* One-shot timer; returns an id usable with cancelTimer. */
    uint64_t addTimer(int64_t delayMs, TimerCallback cb);
    void cancelTimer(uint64_t id);

    /* Run cb on the next loop iteration (loop thread only). */
    void defer(TimerCallback cb);

    /* Thread-safe: queue cb for execution on the loop thread and wake
     * it. Used by helper threads (e.g. the blocking LDAP refresh). */
    void postFromThread(TimerCallback cb);

    void run();
    void stop() { running_ = false; }
    bool running() const { return running_; }

    /* Run until pred() is true or timeout; for tests/clients. */
    bool runUntil(const std::function<bool()>& pred, int64_t timeoutMs);

  private:
    void runOnce(int64_t maxWaitMs);
    int64_t nextTimerDelay() const;
    void fireTimers();

    struct Timer {
        int64_t deadline;
        uint64_t id;
        bool operator>(const Timer& o) const {
            return deadline > o.deadline ||
                   (deadline == o.deadline && id > o.id);
        }
    };

    /* Each registration carries a generation; epoll events are tagged
     * with it (data.u64 = gen<<32 | fd) so a stale queued event for a
     * closed-and-reused fd within one epoll_wait batch is dropped
     * instead of being delivered to the new registration (fd-reuse ABA). */
    struct FdReg {
        uint32_t gen;
        FdCallback cb;
    };

    int epfd_;
    std::atomic<bool> running_{false};
    std::map<int, FdReg> fds_;
    uint32_t nextFdGen_ = 1;
    std::priority_queue<Timer, std::vector<Timer>, std::greater<Timer>> heap_;
    std::map<uint64_t, TimerCallback> timers_;  // id -> cb (absent=cancelled)
    uint64_t nextTimerId_ = 1;
    std::vector<TimerCallback> deferred_;

    int wakeFd_ = -1;
    std::mutex postMutex_;
    std::vector<TimerCallback> posted_;
};

/* fcntl O_NONBLOCK helper; returns false on error. */
bool setNonBlocking(int fd);

}  // namespace bamd
