#include "loop.hpp"

#include <fcntl.h>
#include <sys/epoll.h>
#include <sys/eventfd.h>
#include <unistd.h>

#include <cstdio>
#include <cstring>
#include <stdexcept>

#include "log.hpp"

namespace bamd {

bool setNonBlocking(int fd) {
    int flags = fcntl(fd, F_GETFL, 0);
    if (flags < 0) return false;
    return fcntl(fd, F_SETFL, flags | O_NONBLOCK) == 0;
}

EventLoop::EventLoop() {
    epfd_ = epoll_create1(EPOLL_CLOEXEC);
    if (epfd_ < 0)
        throw std::runtime_error(std::string("epoll_create1: ") +
                                 strerror(errno));
    wakeFd_ = eventfd(0, EFD_NONBLOCK | EFD_CLOEXEC);
    if (wakeFd_ >= 0) {
        addFd(wakeFd_, EPOLLIN, [this](uint32_t) {
            uint64_t v;
            while (read(wakeFd_, &v, sizeof(v)) == sizeof(v)) {
            }
            std::vector<TimerCallback> tasks;
            {
                std::lock_guard<std::mutex> g(postMutex_);
                tasks.swap(posted_);
            }
            for (auto& t : tasks) t();
        });
    }
}

EventLoop::~EventLoop() {
    if (wakeFd_ >= 0) close(wakeFd_);
    close(epfd_);
}

void EventLoop::postFromThread(TimerCallback cb) {
    {
        std::lock_guard<std::mutex> g(postMutex_);
        posted_.push_back(std::move(cb));
    }
    uint64_t one = 1;
    ssize_t rv = write(wakeFd_, &one, sizeof(one));
    (void)rv;
}

void EventLoop::addFd(int fd, uint32_t events, FdCallback cb) {
    uint32_t gen = nextFdGen_++;
    struct epoll_event ev {};
    ev.events = events;
    ev.data.u64 = ((uint64_t)gen << 32) | (uint32_t)fd;
    if (epoll_ctl(epfd_, EPOLL_CTL_ADD, fd, &ev) != 0)
        throw std::runtime_error(std::string("epoll_ctl add: ") +
                                 strerror(errno));
    fds_[fd] = FdReg{gen, std::move(cb)};
}

void EventLoop::modFd(int fd, uint32_t events) {
    auto it = fds_.find(fd);
    if (it == fds_.end()) return;
    struct epoll_event ev {};
    ev.events = events;
    ev.data.u64 = ((uint64_t)it->second.gen << 32) | (uint32_t)fd;
    if (epoll_ctl(epfd_, EPOLL_CTL_MOD, fd, &ev) != 0)
        throw std::runtime_error(std::string("epoll_ctl mod: ") +
                                 strerror(errno));
}

void EventLoop::delFd(int fd) {
    epoll_ctl(epfd_, EPOLL_CTL_DEL, fd, nullptr);
    fds_.erase(fd);
}

uint64_t EventLoop::addTimer(int64_t delayMs, TimerCallback cb) {
    uint64_t id = nextTimerId_++;
    timers_[id] = std::move(cb);
    heap_.push(Timer{monotonicMillis() + delayMs, id});
    return id;
}

void EventLoop::cancelTimer(uint64_t id) { timers_.erase(id); }

void EventLoop::defer(TimerCallback cb) { deferred_.push_back(std::move(cb)); }

int64_t EventLoop::nextTimerDelay() const {
    if (!deferred_.empty()) return 0;
    if (heap_.empty()) return -1;
    int64_t d = heap_.top().deadline - monotonicMillis();
    return d < 0 ? 0 : d;
}

void EventLoop::fireTimers() {
    int64_t now = monotonicMillis();
    while (!heap_.empty() && heap_.top().deadline <= now) {
        uint64_t id = heap_.top().id;
        heap_.pop();
        auto it = timers_.find(id);
        if (it == timers_.end()) continue;  // cancelled
        TimerCallback cb = std::move(it->second);
        timers_.erase(it);
        cb();
    }
}

void EventLoop::runOnce(int64_t maxWaitMs) {
    // Run deferred tasks queued before this iteration.
    if (!deferred_.empty()) {
        std::vector<TimerCallback> tasks;
        tasks.swap(deferred_);
        for (auto& t : tasks) t();
    }

    int64_t wait = nextTimerDelay();
    if (wait < 0 || wait > maxWaitMs) wait = maxWaitMs;

    struct epoll_event evs[64];
    int n = epoll_wait(epfd_, evs, 64, (int)wait);
    if (n < 0) {
        if (errno == EINTR) return;
        throw std::runtime_error(std::string("epoll_wait: ") +
                                 strerror(errno));
    }
    for (int i = 0; i < n; ++i) {
        int fd = (int)(uint32_t)evs[i].data.u64;
        uint32_t gen = (uint32_t)(evs[i].data.u64 >> 32);
        auto it = fds_.find(fd);
        if (it == fds_.end()) continue;  // removed by earlier callback
        if (it->second.gen != gen) continue;  // fd reused in this batch
        // Copy: callback may delFd itself.
        FdCallback cb = it->second.cb;
        cb(evs[i].events);
    }
    fireTimers();
}

void EventLoop::run() {
    running_ = true;
    while (running_) runOnce(1000);
}

bool EventLoop::runUntil(const std::function<bool()>& pred,
                         int64_t timeoutMs) {
    int64_t deadline = monotonicMillis() + timeoutMs;
    running_ = true;
    while (running_) {
        if (pred()) return true;
        int64_t left = deadline - monotonicMillis();
        if (left <= 0) break;
        runOnce(left > 50 ? 50 : left);
    }
    running_ = false;
    return pred();
}

}  // namespace bamd
