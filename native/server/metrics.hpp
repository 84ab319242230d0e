/*
 * binder-amd: Prometheus metrics (artedi/triton-metrics equivalent).
 *
 * The reference exposes three binder metrics labeled by query type plus
 * static labels, over a restify HTTP server on port+1000
 * (/root/reference/main.js:134-152, lib/server.js:31-34, 456-469,
 * 520-535). Here: a lock-free single-threaded collector plus a minimal
 * HTTP/1.1 exposition endpoint on the same event loop.
 */
#pragma once

#include <cstdint>
#include <map>
#include <memory>
#include <string>
#include <vector>

#include "../common/loop.hpp"

namespace bamd {

class Counter {
  public:
    void increment(const std::string& labels, uint64_t by = 1) {
        vals_[labels] += by;
    }
    /* hot-path form: the returned slot is stable (map node), so
     * callers cache it once and bump it without a string lookup */
    uint64_t& slot(const std::string& labels) { return vals_[labels]; }
    void set(const std::string& labels, uint64_t v) {  // gauges
        vals_[labels] = v;
    }
    const std::map<std::string, uint64_t>& values() const { return vals_; }

  private:
    std::map<std::string, uint64_t> vals_;
};

class Histogram {
  public:
    Histogram();
    void observe(const std::string& labels, double v);

    struct Series;
    /* hot-path form: resolve the labelled series once (stable map
     * node), then observeFast per event without a string lookup */
    Series& seriesRef(const std::string& labels);
    void observeFast(Series& s, double v) {
        size_t i = 0;
        while (i < bounds_.size() && v > bounds_[i]) ++i;
        s.bucketCounts[i]++;
        s.sum += v;
        s.count++;
    }

    struct Series {
        std::vector<uint64_t> bucketCounts;
        double sum = 0;
        uint64_t count = 0;
    };
    const std::vector<double>& bounds() const { return bounds_; }
    const std::map<std::string, Series>& series() const { return series_; }

  private:
    std::vector<double> bounds_;
    std::map<std::string, Series> series_;
};

class Collector {
  public:
    Counter* counter(const std::string& name, const std::string& help);
    Histogram* histogram(const std::string& name, const std::string& help);
    Counter* gauge(const std::string& name, const std::string& help);

    /* Prometheus text format. staticLabels pre-rendered as
     * 'k="v",k2="v2"' (may be empty). */
    std::string expose(const std::string& staticLabels) const;

  private:
    struct Metric {
        std::string name, help, kind;
        std::unique_ptr<Counter> counter;
        std::unique_ptr<Histogram> histogram;
    };
    std::vector<Metric> metrics_;
};

/*
 * Minimal HTTP server for GET /metrics (any path actually returns
 * metrics, like triton-metrics' behavior of serving the collector).
 */
class MetricsHttpServer {
  public:
    MetricsHttpServer(EventLoop* loop, Collector* collector,
                      std::string staticLabels);
    ~MetricsHttpServer();

    /* Returns false if bind failed. */
    bool listen(const std::string& address, uint16_t port);
    void close();
    uint16_t port() const { return port_; }

  private:
    void onAccept();
    void onConn(int fd, uint32_t events);

    EventLoop* loop_;
    Collector* collector_;
    std::string staticLabels_;
    int listenFd_ = -1;
    uint16_t port_ = 0;
    std::map<int, std::string> connBufs_;
};

/* Render 'k="v"' pairs, escaping label values. */
std::string renderLabels(
    const std::vector<std::pair<std::string, std::string>>& labels);

}  // namespace bamd
