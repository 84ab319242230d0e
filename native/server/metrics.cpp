#include "metrics.hpp"

#include <arpa/inet.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/epoll.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cstdio>
#include <cstring>

namespace bamd {

Histogram::Histogram()
    : bounds_({0.0001, 0.00025, 0.0005, 0.001, 0.0025, 0.005, 0.01, 0.025,
               0.05, 0.1, 0.25, 0.5, 1, 2.5, 5, 10}) {}

Histogram::Series& Histogram::seriesRef(const std::string& labels) {
    Series& s = series_[labels];
    if (s.bucketCounts.empty())
        s.bucketCounts.resize(bounds_.size() + 1, 0);
    return s;
}

void Histogram::observe(const std::string& labels, double v) {
    Series& s = series_[labels];
    if (s.bucketCounts.empty()) s.bucketCounts.resize(bounds_.size() + 1, 0);
    size_t i = 0;
    while (i < bounds_.size() && v > bounds_[i]) ++i;
    s.bucketCounts[i]++;
    s.sum += v;
    s.count++;
}

Counter* Collector::counter(const std::string& name,
                            const std::string& help) {
    for (auto& m : metrics_)
        if (m.name == name) return m.counter.get();
    metrics_.push_back(
        {name, help, "counter", std::make_unique<Counter>(), nullptr});
    return metrics_.back().counter.get();
}

Counter* Collector::gauge(const std::string& name, const std::string& help) {
    for (auto& m : metrics_)
        if (m.name == name) return m.counter.get();
    metrics_.push_back(
        {name, help, "gauge", std::make_unique<Counter>(), nullptr});
    return metrics_.back().counter.get();
}

Histogram* Collector::histogram(const std::string& name,
                                const std::string& help) {
    for (auto& m : metrics_)
        if (m.name == name) return m.histogram.get();
    metrics_.push_back(
        {name, help, "histogram", nullptr, std::make_unique<Histogram>()});
    return metrics_.back().histogram.get();
}

static void appendLabelSet(std::string& out, const std::string& a,
                           const std::string& b,
                           const std::string& extra = "") {
    bool any = !a.empty() || !b.empty() || !extra.empty();
    if (!any) return;
    out.push_back('{');
    bool first = true;
    for (const std::string* s : {&a, &b, &extra}) {
        if (s->empty()) continue;
        if (!first) out.push_back(',');
        first = false;
        out += *s;
    }
    out.push_back('}');
}

std::string Collector::expose(const std::string& staticLabels) const {
    std::string out;
    out.reserve(4096);
    char buf[64];
    for (const auto& m : metrics_) {
        out += "# HELP " + m.name + " " + m.help + "\n";
        out += "# TYPE " + m.name + " " + m.kind + "\n";
        if (m.counter) {
            if (m.counter->values().empty()) {
                out += m.name;
                appendLabelSet(out, staticLabels, "");
                out += " 0\n";
            }
            for (const auto& [labels, v] : m.counter->values()) {
                out += m.name;
                appendLabelSet(out, staticLabels, labels);
                snprintf(buf, sizeof(buf), " %llu\n",
                         (unsigned long long)v);
                out += buf;
            }
        } else if (m.histogram) {
            const auto& bounds = m.histogram->bounds();
            for (const auto& [labels, s] : m.histogram->series()) {
                uint64_t cum = 0;
                for (size_t i = 0; i <= bounds.size(); ++i) {
                    cum += s.bucketCounts[i];
                    std::string le;
                    if (i < bounds.size()) {
                        snprintf(buf, sizeof(buf), "%g", bounds[i]);
                        le = std::string("le=\"") + buf + "\"";
                    } else {
                        le = "le=\"+Inf\"";
                    }
                    out += m.name + "_bucket";
                    appendLabelSet(out, staticLabels, labels, le);
                    snprintf(buf, sizeof(buf), " %llu\n",
                             (unsigned long long)cum);
                    out += buf;
                }
                out += m.name + "_sum";
                appendLabelSet(out, staticLabels, labels);
                snprintf(buf, sizeof(buf), " %g\n", s.sum);
                out += buf;
                out += m.name + "_count";
                appendLabelSet(out, staticLabels, labels);
                snprintf(buf, sizeof(buf), " %llu\n",
                         (unsigned long long)s.count);
                out += buf;
            }
        }
    }
    return out;
}

std::string renderLabels(
    const std::vector<std::pair<std::string, std::string>>& labels) {
    std::string out;
    bool first = true;
    for (const auto& [k, v] : labels) {
        if (!first) out.push_back(',');
        first = false;
        out += k;
        out += "=\"";
        for (char c : v) {
            if (c == '"' || c == '\\') out.push_back('\\');
            if (c == '\n') {
                out += "\\n";
                continue;
            }
            out.push_back(c);
        }
        out.push_back('"');
    }
    return out;
}

/* ---------------- HTTP exposition ---------------- */

MetricsHttpServer::MetricsHttpServer(EventLoop* loop, Collector* collector,
                                     std::string staticLabels)
    : loop_(loop), collector_(collector),
      staticLabels_(std::move(staticLabels)) {}

MetricsHttpServer::~MetricsHttpServer() { close(); }

bool MetricsHttpServer::listen(const std::string& address, uint16_t port) {
    listenFd_ = socket(AF_INET, SOCK_STREAM | SOCK_NONBLOCK | SOCK_CLOEXEC,
                       0);
    if (listenFd_ < 0) return false;
    int one = 1;
    setsockopt(listenFd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    struct sockaddr_in sa {};
    sa.sin_family = AF_INET;
    sa.sin_port = htons(port);
    if (address.empty() || address == "0.0.0.0")
        sa.sin_addr.s_addr = INADDR_ANY;
    else
        inet_pton(AF_INET, address.c_str(), &sa.sin_addr);
    if (bind(listenFd_, (struct sockaddr*)&sa, sizeof(sa)) != 0 ||
        ::listen(listenFd_, 64) != 0) {
        ::close(listenFd_);
        listenFd_ = -1;
        return false;
    }
    socklen_t slen = sizeof(sa);
    getsockname(listenFd_, (struct sockaddr*)&sa, &slen);
    port_ = ntohs(sa.sin_port);
    loop_->addFd(listenFd_, EPOLLIN, [this](uint32_t) { onAccept(); });
    return true;
}

void MetricsHttpServer::close() {
    if (listenFd_ >= 0) {
        loop_->delFd(listenFd_);
        ::close(listenFd_);
        listenFd_ = -1;
    }
    for (auto& [fd, _] : connBufs_) {
        loop_->delFd(fd);
        ::close(fd);
    }
    connBufs_.clear();
}

void MetricsHttpServer::onAccept() {
    while (true) {
        int fd = accept4(listenFd_, nullptr, nullptr,
                         SOCK_NONBLOCK | SOCK_CLOEXEC);
        if (fd < 0) break;
        if (connBufs_.size() >= 256) {  /* half-open conn guard */
            ::close(fd);
            continue;
        }
        connBufs_[fd] = "";
        loop_->addFd(fd, EPOLLIN,
                     [this, fd](uint32_t ev) { onConn(fd, ev); });
    }
}

void MetricsHttpServer::onConn(int fd, uint32_t events) {
    char buf[4096];
    ssize_t nread = read(fd, buf, sizeof(buf));
    if (nread <= 0) {
        if (nread < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) return;
        loop_->delFd(fd);
        ::close(fd);
        connBufs_.erase(fd);
        return;
    }
    std::string& req = connBufs_[fd];
    req.append(buf, (size_t)nread);
    if (req.find("\r\n\r\n") == std::string::npos &&
        req.find("\n\n") == std::string::npos && req.size() < 65536)
        return;  // keep reading until end of headers

    // Route like the reference's restify mount (triton-metrics serves
    // only GET /metrics; main.js:134-152): anything else is a 404.
    bool found = false;
    if (req.compare(0, 4, "GET ") == 0) {
        size_t sp = req.find(' ', 4);
        if (sp != std::string::npos) {
            std::string_view path(req.data() + 4, sp - 4);
            found = (path == "/metrics");
        }
    }
    std::string resp;
    if (found) {
        std::string body = collector_->expose(staticLabels_);
        char hdr[256];
        snprintf(hdr, sizeof(hdr),
                 "HTTP/1.1 200 OK\r\n"
                 "Content-Type: text/plain; version=0.0.4\r\n"
                 "Content-Length: %zu\r\n"
                 "Connection: close\r\n\r\n",
                 body.size());
        resp = std::string(hdr) + body;
    } else {
        resp = "HTTP/1.1 404 Not Found\r\n"
               "Content-Type: application/json\r\n"
               "Content-Length: 27\r\n"
               "Connection: close\r\n\r\n"
               "{\"code\":\"ResourceNotFound\"}";
    }
    ssize_t rv = write(fd, resp.data(), resp.size());
    (void)rv;
    loop_->delFd(fd);
    ::close(fd);
    connBufs_.erase(fd);
}

}  // namespace bamd
