#include "recursion.hpp"

#include "ldap.hpp"

#include <arpa/inet.h>
#include <ifaddrs.h>
#include <netinet/in.h>
#include <sys/epoll.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cstring>

namespace bamd {

using namespace dns;

static constexpr int64_t kRefreshIntervalMs = 5 * 60 * 1000;
static constexpr int64_t kRetryInitMs = 15 * 1000;
static constexpr int64_t kUpstreamTimeoutMs = 3000;

Recursion::Recursion(EventLoop* loop, Logger log, RecursionOptions opts,
                     const Store* store)
    : loop_(loop),
      log_(log.child({{"component", Json("Recursion")}})),
      opts_(std::move(opts)), store_(store),
      rng_(std::random_device{}()) {}

Recursion::~Recursion() {
    if (ldapThread_.joinable()) ldapThread_.join();
    if (refreshTimer_) loop_->cancelTimer(refreshTimer_);
    for (int fd : poolFds_)
        if (fd >= 0) {
            loop_->delFd(fd);
            close(fd);
        }
}

bool Recursion::ensurePool() {
    if (!poolFds_.empty()) return true;
    poolFds_.reserve(kPoolSize);
    poolPending_.resize(kPoolSize);
    for (int i = 0; i < kPoolSize; ++i) {
        int fd = socket(AF_INET, SOCK_DGRAM | SOCK_NONBLOCK |
                        SOCK_CLOEXEC, 0);
        if (fd < 0) break;
        poolFds_.push_back(fd);
        loop_->addFd(fd, EPOLLIN,
                     [this, i](uint32_t) { onPoolReadable(i); });
    }
    return !poolFds_.empty();
}

void Recursion::emitReady() {
    if (ready_) return;
    ready_ = true;
    if (readyCb_) readyCb_();
}

void Recursion::init() {
    log_.info("Initing Clients...");
    refresh();
}

void Recursion::scheduleRefresh(int64_t ms) {
    if (refreshTimer_) loop_->cancelTimer(refreshTimer_);
    refreshTimer_ = loop_->addTimer(ms, [this]() {
        refreshTimer_ = 0;
        refresh();
    });
}

void Recursion::refresh() {
    const Json& cfg = opts_.config;
    std::string source = cfg.get("source").asString();
    if (source.empty()) {
        if (cfg.get("dcs").isObject())
            source = "static";
        else if (cfg.get("ufds").isObject())
            source = "ufds";  /* the reference's config shape */
        else
            source = "zk";
    }

    if (source == "ufds") {
        refreshViaUfds();
        return;
    }

    if (source == "static") {
        std::map<std::string, std::vector<std::string>> dcs;
        for (const auto& [dc, ips] : cfg.get("dcs").fields()) {
            for (const auto& ip : ips.items())
                if (ip.isString())
                    dcs[dc].push_back(ip.asString());
        }
        dcs_ = std::move(dcs);
        log_.info({{"dcs", Json((int64_t)dcs_.size())}},
                  "setting recursion resolvers (static)");
        emitReady();
        scheduleRefresh(kRefreshIntervalMs);
        return;
    }

    if (source == "zk") {
        /* Bootstrap from the mirror itself: the resolver registry lives
         * at a domain inside our own tree (pluggable replacement for the
         * reference's UFDS-over-LDAP listResolvers; recursion.js:104-127
         * similarly bootstraps UFDS's address from the ZK cache). */
        std::string domain = cfg.get("registryDomain").asString();
        if (domain.empty() || !store_->ready()) {
            log_.warn("unable to refresh resolvers yet; will try again "
                      "in 15 seconds (best effort)");
            emitReady();  // 'ready' fires regardless (recursion.js:174-197)
            scheduleRefresh(kRetryInitMs);
            return;
        }
        const StoreNode* node = store_->lookup(domain);
        if (node == nullptr) {
            log_.warn({{"domain", Json(domain)}},
                      "resolver registry domain not found yet");
            emitReady();
            scheduleRefresh(kRetryInitMs);
            return;
        }
        /* Registry children, two accepted shapes (schema documented
         * in recursion.hpp):
         *   1. typed (preferred): {"type":"resolver","resolver":
         *      {"datacenter":"dc2","address":"10.0.0.5"}} — the
         *      datacenter is explicit;
         *   2. legacy host-like entries named "<dc>-<n>", where the
         *      dc comes from the name prefix. */
        std::map<std::string, std::vector<std::string>> dcs;
        for (const StoreNode* kid : node->children()) {
            const CompiledRecord& r = kid->rec();
            if (!r.valid || r.address.empty()) continue;
            std::string dc;
            if (!r.datacenter.empty()) {
                dc = r.datacenter;
            } else {
                dc = kid->name();
                size_t dash = dc.rfind('-');
                if (dash != std::string::npos) dc = dc.substr(0, dash);
            }
            auto& v = dcs[dc];
            bool dup = false;
            for (const auto& e : v) dup = dup || e == r.address;
            if (!dup) v.push_back(r.address);
        }
        dcs_ = std::move(dcs);
        log_.info({{"dcs", Json((int64_t)dcs_.size())}},
                  "setting recursion resolvers (zk)");
        emitReady();
        scheduleRefresh(kRefreshIntervalMs);
        return;
    }

    log_.warn({{"source", Json(source)}},
              "unknown resolver source; recursion disabled "
              "(best effort)");
    emitReady();
    scheduleRefresh(kRetryInitMs);
}

/*
 * UFDS source: bootstrap the LDAP address from our own mirror, then
 * listResolvers(region) — recursion.js:104-127 + 202-249. The blocking
 * LDAP conversation runs on a helper thread; only the result lands on
 * the loop thread.
 */
void Recursion::refreshViaUfds() {
    if (ldapBusy_) {
        scheduleRefresh(kRetryInitMs);
        return;
    }
    const Json& ucfg = opts_.config.get("ufds");
    std::string url = ucfg.get("url").asString();
    bool tls = url.rfind("ldaps://", 0) == 0;
    std::string domain = url;
    for (const char* scheme : {"ldaps://", "ldap://"}) {
        if (domain.rfind(scheme, 0) == 0) {
            domain = domain.substr(strlen(scheme));
            break;
        }
    }
    uint16_t port = tls ? 636 : 389;
    size_t colon = domain.rfind(':');
    if (colon != std::string::npos &&
        domain.find(':') == colon /* not v6 */) {
        port = (uint16_t)atoi(domain.c_str() + colon + 1);
        domain = domain.substr(0, colon);
    }

    /* resolveUfds: must be a service node with a first child */
    std::string addr = domain;
    bool isIp = true;
    for (char c : domain)
        isIp = isIp && ((c >= '0' && c <= '9') || c == '.');
    if (!isIp) {
        if (!store_->ready()) {
            log_.warn("Recursion: ZK is not yet available");
            emitReady();
            scheduleRefresh(kRetryInitMs);
            return;
        }
        const StoreNode* node = store_->lookup(domain);
        const StoreNode* kid = nullptr;
        if (node != nullptr && node->rec().type == RecType::Service) {
            auto kids = node->children();
            if (!kids.empty()) kid = kids[0];
        }
        if (kid == nullptr || !kid->rec().valid ||
            kid->rec().address.empty()) {
            log_.warn("Recursion: not yet able to resolve ufds");
            emitReady();
            scheduleRefresh(kRetryInitMs);
            return;
        }
        addr = kid->rec().address;
    }

    ldap::Options lopts;
    lopts.host = addr;
    lopts.port = port;
    lopts.tls = tls;
    lopts.tlsVerify = ucfg.get("tlsVerify").asBool(true);
    lopts.caFile = ucfg.get("caFile").asString();
    lopts.bindDn = ucfg.get("bindDN").asString();
    lopts.bindPassword = ucfg.get("bindPassword").asString();
    std::string region = opts_.regionName;

    ldapBusy_ = true;
    if (ldapThread_.joinable()) ldapThread_.join();
    EventLoop* loop = loop_;
    Recursion* self = this;
    ldapThread_ = std::thread([self, loop, lopts, region]() {
        /* reuse the held connection when the target is unchanged */
        ldap::Client* client = self->ldapClient_.get();
        bool fresh = false;
        if (client == nullptr || !client->isConnected() ||
            client->options().host != lopts.host ||
            client->options().port != lopts.port) {
            self->ldapClient_ = std::make_unique<ldap::Client>(lopts);
            client = self->ldapClient_.get();
            fresh = true;
        }
        std::vector<ldap::Entry> entries;
        std::string errMsg;
        std::string base = "region=" + region + ", o=smartdc";
        bool ok = true;
        if (fresh) ok = client->connect();
        if (ok)
            ok = client->search(base, "objectclass", "resolver",
                                entries);
        if (!ok && !fresh) {
            /* held connection went stale: reconnect once */
            entries.clear();
            self->ldapClient_ = std::make_unique<ldap::Client>(lopts);
            client = self->ldapClient_.get();
            ok = client->connect() &&
                 client->search(base, "objectclass", "resolver",
                                entries);
        }
        if (!ok) {
            errMsg = client->error();
            self->ldapClient_.reset();  /* reconnect next refresh */
        }
        loop->postFromThread([self, ok, errMsg,
                              entries = std::move(entries)]() {
            self->ldapBusy_ = false;
            if (!ok) {
                self->log_.warn(
                    {{"err", Json(errMsg)}},
                    "Recursion: Binder is configured for recursive dns "
                    "but is unable to reach UFDS. Will try again in 15 "
                    "seconds (best effort).");
                self->emitReady();
                self->scheduleRefresh(kRetryInitMs);
                return;
            }
            std::map<std::string, std::vector<std::string>> dcs;
            for (const auto& e : entries) {
                auto dcIt = e.find("datacenter");
                auto ipIt = e.find("ip");
                if (dcIt == e.end() || ipIt == e.end() ||
                    dcIt->second.empty() || ipIt->second.empty())
                    continue;
                const std::string& dc = dcIt->second[0];
                const std::string& ip = ipIt->second[0];
                auto& v = dcs[dc];
                bool dup = false;
                for (const auto& x : v) dup = dup || x == ip;
                if (!dup) v.push_back(ip);
            }
            self->dcs_ = std::move(dcs);
            self->log_.info({{"dcs", Json((int64_t)self->dcs_.size())}},
                            "setting recursion resolvers (ufds)");
            self->emitReady();
            self->scheduleRefresh(kRefreshIntervalMs);
        });
    });
}

std::vector<std::string> Recursion::ownAddrs() {
    int64_t now = monotonicMillis();
    if (!nicCache_.empty() && now - nicCacheAtMs_ <= 30000)
        return nicCache_;
    std::vector<std::string> out;
    struct ifaddrs* ifap = nullptr;
    if (getifaddrs(&ifap) == 0) {
        for (struct ifaddrs* ifa = ifap; ifa; ifa = ifa->ifa_next) {
            if (ifa->ifa_addr == nullptr) continue;
            char buf[INET6_ADDRSTRLEN] = {0};
            if (ifa->ifa_addr->sa_family == AF_INET) {
                auto* sa = (struct sockaddr_in*)ifa->ifa_addr;
                inet_ntop(AF_INET, &sa->sin_addr, buf, sizeof(buf));
            } else if (ifa->ifa_addr->sa_family == AF_INET6) {
                auto* sa = (struct sockaddr_in6*)ifa->ifa_addr;
                inet_ntop(AF_INET6, &sa->sin6_addr, buf, sizeof(buf));
            } else {
                continue;
            }
            out.push_back(buf);
        }
        freeifaddrs(ifap);
    }
    nicCache_ = out;
    nicCacheAtMs_ = now;
    return out;
}

void Recursion::resolve(const Message& query, Message& resp,
                        std::function<void()> done) {
    const std::string domain =
        query.questions.empty() ? "" : query.questions[0].name;
    uint16_t qtype =
        query.questions.empty() ? (uint16_t)TYPE_A
                                : query.questions[0].qtype;
    bool isPtr = qtype == TYPE_PTR;

    auto refuse = [&]() {
        resp.header.rcode = RCODE_REFUSED;
        done();
    };

    /* Right dns domain? (recursion.js:330-333) */
    if (!isPtr && !opts_.dnsDomain.empty()) {
        if (domain.size() < opts_.dnsDomain.size() ||
            domain.compare(domain.size() - opts_.dnsDomain.size(),
                           opts_.dnsDomain.size(), opts_.dnsDomain) != 0) {
            refuse();
            return;
        }
    }

    std::vector<std::string> upstreams;
    if (!isPtr) {
        /* DC label immediately left of dnsDomain (recursion.js:335-345) */
        if (domain.size() <= opts_.dnsDomain.size()) {
            refuse();
            return;
        }
        std::string p = domain.substr(
            0, domain.size() - opts_.dnsDomain.size() - 1);
        size_t dot = p.rfind('.');
        std::string dc = dot == std::string::npos ? p : p.substr(dot + 1);
        auto it = dcs_.find(dc);
        if (it == dcs_.end()) {
            refuse();
            return;
        }
        upstreams = it->second;
    } else {
        for (const auto& [dc, ips] : dcs_)
            for (const auto& ip : ips) upstreams.push_back(ip);
    }

    /* Filter own NICs (recursion.js:356-376). */
    std::vector<std::string> mine = ownAddrs();
    std::vector<std::string> filtered;
    for (const auto& u : upstreams) {
        bool self = false;
        for (const auto& m : mine) self = self || m == u;
        if (!self) filtered.push_back(u);
    }
    if (filtered.empty()) {
        refuse();
        return;
    }
    if (activeLookups_ >= 4096 || !ensurePool()) {
        /* bound in-flight lookups: shed load (best effort) */
        refuse();
        return;
    }

    auto up = std::make_shared<Upstream>();
    /* random socket from the outgoing pool + random non-colliding qid
     * (the port+qid pair is what an off-path forger must guess) */
    up->poolIdx = (int)(rng_() % poolFds_.size());
    up->fd = poolFds_[up->poolIdx];
    auto& pending = poolPending_[up->poolIdx];
    uint16_t qid = 0;
    for (int tries = 0; tries < 16; ++tries) {
        qid = (uint16_t)(rng_() & 0xffff);
        if (qid != 0 && pending.count(qid) == 0) break;
        qid = 0;
    }
    if (qid == 0) {
        refuse();
        return;
    }
    up->qid = qid;
    up->hosts = std::move(filtered);
    up->maxConcurrency = isPtr ? 100 : 2;  // recursion.js:64-78
    up->resp = &resp;
    up->done = std::move(done);
    up->qname = domain;
    up->question = query.questions.empty() ? Question{} : query.questions[0];

    /* outgoing query: same question, rd cleared (recursion.js:258-261) */
    Message out;
    out.header.id = up->qid;
    out.header.rd = false;
    out.questions = query.questions;
    up->wire = out.encode(0);

    activeLookups_++;
    poolPending_[up->poolIdx][up->qid] = up;

    up->timeoutTimer = loop_->addTimer(kUpstreamTimeoutMs, [this, up]() {
        up->timeoutTimer = 0;
        finish(up, nullptr);
    });

    sendNext(up);
}

void Recursion::sendNext(const std::shared_ptr<Upstream>& up) {
    while (up->inFlight < up->maxConcurrency && !up->hosts.empty()) {
        std::string host = up->hosts.front();
        up->hosts.erase(up->hosts.begin());
        struct sockaddr_in sa {};
        sa.sin_family = AF_INET;
        sa.sin_port = htons(opts_.upstreamPort);
        if (inet_pton(AF_INET, host.c_str(), &sa.sin_addr) != 1) {
            up->errors++;
            continue;
        }
        ssize_t rv = sendto(up->fd, up->wire.data(), up->wire.size(), 0,
                            (struct sockaddr*)&sa, sizeof(sa));
        if (rv < 0) {
            up->errors++;
            continue;
        }
        up->queried.push_back(sa.sin_addr.s_addr);
        up->inFlight++;
    }
    if (up->inFlight == 0) finish(up, nullptr);
}

void Recursion::onPoolReadable(int idx) {
    uint8_t buf[4096];
    int fd = poolFds_[idx];
    auto& pending = poolPending_[idx];
    while (true) {
        struct sockaddr_in src {};
        socklen_t slen = sizeof(src);
        ssize_t nr = recvfrom(fd, buf, sizeof(buf), 0,
                              (struct sockaddr*)&src, &slen);
        if (nr <= 0) return;
        if (nr < 12) continue;
        /* match the lookup by qid on THIS socket */
        uint16_t qid = (uint16_t)(((uint16_t)buf[0] << 8) | buf[1]);
        auto it = pending.find(qid);
        if (it == pending.end()) continue;
        auto up = it->second;
        if (up->finished) continue;
        /* Source must be an upstream we actually queried, replying
         * from the DNS port we sent to. */
        if (src.sin_family != AF_INET ||
            src.sin_port != htons(opts_.upstreamPort))
            continue;
        bool known = false;
        for (uint32_t a : up->queried)
            known = known || a == src.sin_addr.s_addr;
        if (!known) continue;
        auto msg = Message::decode(buf, (size_t)nr);
        if (!msg || !msg->header.qr) continue;
        if (msg->header.id != up->qid) continue;
        /* The echoed question must match what we asked
         * (case-insensitive, compared in place). */
        if (msg->questions.size() != 1) continue;
        const std::string& qn = msg->questions[0].name;
        const std::string& expect = up->question.name;
        bool sameName = qn.size() == expect.size();
        for (size_t i = 0; sameName && i < qn.size(); ++i) {
            char a = qn[i], b = expect[i];
            if (a >= 'A' && a <= 'Z') a = (char)(a + 32);
            if (b >= 'A' && b <= 'Z') b = (char)(b + 32);
            sameName = a == b;
        }
        if (!sameName ||
            msg->questions[0].qtype != up->question.qtype ||
            msg->questions[0].qclass != up->question.qclass)
            continue;
        up->inFlight--;
        if (msg->header.rcode == RCODE_NOERROR && !msg->answers.empty()) {
            finish(up, &*msg);
        } else {
            up->errors++;
            if (up->inFlight == 0 && up->hosts.empty())
                finish(up, nullptr);
            else if (!up->hosts.empty())
                sendNext(up);  /* keep trying further resolvers */
        }
    }
}

void Recursion::finish(const std::shared_ptr<Upstream>& up,
                       const Message* answer) {
    if (up->finished) return;
    up->finished = true;
    if (up->timeoutTimer) {
        loop_->cancelTimer(up->timeoutTimer);
        up->timeoutTimer = 0;
    }
    if (up->poolIdx >= 0) {
        poolPending_[up->poolIdx].erase(up->qid);
        up->poolIdx = -1;
        activeLookups_--;
    }

    Message& resp = *up->resp;
    size_t accepted = 0;
    if (answer != nullptr) {
        for (const auto& rec : answer->answers) {
            /* accepted types only (recursion.js:299-323) */
            switch (rec.type) {
            case TYPE_A:
            case TYPE_AAAA:
            case TYPE_TXT:
            case TYPE_PTR:
            case TYPE_CNAME:
            case TYPE_SRV: {
                Record r = rec;
                r.name = up->qname;  // answers under the original name
                resp.answers.push_back(std::move(r));
                accepted++;
                break;
            }
            default:
                log_.warn("upstream ns returned unsupported record "
                          "type, dropping");
                break;
            }
        }
    }
    if (accepted == 0) resp.header.rcode = RCODE_REFUSED;
    up->done();
}

}  // namespace bamd
