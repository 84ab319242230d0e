/*
 * binder-amd: cross-DC recursion forwarder (lib/recursion.js equivalent).
 *
 * Best-effort forwarding of cache misses to other datacenters' binders:
 *   - resolver registry refreshed every 5 minutes
 *     (recursion.js:40, 202-249), init retried every 15 s, 'ready' fires
 *     regardless (174-197);
 *   - non-PTR queries route to the DC named by the label immediately
 *     left of dnsDomain (335-345); PTR fans out to every known resolver
 *     (346-354);
 *   - own-NIC addresses filtered with a 30 s cache (356-376);
 *   - upstream queries have RD cleared and a 3 s timeout (257-261);
 *   - only A/AAAA/TXT/PTR/CNAME/SRV upstream answers are accepted
 *     (299-323); empty results => REFUSED (292-296).
 *
 * Registry sources are pluggable where the reference hardcodes UFDS/LDAP
 * (its listResolvers(region) call, recursion.js:210-219):
 *   "static": {"source":"static","dcs":{"dc1":["10.0.0.5",...]}}
 *   "zk":     {"source":"zk","registryDomain":"resolvers.<domain>"}.
 *             Children of that domain are resolver entries; the typed
 *             schema is {"type":"resolver","resolver":{"datacenter":
 *             "<dc>","address":"<ip>"}} (explicit dc); host-like
 *             children keyed "<dc>-<n>" are accepted as the legacy
 *             name-prefix form.
 *   "ufds":   the reference's own path: resolve the UFDS address from
 *             the mirror (recursion.js:104-127), then LDAP-search
 *             resolvers for the region (ldap.hpp) every 5 minutes on a
 *             helper thread.
 */
#pragma once

#include <functional>
#include <map>
#include <memory>
#include <random>
#include <string>
#include <thread>
#include <vector>

#include "../common/json.hpp"
#include "../common/log.hpp"
#include "../common/loop.hpp"
#include "../engine/store.hpp"
#include "ldap.hpp"
#include "server.hpp"

namespace bamd {

struct RecursionOptions {
    std::string regionName;
    std::string datacenterName;
    std::string dnsDomain;
    uint16_t upstreamPort = 53;  // tests point this at ephemeral ports
    Json config;  // the whole `recursion` config block
};

class Recursion : public RecursionIface {
  public:
    Recursion(EventLoop* loop, Logger log, RecursionOptions opts,
              const Store* store);
    ~Recursion() override;

    void init();
    bool ready() const { return ready_; }
    void onReady(std::function<void()> cb) { readyCb_ = std::move(cb); }

    /* RecursionIface */
    void resolve(const dns::Message& query, dns::Message& resp,
                 std::function<void()> done) override;

    /* Direct registry injection (tests / static source). */
    void setDcs(std::map<std::string, std::vector<std::string>> dcs) {
        dcs_ = std::move(dcs);
    }
    const std::map<std::string, std::vector<std::string>>& dcs() const {
        return dcs_;
    }

  private:
    /*
     * One in-flight lookup. Lookups draw a random socket from a pool
     * of pre-bound ephemeral-port sockets (the production-resolver
     * "outgoing port range" design: 256 unpredictable source ports x
     * a random qid ~ 24 bits an off-path forger must hit) plus a
     * random qid, and record the upstream addresses actually queried:
     * replies are only accepted when source address+port, qid AND
     * question section all match. A fresh socket per lookup gives the
     * same properties but costs socket+epoll+close syscalls per query
     * (measured 250k -> 92k qps on config 4); the pool keeps the
     * entropy without the churn.
     */
    struct Upstream {
        int fd = -1;
        int poolIdx = -1;
        uint16_t qid;
        std::vector<std::string> hosts;  // remaining unsent
        std::vector<uint32_t> queried;   // in_addr.s_addr actually sent to
        std::vector<uint8_t> wire;       // encoded outgoing query
        dns::Question question;          // for reply validation
        int inFlight = 0;
        int errors = 0;
        int maxConcurrency;
        dns::Message* resp;
        std::function<void()> done;
        std::string qname;
        uint64_t timeoutTimer = 0;
        bool finished = false;
    };

    void refresh();
    void scheduleRefresh(int64_t ms);
    void emitReady();
    std::vector<std::string> ownAddrs();
    bool ensurePool();
    void onPoolReadable(int idx);
    void sendNext(const std::shared_ptr<Upstream>& up);
    void finish(const std::shared_ptr<Upstream>& up,
                const dns::Message* answer);

    EventLoop* loop_;
    Logger log_;
    RecursionOptions opts_;
    const Store* store_;

    std::map<std::string, std::vector<std::string>> dcs_;
    bool ready_ = false;
    std::function<void()> readyCb_;
    uint64_t refreshTimer_ = 0;

    std::mt19937 rng_;   // seeded from std::random_device (qid entropy)
    /* outgoing socket pool: fds + per-socket qid->lookup maps */
    static constexpr int kPoolSize = 256;
    std::vector<int> poolFds_;
    std::vector<std::map<uint16_t, std::shared_ptr<Upstream>>> poolPending_;
    int activeLookups_ = 0;  // bounds total in-flight lookups

    std::vector<std::string> nicCache_;
    int64_t nicCacheAtMs_ = 0;

    /* UFDS/LDAP refresh runs on a helper thread (blocking client);
     * results are posted back via EventLoop::postFromThread. The
     * connection persists across 5-minute refreshes like the
     * reference's long-lived UFDS client (recursion.js:129-148);
     * only the helper thread touches ldapClient_ (ldapBusy_ gates). */
    std::thread ldapThread_;
    bool ldapBusy_ = false;
    std::unique_ptr<ldap::Client> ldapClient_;
    void refreshViaUfds();
};

}  // namespace bamd
