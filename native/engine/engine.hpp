/*
 * binder-amd: DNS resolution engine.
 *
 * Re-implements the query semantics of /root/reference/lib/server.js
 * (resolve: 136-429, resolvePtr: 67-134, dispatch: 491-506) against the
 * abstract Store. Deliberately preserved reference policies:
 *   - unknown names => REFUSED, not NXDOMAIN (server.js:227-246);
 *   - store-not-ready => SERVFAIL ('eserver', server.js:186-192);
 *   - SRV on a non-service => NODATA + SOA authority (server.js:276-292);
 *   - SRV with wrong service/proto labels => NXDOMAIN (server.js:334-345);
 *   - unsupported qtypes (incl. AAAA!) => NOTIMP (server.js:500-505);
 *   - invalid member record mid-service => SERVFAIL with partial answers
 *     (the reference `break`s out of the loop, server.js:366-376);
 *   - dnsDomain suffix check is case-sensitive and happens before
 *     lowercasing (server.js:156-176 vs 207);
 *   - member shuffle (Fisher-Yates, server.js:40-53, 361).
 * Deviation (documented): the reference's doubled-suffix check
 * (server.js:167-175) can never fire because stripSuffix() appends "..."
 * before the isSuffix test; we implement the evidently intended check.
 * Observable only when recursion is enabled (miss is REFUSED either way).
 */
#pragma once

#include <cstdint>
#include <functional>
#include <random>
#include <string>

#include "../dns/codec.hpp"
#include "store.hpp"

namespace bamd {

struct EngineConfig {
    std::string dnsDomain;        // may be empty (no suffix enforcement)
    std::string datacenterName;
    bool recursionEnabled = false;
};

struct QueryResult {
    enum class Action {
        Respond,   // resp is complete, send it
        Recurse,   // hand off to the recursion forwarder
    };
    Action action = Action::Respond;
    /* For logging parity (server.js:484-490, 537-590): */
    std::string logName;      // stripped name (with "..." marker) or name
    std::string srvLabel;     // "_svc._proto" when SRV-shaped
};

class Engine {
  public:
    Engine(const EngineConfig& cfg, const Store* store)
        : cfg_(cfg), store_(store), rng_(std::random_device{}()) {}

    /*
     * Handle one decoded query; fills `resp` (header/question echo done
     * here). Returns Recurse only when recursion is enabled AND the query
     * had RD AND the name missed the store.
     */
    QueryResult handle(const dns::Message& query, dns::Message& resp);

    void setStore(const Store* s) { store_ = s; }
    const EngineConfig& config() const { return cfg_; }

  private:
    void resolve(const dns::Question& q, bool rd, dns::Message& resp,
                 QueryResult& qr);
    void resolvePtr(const dns::Question& q, bool rd, dns::Message& resp,
                    QueryResult& qr);
    void addSoaAuthority(dns::Message& resp, const std::string& name,
                         uint32_t ttl);

    EngineConfig cfg_;
    const Store* store_;
    std::mt19937 rng_;
};

/* suffix helpers (server.js:55-65). stripSuffix appends "..." — used for
 * log lines only. */
bool isSuffix(const std::string& suffix, const std::string& str);
std::string stripSuffix(const std::string& suffix, const std::string& str);

}  // namespace bamd
