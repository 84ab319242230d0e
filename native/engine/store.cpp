#include "store.hpp"

#include <arpa/inet.h>

#include <algorithm>

namespace bamd {

bool recTypeIsHostLike(RecType t) {
    switch (t) {
    case RecType::Host:
    case RecType::DbHost:
    case RecType::LoadBalancer:
    case RecType::MorayHost:
    case RecType::RedisHost:
    case RecType::OpsHost:
    case RecType::RrHost:
        return true;
    default:
        return false;
    }
}

/*
 * Which member types are served under a service (A/SRV synthesis).
 * Note: 'host' and 'db_host' are deliberately NOT in this set, matching
 * lib/server.js:352-360.
 */
bool recTypeServesUnderService(RecType t) {
    switch (t) {
    case RecType::LoadBalancer:
    case RecType::MorayHost:
    case RecType::OpsHost:
    case RecType::RrHost:
    case RecType::RedisHost:
        return true;
    default:
        return false;
    }
}

RecType recTypeFromString(const std::string& s) {
    if (s == "host") return RecType::Host;
    if (s == "db_host") return RecType::DbHost;
    if (s == "load_balancer") return RecType::LoadBalancer;
    if (s == "moray_host") return RecType::MorayHost;
    if (s == "redis_host") return RecType::RedisHost;
    if (s == "ops_host") return RecType::OpsHost;
    if (s == "rr_host") return RecType::RrHost;
    if (s == "database") return RecType::Database;
    if (s == "service") return RecType::Service;
    return RecType::Unknown;
}

const char* recTypeName(RecType t) {
    switch (t) {
    case RecType::Host: return "host";
    case RecType::DbHost: return "db_host";
    case RecType::LoadBalancer: return "load_balancer";
    case RecType::MorayHost: return "moray_host";
    case RecType::RedisHost: return "redis_host";
    case RecType::OpsHost: return "ops_host";
    case RecType::RrHost: return "rr_host";
    case RecType::Database: return "database";
    case RecType::Service: return "service";
    default: return "unknown";
    }
}

std::string urlHostname(const std::string& url) {
    size_t start = 0;
    size_t scheme = url.find("://");
    if (scheme != std::string::npos) start = scheme + 3;
    size_t end = url.find_first_of("/?#", start);
    std::string auth =
        url.substr(start, end == std::string::npos ? end : end - start);
    size_t at = auth.rfind('@');
    if (at != std::string::npos) auth = auth.substr(at + 1);
    if (!auth.empty() && auth[0] == '[') {
        size_t close = auth.find(']');
        if (close != std::string::npos) return auth.substr(1, close - 1);
    }
    size_t colon = auth.find(':');
    if (colon != std::string::npos) auth = auth.substr(0, colon);
    return auth;
}

CompiledRecord compileRecord(const Json& data) {
    CompiledRecord out;
    /* JS typeof semantics (lib/zk.js:149-155): null and arrays are
     * 'object' and DO get assigned (queries then SERVFAIL on them);
     * strings/numbers/bools are rejected (caller keeps previous data). */
    if (data.isNull() || data.isArray()) {
        out.hasData = true;
        return out;
    }
    if (!data.isObject()) return out;  // ignored by caller
    out.hasData = true;

    const Json& typeJ = data.get("type");
    if (!typeJ.isString()) return out;
    out.typeName = typeJ.asString();
    out.type = recTypeFromString(out.typeName);

    const Json& sub = data.get(out.typeName);
    if (!sub.isObject()) return out;  // record[type] must be an object
    out.valid = true;

    /* Explicit resolver-registry schema (no reference counterpart —
     * the reference only discovers resolvers via UFDS; this is the
     * typed form of the registryDomain convention, recursion.hpp). */
    if (out.typeName == "resolver") {
        if (sub.get("address").isString())
            out.address = sub.get("address").asString();
        if (sub.get("datacenter").isString())
            out.datacenter = sub.get("datacenter").asString();
    }

    /* TTL precedence chain (deepest wins). */
    uint32_t ttl = 30;
    if (data.get("ttl").isNumber()) ttl = (uint32_t)data.get("ttl").asInt();
    if (sub.get("ttl").isNumber()) ttl = (uint32_t)sub.get("ttl").asInt();

    /* Member-level TTL override (server.js:389-393): only krec.ttl /
     * krec[type].ttl — identical inputs, so reuse the chain when set. */
    if (data.get("ttl").isNumber() || sub.get("ttl").isNumber())
        out.memberTtlOverride = ttl;

    if (recTypeIsHostLike(out.type)) {
        const Json& addr = sub.get("address");
        if (addr.isString()) out.address = addr.asString();
        /* non-IPv4 address (garbage, or IPv6 in an A-only server):
         * treat as missing — the reference would abort constructing
         * the ARecord (deviations ledger #3) */
        struct in_addr a4h;
        if (!out.address.empty() &&
            inet_pton(AF_INET, out.address.c_str(), &a4h) != 1)
            out.address.clear();
        const Json& portsJ = sub.get("ports");
        if (portsJ.isArray()) {
            for (const auto& p : portsJ.items())
                if (p.isNumber())
                    out.ports.push_back((uint16_t)p.asInt());
        }
    } else if (out.type == RecType::Database) {
        const Json& prim = sub.get("primary");
        if (prim.isString()) out.address = urlHostname(prim.asString());
        /* a primary URL whose host is not an IPv4 literal cannot be
         * served as an A answer: treat like a missing address (empty
         * NOERROR — the documented deviation for records the
         * reference would abort on) rather than emitting 0.0.0.0 */
        struct in_addr a4;
        if (!out.address.empty() &&
            inet_pton(AF_INET, out.address.c_str(), &a4) != 1)
            out.address.clear();
    } else if (out.type == RecType::Service) {
        /* The service body may be nested one level as service.service
         * (server.js:324-332). */
        const Json* s = &sub;
        const Json& nested = sub.get("service");
        if (nested.isObject()) s = &nested;
        if (s->get("ttl").isNumber()) ttl = (uint32_t)s->get("ttl").asInt();
        if (s->get("srvce").isString()) out.srvce = s->get("srvce").asString();
        if (s->get("proto").isString()) out.proto = s->get("proto").asString();
        if (s->get("port").isNumber()) {
            out.defaultPort = (uint16_t)s->get("port").asInt();
            out.hasDefaultPort = true;
        }
    }
    out.ttl = ttl;
    return out;
}

std::string domainToPath(const std::string& domain) {
    std::vector<std::string> parts;
    size_t start = 0;
    while (start <= domain.size()) {
        size_t dot = domain.find('.', start);
        if (dot == std::string::npos) {
            parts.push_back(domain.substr(start));
            break;
        }
        parts.push_back(domain.substr(start, dot - start));
        start = dot + 1;
    }
    std::string path;
    for (auto it = parts.rbegin(); it != parts.rend(); ++it) {
        path.push_back('/');
        path += *it;
    }
    return path;
}

std::string pathToDomain(const std::string& path) {
    std::vector<std::string> parts;
    size_t start = 1;
    while (start <= path.size()) {
        size_t slash = path.find('/', start);
        if (slash == std::string::npos) {
            parts.push_back(path.substr(start));
            break;
        }
        parts.push_back(path.substr(start, slash - start));
        start = slash + 1;
    }
    std::string domain;
    for (auto it = parts.rbegin(); it != parts.rend(); ++it) {
        if (!domain.empty()) domain.push_back('.');
        domain += *it;
    }
    return domain;
}

/* ---------------- StubStore ---------------- */

static void dnsLower(std::string& s) {
    for (char& c : s)
        if (c >= 'A' && c <= 'Z') c += 32;
}

StubStore::Node::Node(StubStore* store, std::string domain)
    : store_(store), domain_(std::move(domain)) {
    size_t dot = domain_.find('.');
    name_ = dot == std::string::npos ? domain_ : domain_.substr(0, dot);
}

std::vector<const StoreNode*> StubStore::Node::children() const {
    std::vector<const StoreNode*> out;
    out.reserve(childDomains_.size());
    for (const auto& d : childDomains_) {
        auto it = store_->nodes_.find(d);
        if (it != store_->nodes_.end()) out.push_back(it->second.get());
    }
    return out;
}

StubStore::Node* StubStore::ensure(const std::string& domain) {
    auto it = nodes_.find(domain);
    if (it != nodes_.end()) return it->second.get();
    auto node = std::make_unique<Node>(this, domain);
    Node* raw = node.get();
    nodes_[domain] = std::move(node);
    /* Link into parent (create parents up the chain). */
    size_t dot = domain.find('.');
    if (dot != std::string::npos) {
        std::string parent = domain.substr(dot + 1);
        Node* p = ensure(parent);
        if (std::find(p->childDomains_.begin(), p->childDomains_.end(),
                      domain) == p->childDomains_.end())
            p->childDomains_.push_back(domain);
    }
    return raw;
}

void StubStore::reindex(Node* n, const std::string& oldAddr) {
    if (!oldAddr.empty()) {
        auto it = rev_.find(oldAddr);
        if (it != rev_.end() && it->second == n) rev_.erase(it);
    }
    if (recTypeIsHostLike(n->rec_.type) && n->rec_.valid &&
        !n->rec_.address.empty())
        rev_[n->rec_.address] = n;
}

void StubStore::clearParentSvcCache(const std::string& domain) {
    size_t dot = domain.find('.');
    if (dot == std::string::npos) return;
    auto it = nodes_.find(domain.substr(dot + 1));
    if (it != nodes_.end()) it->second->rec_.clearWireCaches();
}

void StubStore::put(const std::string& domain, const Json& data) {
    std::string d = domain;
    dnsLower(d);
    Node* n = ensure(d);
    std::string oldAddr;
    if (recTypeIsHostLike(n->rec_.type)) oldAddr = n->rec_.address;
    CompiledRecord rec = compileRecord(data);
    /* Parity: unparseable / string / number payloads leave previous data
     * in place (lib/zk.js:139-154); null / array / object are assigned. */
    if (rec.hasData) {
        n->rec_ = std::move(rec);
        reindex(n, oldAddr);
        clearParentSvcCache(d);
    }
}

void StubStore::remove(const std::string& domain) {
    std::string d = domain;
    dnsLower(d);
    auto it = nodes_.find(d);
    if (it == nodes_.end()) return;
    Node* n = it->second.get();
    if (!n->rec_.address.empty()) {
        auto rit = rev_.find(n->rec_.address);
        if (rit != rev_.end() && rit->second == n) rev_.erase(rit);
    }
    /* Remove children recursively. */
    std::vector<std::string> kids = n->childDomains_;
    for (const auto& k : kids) remove(k);
    /* Unlink from parent. */
    size_t dot = d.find('.');
    if (dot != std::string::npos) {
        auto pit = nodes_.find(d.substr(dot + 1));
        if (pit != nodes_.end()) {
            auto& cd = pit->second->childDomains_;
            cd.erase(std::remove(cd.begin(), cd.end(), d), cd.end());
        }
    }
    nodes_.erase(d);
    clearParentSvcCache(d);
}

const StoreNode* StubStore::lookup(const std::string& domain) const {
    auto it = nodes_.find(domain);
    return it == nodes_.end() ? nullptr : it->second.get();
}

const StoreNode* StubStore::lookupView(std::string_view domain) const {
    auto it = nodes_.find(domain);
    return it == nodes_.end() ? nullptr : it->second.get();
}

const StoreNode* StubStore::reverseLookup(const std::string& ip) const {
    auto it = rev_.find(ip);
    return it == rev_.end() ? nullptr : it->second;
}

}  // namespace bamd
