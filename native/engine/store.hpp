/*
 * binder-amd: record store interface + record compilation.
 *
 * The reference keeps raw JSON-parsed objects in its ZK mirror and walks
 * them on every query (/root/reference/lib/server.js:249-424). We instead
 * compile each znode's JSON payload into a flat CompiledRecord ONCE at
 * update time (mirror watch delivery), so the query hot path touches only
 * PODs and interned strings. Two Store implementations exist:
 *   - StubStore: in-memory, fed directly (tests + BASELINE config 1);
 *   - ZkMirror: watch-driven full mirror of a ZooKeeper tree (native/zk/).
 */
#pragma once

#include <cstdint>
#include <memory>
#include <optional>
#include <string>
#include <string_view>
#include <unordered_map>
#include <vector>

#include "../common/json.hpp"

namespace bamd {

/*
 * Registration record types understood by binder
 * (lib/server.js:302-308, 352-360; lib/zk.js:172-189).
 */
enum class RecType : uint8_t {
    Unknown = 0,
    Host,
    DbHost,
    LoadBalancer,
    MorayHost,
    RedisHost,
    OpsHost,
    RrHost,
    Database,
    Service,
};

bool recTypeIsHostLike(RecType t);       // the 7 *_host/load_balancer types
bool recTypeServesUnderService(RecType t);  // server.js:352-360 filter set
RecType recTypeFromString(const std::string& s);
const char* recTypeName(RecType t);

struct CompiledRecord {
    bool hasData = false;   // znode payload parsed to a JSON object
    bool valid = false;     // .type is a string AND [.type] is an object
    RecType type = RecType::Unknown;
    std::string typeName;

    /* host-like + database (URL-parsed hostname) + resolver: */
    std::string address;    // empty = null/absent address
    /* resolver records only ({"type":"resolver","resolver":
     * {"datacenter":...,"address":...}}) — the explicit schema for
     * the recursion resolver registry (recursion.hpp). */
    std::string datacenter;
    /* TTL via the precedence chain: 30 -> record.ttl -> record[type].ttl
     * -> (service, nested) record.service.service.ttl
     * (lib/server.js:262-274, 324-332). */
    uint32_t ttl = 30;
    /* krec.ttl / krec[type].ttl override when this node is a service
     * member (server.js:389-393). */
    std::optional<uint32_t> memberTtlOverride;

    /* service-only (after the optional service.service dive): */
    std::string srvce;      // e.g. "_http"
    std::string proto;      // e.g. "_tcp"
    uint16_t defaultPort = 0;
    bool hasDefaultPort = false;

    /* member-only: per-member port list (server.js:383-385) */
    std::vector<uint16_t> ports;

    /* binderd fast-path caches (built lazily in server.cpp; a fresh
     * CompiledRecord starts empty, so recompiles invalidate them; the
     * stores additionally clear the PARENT's service cache when a
     * member changes). */
    mutable std::vector<uint8_t> wireA;  /* host-like A response */
    mutable std::string logA;  /* pre-escaped answers element for wireA */
    /* service responses, segment-permuted per query to preserve the
     * reference's Fisher-Yates member shuffle: */
    struct ServiceCache {
        bool usable = false;     /* false => always slow path */
        std::vector<uint8_t> headA;    /* header+question, A query */
        std::vector<uint8_t> headSrv;  /* header+question, SRV query */
        /* per member: plain-A answer segment / SRV answer segment(s) /
         * additional-A segment, plus the matching pre-escaped log-line
         * fragments (comma-joined quoted JSON elements) so the fast
         * path can emit the per-query info log without the slow path */
        struct Member {
            std::vector<uint8_t> aSeg;
            std::vector<uint8_t> srvSeg;
            std::vector<uint8_t> addSeg;
            std::string aLog, srvLog, addLog;
        };
        std::vector<Member> members;
        uint16_t srvAnCount = 0;  /* total SRV records over members */
    };
    mutable std::unique_ptr<ServiceCache> svc;
    void clearWireCaches() const {
        wireA.clear();
        logA.clear();
        svc.reset();
    }
};

/* Compile a znode JSON payload. `parsed`=false when payload was not valid
 * JSON (such nodes are kept but data is ignored; lib/zk.js:139-154). */
CompiledRecord compileRecord(const Json& data);

class StoreNode {
  public:
    virtual ~StoreNode() = default;
    virtual const CompiledRecord& rec() const = 0;
    virtual const std::string& domain() const = 0;   // fqdn, lowercase
    virtual const std::string& name() const = 0;     // leftmost label
    virtual std::vector<const StoreNode*> children() const = 0;
};

/* heterogeneous lookup support: find by string_view without building
 * a std::string per query */
struct SvHash {
    using is_transparent = void;
    size_t operator()(std::string_view sv) const {
        return std::hash<std::string_view>{}(sv);
    }
    size_t operator()(const std::string& s) const {
        return std::hash<std::string_view>{}(s);
    }
};
struct SvEq {
    using is_transparent = void;
    bool operator()(std::string_view a, std::string_view b) const {
        return a == b;
    }
};

class Store {
  public:
    virtual ~Store() = default;
    virtual const StoreNode* lookup(const std::string& domain) const = 0;
    virtual const StoreNode* lookupView(std::string_view domain) const {
        return lookup(std::string(domain));
    }
    virtual const StoreNode* reverseLookup(const std::string& ip) const = 0;
    virtual bool ready() const = 0;
};

/*
 * StubStore: hash-map store fed directly with (domain -> JSON payload)
 * pairs; maintains the same reverse IP index the mirror does.
 */
class StubStore : public Store {
  public:
    class Node : public StoreNode {
      public:
        Node(StubStore* store, std::string domain);
        const CompiledRecord& rec() const override { return rec_; }
        const std::string& domain() const override { return domain_; }
        const std::string& name() const override { return name_; }
        std::vector<const StoreNode*> children() const override;

        StubStore* store_;
        std::string domain_;
        std::string name_;
        CompiledRecord rec_;
        std::vector<std::string> childDomains_;
    };

    /* Set/replace a node's payload; creates intermediate parents. */
    void put(const std::string& domain, const Json& data);
    void remove(const std::string& domain);
    void setReady(bool r) { ready_ = r; }

    const StoreNode* lookup(const std::string& domain) const override;
    const StoreNode* lookupView(std::string_view domain) const override;
    const StoreNode* reverseLookup(const std::string& ip) const override;
    bool ready() const override { return ready_; }

  private:
    friend class Node;
    Node* ensure(const std::string& domain);
    void reindex(Node* n, const std::string& oldAddr);
    void clearParentSvcCache(const std::string& domain);

    std::unordered_map<std::string, std::unique_ptr<Node>, SvHash, SvEq>
        nodes_;
    std::unordered_map<std::string, Node*> rev_;
    bool ready_ = true;
};

/* "foo.com" -> "/com/foo" and back (lib/zk.js:225-228). */
std::string domainToPath(const std::string& domain);
std::string pathToDomain(const std::string& path);

/* Extract hostname from a URL like "tcp://user@host:123/db"
 * (lib/server.js:296-300 uses node's url.parse().hostname). */
std::string urlHostname(const std::string& url);

}  // namespace bamd
