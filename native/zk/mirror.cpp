#include "mirror.hpp"

namespace bamd {

using namespace zk;

static void lowerAscii(std::string& s) {
    for (char& c : s)
        if (c >= 'A' && c <= 'Z') c += 32;
}

ZkMirror::Node::Node(ZkMirror* m, std::string domain)
    : mirror_(m), domain_(std::move(domain)) {
    size_t dot = domain_.find('.');
    name_ = dot == std::string::npos ? domain_ : domain_.substr(0, dot);
    path_ = domainToPath(domain_);
    lowerAscii(domain_);  // tn_domain lowercased (zk.js:84)
}

std::vector<const StoreNode*> ZkMirror::Node::children() const {
    std::vector<const StoreNode*> out;
    out.reserve(kids_.size());
    for (const auto& [label, n] : kids_) out.push_back(n);
    return out;
}

ZkMirror::ZkMirror(EventLoop* loop, Logger log, ZkMirrorOptions opts,
                   Collector* collector)
    : loop_(loop),
      log_(log.child({{"component", Json("ZkMirror")}})),
      opts_(std::move(opts)) {
    ZkConfig zc;
    zc.host = opts_.host;
    zc.port = opts_.port;
    zc.sessionTimeoutMs = opts_.sessionTimeoutMs;
    client_ = std::make_unique<ZkClient>(loop_, log_, zc);
    client_->onSession([this]() { rebuild(); });
    client_->onWatch([this](int32_t type, const std::string& path) {
        onWatch(type, path);
    });
    if (collector != nullptr) {
        nodesGauge_ = collector->gauge("binder_zk_mirror_nodes",
                                       "znodes currently mirrored");
        sessionsCounter_ =
            collector->counter("binder_zk_sessions_established",
                               "ZooKeeper sessions established");
    }
}

ZkMirror::~ZkMirror() { stop(); }

void ZkMirror::start() { client_->start(); }

void ZkMirror::stop() {
    if (client_) client_->close();
}

const StoreNode* ZkMirror::lookup(const std::string& domain) const {
    auto it = byDomain_.find(domain);
    return it == byDomain_.end() ? nullptr : it->second;
}

const StoreNode* ZkMirror::lookupView(std::string_view domain) const {
    auto it = byDomain_.find(domain);
    return it == byDomain_.end() ? nullptr : it->second;
}

const StoreNode* ZkMirror::reverseLookup(const std::string& ip) const {
    auto it = rev_.find(ip);
    return it == rev_.end() ? nullptr : it->second;
}

bool ZkMirror::ready() const {
    /* true once the root node object exists, i.e. after the first
     * session event — even before data lands (zk.js:55-58). */
    return byDomain_.count(opts_.domain) > 0;
}

ZkMirror::Node* ZkMirror::nodeAt(const std::string& path) {
    auto it = byPath_.find(path);
    return it == byPath_.end() ? nullptr : it->second.get();
}

void ZkMirror::rebuild() {
    if (sessionsCounter_) sessionsCounter_->increment("");
    if (root_ == nullptr) {
        auto n = std::make_unique<Node>(this, opts_.domain);
        root_ = n.get();
        byDomain_[root_->domain_] = root_;
        byPath_[root_->path_] = std::move(n);
    }
    /* Re-bind the whole existing tree with fresh reads+watches. */
    std::vector<Node*> stack{root_};
    while (!stack.empty()) {
        Node* n = stack.back();
        stack.pop_back();
        bind(n);
        for (auto& [label, kid] : n->kids_) stack.push_back(kid);
    }
    log_.info({{"nodes", Json((int64_t)byPath_.size())}},
              "ZK session established; mirror resync started");
}

void ZkMirror::bind(Node* n) {
    const std::string path = n->path_;
    client_->getData(path, true,
                     [this, path](int32_t rc, const std::string& data,
                                  const Stat&) {
                         if (rc == ZOK) onData(path, data);
                     });
    client_->getChildren(
        path, true,
        [this, path](int32_t rc, const std::vector<std::string>& kids) {
            if (rc == ZOK) onChildren(path, kids);
        });
}

void ZkMirror::onWatch(int32_t type, const std::string& path) {
    Node* n = nodeAt(path);
    if (n == nullptr) return;  // removed meanwhile
    switch (type) {
    case EV_NODE_DATA_CHANGED:
    case EV_NODE_CREATED:
    case EV_NODE_DELETED: {
        /* NodeDeleted also re-probes: if the znode was deleted and
         * recreated before our (delete-triggered) getChildren ran, the
         * parent diff keeps this node object — whose data watch was
         * just consumed. Re-reading re-arms it on the recreated node;
         * a true deletion answers ZNONODE (no watch) and the parent's
         * childrenChanged removes the node. */
        const std::string p = path;
        client_->getData(p, true,
                         [this, p](int32_t rc, const std::string& data,
                                   const Stat&) {
                             if (rc == ZOK) onData(p, data);
                         });
        break;
    }
    case EV_NODE_CHILDREN_CHANGED: {
        const std::string p = path;
        client_->getChildren(
            p, true,
            [this, p](int32_t rc, const std::vector<std::string>& kids) {
                if (rc == ZOK) onChildren(p, kids);
            });
        break;
    }
    default:
        break;
    }
}

void ZkMirror::onChildren(const std::string& path,
                          const std::vector<std::string>& kids) {
    Node* n = nodeAt(path);
    if (n == nullptr) return;
    if (nodesGauge_ != nullptr)
        nodesGauge_->set("", byPath_.size());

    std::unordered_map<std::string, Node*> newKids;
    for (const std::string& kid : kids) {
        auto it = n->kids_.find(kid);
        if (it != n->kids_.end()) {
            newKids[kid] = it->second;
            n->kids_.erase(it);
        } else {
            std::string kidDomain = kid + "." + n->domain_;
            auto node = std::make_unique<Node>(this, kidDomain);
            Node* raw = node.get();
            /* A node replaces any same-domain entry (shouldn't happen
             * under distinct paths, but guard like zk.js:205-207). */
            byDomain_[raw->domain_] = raw;
            byPath_[raw->path_] = std::move(node);
            newKids[kid] = raw;
            bind(raw);
        }
    }
    /* whatever remains in n->kids_ was removed */
    for (auto& [label, old] : n->kids_) unbind(old);
    n->kids_ = std::move(newKids);
    n->rec_.clearWireCaches();  /* membership feeds service responses */
}

void ZkMirror::unbind(Node* n) {
    for (auto& [label, kid] : n->kids_) unbind(kid);
    n->kids_.clear();
    if (!n->ip_.empty()) {
        auto it = rev_.find(n->ip_);
        if (it != rev_.end() && it->second == n) rev_.erase(it);
    }
    auto dit = byDomain_.find(n->domain_);
    if (dit != byDomain_.end() && dit->second == n) byDomain_.erase(dit);
    byPath_.erase(n->path_);  // frees n
}

void ZkMirror::onData(const std::string& path, const std::string& data) {
    Node* n = nodeAt(path);
    if (n == nullptr) return;
    auto parsed = Json::parse(data);
    if (!parsed) {
        log_.warn({{"path", Json(path)}},
                  "ignoring node: failed to parse data");
        return;
    }
    CompiledRecord rec = compileRecord(*parsed);
    if (!rec.hasData) {
        /* string/number payloads: 'Parsed JSON data is not an object' —
         * previous data retained (zk.js:149-154). */
        log_.warn({{"path", Json(path)}},
                  "ignoring node: parsed JSON data is not an object");
        return;
    }
    /* reverse map maintenance (zk.js:172-189) */
    std::string newIp;
    if (recTypeIsHostLike(rec.type) && rec.valid) newIp = rec.address;
    if (n->ip_ != newIp) {
        if (!n->ip_.empty()) {
            auto it = rev_.find(n->ip_);
            if (it != rev_.end() && it->second == n) rev_.erase(it);
        }
        if (!newIp.empty()) rev_[newIp] = n;
        n->ip_ = newIp;
    }
    n->rec_ = std::move(rec);
    /* member data feeds the parent's cached service responses */
    size_t slash = path.rfind('/');
    if (slash != std::string::npos && slash > 0) {
        Node* parent = nodeAt(path.substr(0, slash));
        if (parent != nullptr) parent->rec_.clearWireCaches();
    }
}

}  // namespace bamd
