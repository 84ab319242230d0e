/*
 * binder-amd: watch-driven ZooKeeper tree mirror (lib/zk.js equivalent).
 *
 * Mirrors the registration tree under the dnsDomain znode into memory:
 * one node per znode, forward map domain->node, reverse map ip->node.
 * Queries NEVER touch ZooKeeper — the load-bearing design fact of the
 * reference. On every (re)established session the whole
 * tree is re-bound: every node re-issues getData+getChildren with fresh
 * watches, which both re-arms and resyncs (lib/zk.js:45-47, 209-223).
 *
 * Improvement over the reference (documented deviation): unbind removes
 * stale reverse-map entries; the reference leaks them until another node
 * claims the IP (lib/zk.js:195-208 never touches ca_revLookup).
 */
#pragma once

#include <memory>
#include <string>
#include <unordered_map>

#include "../common/log.hpp"
#include "../common/loop.hpp"
#include "../engine/store.hpp"
#include "../server/metrics.hpp"
#include "client.hpp"

namespace bamd {

struct ZkMirrorOptions {
    std::string host = "127.0.0.1";
    uint16_t port = 2181;
    std::string domain;          // dnsDomain, e.g. "foo.com"
    int sessionTimeoutMs = 30000;
};

class ZkMirror : public Store {
  public:
    class Node : public StoreNode {
      public:
        Node(ZkMirror* m, std::string domain);
        const CompiledRecord& rec() const override { return rec_; }
        const std::string& domain() const override { return domain_; }
        const std::string& name() const override { return name_; }
        std::vector<const StoreNode*> children() const override;

        ZkMirror* mirror_;
        std::string domain_;
        std::string name_;
        std::string path_;
        CompiledRecord rec_;
        std::string ip_;  // current reverse-map registration
        std::unordered_map<std::string, Node*> kids_;  // label -> node
    };

    ZkMirror(EventLoop* loop, Logger log, ZkMirrorOptions opts,
             Collector* collector);
    ~ZkMirror() override;

    void start();
    void stop();

    /* Store */
    const StoreNode* lookup(const std::string& domain) const override;
    const StoreNode* lookupView(std::string_view domain) const override;
    const StoreNode* reverseLookup(const std::string& ip) const override;
    bool ready() const override;

    zk::ZkClient& client() { return *client_; }
    size_t nodeCount() const { return byPath_.size(); }

  private:
    friend class Node;
    void rebuild();                    // 'session' handler
    void bind(Node* n);                // fresh getData+getChildren+watches
    void onWatch(int32_t type, const std::string& path);
    void onChildren(const std::string& path,
                    const std::vector<std::string>& kids);
    void onData(const std::string& path, const std::string& data);
    void unbind(Node* n);              // recursive removal
    Node* nodeAt(const std::string& path);

    EventLoop* loop_;
    Logger log_;
    ZkMirrorOptions opts_;
    std::unique_ptr<zk::ZkClient> client_;

    std::unordered_map<std::string, std::unique_ptr<Node>> byPath_;
    std::unordered_map<std::string, Node*, SvHash, SvEq> byDomain_;
    std::unordered_map<std::string, Node*> rev_;
    Node* root_ = nullptr;

    Counter* nodesGauge_ = nullptr;
    Counter* sessionsCounter_ = nullptr;
};

}  // namespace bamd
