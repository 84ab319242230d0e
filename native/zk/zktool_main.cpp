/*
 * zktool: minimal native ZooKeeper CLI over binder-amd's own client
 * (zkCli-style operator tool; the reference leans on zkCli.sh from the
 * bundled ZooKeeper distribution for the same purpose).
 *
 * usage: zktool [-s host:port] <op> <path> [data]
 *   ops: get | ls | create | set | rm | rmr | stat
 * Exit: 0 ok, 1 usage/connect error, 2 op error (e.g. no such node).
 */
#include <unistd.h>

#include <cstdio>
#include <cstring>
#include <functional>
#include <string>
#include <vector>

#include "../common/log.hpp"
#include "../common/loop.hpp"
#include "client.hpp"

using namespace bamd;
using namespace bamd::zk;

namespace {

struct Ctx {
    EventLoop loop;
    std::unique_ptr<ZkClient> client;
    int rc = 0;
    bool done = false;

    void finish(int code) {
        rc = code;
        done = true;
        loop.stop();
    }

    bool run(int64_t timeoutMs) {
        return loop.runUntil([this]() { return done; }, timeoutMs);
    }
};

}  // namespace

int main(int argc, char** argv) {
    std::string server = "127.0.0.1:2181";
    const char* zh = getenv("ZK_HOST");
    const char* zp = getenv("ZK_PORT");
    if (zh && *zh)
        server = std::string(zh) + ":" + (zp && *zp ? zp : "2181");
    int c;
    while ((c = getopt(argc, argv, "hs:")) != -1) {
        switch (c) {
        case 's': server = optarg; break;
        case 'h':
        default:
            fprintf(stderr,
                    "usage: zktool [-s host:port] "
                    "get|ls|create|set|rm|rmr|stat <path> [data]\n");
            return c == 'h' ? 0 : 1;
        }
    }
    if (argc - optind < 2) {
        fprintf(stderr, "zktool: need <op> <path>\n");
        return 1;
    }
    std::string op = argv[optind];
    std::string path = argv[optind + 1];
    std::string data =
        argc - optind >= 3 ? argv[optind + 2] : std::string("null");

    size_t colon = server.rfind(':');
    ZkConfig cfg;
    cfg.host = colon == std::string::npos ? server
                                          : server.substr(0, colon);
    cfg.port = colon == std::string::npos
                   ? 2181
                   : (uint16_t)atoi(server.c_str() + colon + 1);
    cfg.sessionTimeoutMs = 10000;

    Ctx ctx;
    Logger log("zktool", LogLevel::Warn, 2);
    ctx.client = std::make_unique<ZkClient>(&ctx.loop, log, cfg);

    ctx.client->onSession([&ctx, &op, &path, &data]() {
        ZkClient& zk = *ctx.client;
        if (op == "get") {
            zk.getData(path, false,
                       [&ctx](int32_t rc, const std::string& d,
                              const Stat&) {
                           if (rc != ZOK) {
                               fprintf(stderr, "err %d\n", rc);
                               ctx.finish(2);
                               return;
                           }
                           fwrite(d.data(), 1, d.size(), stdout);
                           fputc('\n', stdout);
                           ctx.finish(0);
                       });
        } else if (op == "ls") {
            zk.getChildren(
                path, false,
                [&ctx](int32_t rc, const std::vector<std::string>& kids) {
                    if (rc != ZOK) {
                        fprintf(stderr, "err %d\n", rc);
                        ctx.finish(2);
                        return;
                    }
                    for (const auto& k : kids) printf("%s\n", k.c_str());
                    ctx.finish(0);
                });
        } else if (op == "stat") {
            zk.exists(path, false, [&ctx](int32_t rc, const Stat& st) {
                if (rc != ZOK) {
                    fprintf(stderr, "err %d\n", rc);
                    ctx.finish(2);
                    return;
                }
                printf("version=%d cversion=%d dataLength=%d "
                       "numChildren=%d\n",
                       st.version, st.cversion, st.dataLength,
                       st.numChildren);
                ctx.finish(0);
            });
        } else if (op == "create") {
            zk.create(path, data, 0,
                      [&ctx](int32_t rc, const std::string& created) {
                          if (rc != ZOK) {
                              fprintf(stderr, "err %d\n", rc);
                              ctx.finish(2);
                              return;
                          }
                          printf("%s\n", created.c_str());
                          ctx.finish(0);
                      });
        } else if (op == "set") {
            zk.setData(path, data, -1, [&ctx](int32_t rc) {
                if (rc != ZOK) {
                    fprintf(stderr, "err %d\n", rc);
                    ctx.finish(2);
                    return;
                }
                ctx.finish(0);
            });
        } else if (op == "rm") {
            zk.del(path, -1, [&ctx](int32_t rc) {
                if (rc != ZOK) {
                    fprintf(stderr, "err %d\n", rc);
                    ctx.finish(2);
                    return;
                }
                ctx.finish(0);
            });
        } else if (op == "rmr") {
            auto walk = std::make_shared<
                std::function<void(std::string)>>();
            auto target = std::make_shared<std::string>(path);
            *walk = [&ctx, walk, target](std::string p) {
                ctx.client->getChildren(
                    p, false,
                    [&ctx, walk, target, p](
                        int32_t rc,
                        const std::vector<std::string>& kids) {
                        if (rc == ZNONODE) {
                            if (p == *target) ctx.finish(0);
                            else (*walk)(*target);
                            return;
                        }
                        if (rc != ZOK) {
                            fprintf(stderr, "err %d\n", rc);
                            ctx.finish(2);
                            return;
                        }
                        if (!kids.empty()) {
                            (*walk)(p == "/" ? "/" + kids[0]
                                             : p + "/" + kids[0]);
                            return;
                        }
                        ctx.client->del(
                            p, -1, [&ctx, walk, target, p](int32_t rc2) {
                                if (rc2 != ZOK && rc2 != ZNONODE) {
                                    fprintf(stderr, "err %d\n", rc2);
                                    ctx.finish(2);
                                    return;
                                }
                                if (p == *target)
                                    ctx.finish(0);
                                else
                                    (*walk)(*target);
                            });
                    });
            };
            (*walk)(path);
        } else {
            fprintf(stderr, "zktool: unknown op %s\n", op.c_str());
            ctx.finish(1);
        }
    });
    ctx.client->start();

    if (!ctx.run(15000)) {
        fprintf(stderr, "zktool: timeout/connect failure\n");
        return 1;
    }
    ctx.client->close();
    return ctx.rc;
}
