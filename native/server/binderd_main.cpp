/*
 * binderd: the binder-amd DNS server process (main.js equivalent).
 *
 * CLI parity with /root/reference/main.js:51-108:
 *   -a <ms>    cache expiry  (vestigial in the reference too; accepted)
 *   -b <path>  balancer UNIX socket path
 *   -f <file>  JSON config file (default ./etc/config.json)
 *   -p <port>  DNS service port (default 53)
 *   -s <n>     cache size (vestigial; accepted)
 *   -v         increase verbosity (repeatable)
 *   -h         usage
 * plus binder-amd extensions:
 *   -S <mode>  store mode: "zk" (default) or "file:<path>" (static JSON
 *              tree, used by tests/bench config 1)
 * Env: ZK_HOST (lib/zk.js:34), ZK_PORT (test/helper.js:54), LOG_LEVEL.
 * Option precedence: defaults < config file < CLI (main.js:105).
 * Metrics HTTP on port+1000 (main.js:144-152).
 */
#include <fcntl.h>
#include <signal.h>
#include <sys/epoll.h>
#include <sys/signalfd.h>
#include <unistd.h>

#include <cstdio>
#include <cstring>
#include <fstream>
#include <sstream>

#include "../common/json.hpp"
#include "../common/log.hpp"
#include "../common/loop.hpp"
#include "../engine/engine.hpp"
#include "../engine/store.hpp"
#include "../zk/mirror.hpp"
#include "metrics.hpp"
#include "recursion.hpp"
#include "server.hpp"

using namespace bamd;

static const char* kVersion = "binder-amd 0.2.0";

static void usage(const char* name) {
    fprintf(stderr,
            "usage: %s [-v] [-V] [-a cacheExpiry] [-s cacheSize] "
            "[-p port] [-f file] [-b balancerSocket] "
            "[-S zk|file:<path>]\n",
            name);
}

static bool loadFileStore(StubStore& store, const std::string& path,
                          Logger& log) {
    std::ifstream f(path);
    if (!f) {
        log.error({{"path", Json(path)}}, "cannot open store file");
        return false;
    }
    std::stringstream ss;
    ss << f.rdbuf();
    auto parsed = Json::parse(ss.str());
    if (!parsed || !parsed->isObject()) {
        log.error({{"path", Json(path)}}, "store file is not a JSON object");
        return false;
    }
    for (const auto& [domain, rec] : parsed->fields())
        store.put(domain, rec);
    return true;
}

int main(int argc, char** argv) {
    /* a peer closing mid-write must be an EPIPE errno, not process
     * death */
    signal(SIGPIPE, SIG_IGN);
    const char* lvl = getenv("LOG_LEVEL");
    LogLevel level = logLevelFromName(lvl ? lvl : "info", LogLevel::Info);
    Logger log("binder", level);

    /* defaults (main.js:34-38) */
    Json opts = Json::object();
    opts.set("expiry", Json((int64_t)60000));
    opts.set("size", Json((int64_t)10000));
    opts.set("port", Json((int64_t)53));

    std::string configFile = "./etc/config.json";
    Json cli = Json::object();
    int c;
    int verbosity = 0;
    while ((c = getopt(argc, argv, "hvVa:b:d:s:p:f:S:")) != -1) {
        switch (c) {
        case 'V': printf("%s\n", kVersion); return 0;
        case 'd': verbosity = atoi(optarg); break;  /* README.md:46-50
            documents `-d 2` for debug spew; main.js actually
            implements -v (flag drift noted in SURVEY.md §5.5) —
            accept both */
        case 'a': cli.set("expiry", Json((int64_t)atoi(optarg))); break;
        case 'b': cli.set("balancerSocket", Json(std::string(optarg))); break;
        case 'f': configFile = optarg; break;
        case 'p': cli.set("port", Json((int64_t)atoi(optarg))); break;
        case 's': cli.set("size", Json((int64_t)atoi(optarg))); break;
        case 'S': cli.set("store", Json(std::string(optarg))); break;
        case 'v': verbosity++; break;
        case 'h': usage(argv[0]); return 0;
        default: return 1;
        }
    }
    if (verbosity >= 2)
        log.setLevel(LogLevel::Trace);
    else if (verbosity == 1)
        log.setLevel(LogLevel::Debug);

    /* config file merge: defaults < file < CLI (main.js:96-107). The
     * reference exits fatally when the file is unreadable; we only do so
     * when -f was given explicitly (the default ./etc/config.json may not
     * exist in dev). */
    {
        std::ifstream f(configFile);
        if (f) {
            std::stringstream ss;
            ss << f.rdbuf();
            auto parsed = Json::parse(ss.str());
            if (!parsed) {
                log.log(LogLevel::Fatal, "config file is not valid JSON",
                        {{"file", Json(configFile)}});
                return 1;
            }
            for (const auto& [k, v] : parsed->fields()) opts.set(k, v);
        } else if (std::string(configFile) != "./etc/config.json") {
            log.log(LogLevel::Fatal, "cannot read config file",
                    {{"file", Json(configFile)}});
            return 1;
        }
    }
    for (const auto& [k, v] : cli.fields()) opts.set(k, v);

    log.info(opts.fields(), "starting with options");

    uint16_t port = (uint16_t)opts.get("port").asInt(53);
    std::string dnsDomain = opts.get("dnsDomain").asString();
    std::string dcName = opts.get("datacenterName").asString();

    EventLoop loop;

    /* metrics manager on port+1000 (main.js:134-152) */
    Collector collector;
    std::string staticLabels = renderLabels({
        {"datacenter", dcName},
        {"instance", opts.get("instance_uuid").asString()},
        {"server", opts.get("server_uuid").asString()},
        {"service", opts.get("service_name").asString()},
        {"port", std::to_string(port)},
    });
    MetricsHttpServer metricsSrv(&loop, &collector, staticLabels);
    uint16_t metricsPort =
        (uint16_t)opts.get("metricsPort").asInt(port + 1000);
    if (!metricsSrv.listen("0.0.0.0", metricsPort))
        log.warn({{"port", Json((int)metricsPort)}},
                 "could not bind metrics port");

    /* store */
    std::string storeMode = opts.get("store").asString();
    if (storeMode.empty()) storeMode = "zk";
    StubStore fileStore;
    std::unique_ptr<ZkMirror> mirror;
    Store* store = nullptr;
    if (storeMode.rfind("file:", 0) == 0) {
        if (!loadFileStore(fileStore, storeMode.substr(5), log)) return 1;
        store = &fileStore;
    } else if (storeMode == "zk") {
        if (dnsDomain.empty()) {
            /* the reference's ZKCache asserts options.domain
             * (lib/zk.js:23); an empty domain would mirror the whole
             * ZK root here */
            log.fatal("dnsDomain is required with the zk store");
            return 1;
        }
        const char* zh = getenv("ZK_HOST");
        const char* zp = getenv("ZK_PORT");
        std::string zkHost = zh && *zh ? zh : "127.0.0.1";
        if (!zh && opts.get("zookeeper").isObject() &&
            opts.get("zookeeper").get("host").isString())
            zkHost = opts.get("zookeeper").get("host").asString();
        uint16_t zkPort = (uint16_t)(zp && *zp ? atoi(zp) : 2181);
        mirror = std::make_unique<ZkMirror>(
            &loop, log, ZkMirrorOptions{zkHost, zkPort, dnsDomain, 30000},
            &collector);
        mirror->start();
        store = mirror.get();
    } else {
        log.log(LogLevel::Fatal, "unknown store mode",
                {{"store", Json(storeMode)}});
        return 1;
    }

    EngineConfig ecfg;
    ecfg.dnsDomain = dnsDomain;
    ecfg.datacenterName = dcName;
    ecfg.recursionEnabled = opts.get("recursion").isObject();
    Engine engine(ecfg, store);

    ServerOptions sopts;
    sopts.host = opts.get("host").asString();
    sopts.port = port;
    sopts.balancerSocket = opts.get("balancerSocket").asString();

    DnsServer server(&loop, log, sopts, &engine, &collector);
    server.setStore(store);

    /* recursion (lib/recursion.js equivalent; config block per
     * sapi_manifests/binder/template) */
    std::unique_ptr<Recursion> recursion;
    if (opts.get("recursion").isObject()) {
        RecursionOptions ropts;
        const Json& rj = opts.get("recursion");
        ropts.regionName = rj.get("regionName").asString();
        ropts.datacenterName =
            rj.get("datacenterName").isString()
                ? rj.get("datacenterName").asString() : dcName;
        ropts.dnsDomain = rj.get("dnsDomain").isString()
                              ? rj.get("dnsDomain").asString() : dnsDomain;
        if (rj.get("upstreamPort").isNumber())
            ropts.upstreamPort = (uint16_t)rj.get("upstreamPort").asInt();
        ropts.config = rj;
        recursion = std::make_unique<Recursion>(&loop, log, ropts, store);
        recursion->init();
        server.setRecursion(recursion.get());
    }

    if (!server.start()) {
        log.fatal("error initing binder");
        return 1;
    }
    log.info("done with binder init");

    /* SIGTERM: unlink balancer socket then exit (main.js:181-193);
     * SIGINT for interactive use. */
    sigset_t mask;
    sigemptyset(&mask);
    sigaddset(&mask, SIGTERM);
    sigaddset(&mask, SIGINT);
    sigprocmask(SIG_BLOCK, &mask, nullptr);
    int sfd = signalfd(-1, &mask, SFD_NONBLOCK | SFD_CLOEXEC);
    loop.addFd(sfd, EPOLLIN, [&](uint32_t) {
        struct signalfd_siginfo si;
        ssize_t rv = read(sfd, &si, sizeof(si));
        (void)rv;
        if (!sopts.balancerSocket.empty()) {
            log.info({{"path", Json(sopts.balancerSocket)}},
                     "caught SIGTERM; unlinking socket");
            unlink(sopts.balancerSocket.c_str());
        }
        loop.stop();
    });

    loop.run();
    server.stop();
    return 0;
}
