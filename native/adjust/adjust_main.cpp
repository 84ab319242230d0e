/*
 * binder-adjust: idempotently converge the set of binder server
 * instances to a plan (src/smf_adjust.c equivalent, SURVEY.md §2 row 6;
 * the illumos original drives libscf/SMF — this drives the
 * binder-supervisor state directory, same convergence semantics):
 *
 *   -b <base>   instance base name            (smf_adjust -b)
 *   -B <port>   base port                     (smf_adjust -B)
 *   -i <count>  instance count, 0..32        (smf_adjust -i, capped 32
 *               like boot/setup.sh:15)
 *   -d <dir>    supervisor state directory
 *   -s <dir>    balancer socket directory (default <dir>/sockets)
 *   -f <file>   shared binderd config file passed to each instance
 *   -S <store>  store mode passed to each instance (e.g. zk)
 *   -w <secs>   wait until all planned instances are online
 *               (smf_adjust's enable-and-wait, smf_adjust.c:457-544)
 *
 * Convergence: plan = {<base>-<port> for port in B..B+i-1}. Unplanned
 * instance files are removed (supervisor drains them); missing ones are
 * created; existing ones are rewritten ONLY if their JSON differs
 * (deep equality via the canonical dump — the nvlist_equal analog,
 * src/nvlist_equal.c). Exit 0 on success.
 */
#include <signal.h>
#include <dirent.h>
#include <sys/stat.h>
#include <unistd.h>

#include <cstring>
#include <fstream>
#include <map>
#include <set>
#include <sstream>
#include <string>

#include "../common/json.hpp"
#include "../common/log.hpp"

using namespace bamd;

static Json readJson(const std::string& path) {
    std::ifstream f(path);
    if (!f) return Json();
    std::stringstream ss;
    ss << f.rdbuf();
    auto p = Json::parse(ss.str());
    return p ? *p : Json();
}

int main(int argc, char** argv) {
    /* a peer closing mid-write must be an EPIPE errno, not process
     * death */
    signal(SIGPIPE, SIG_IGN);
    const char* lvl = getenv("LOG_LEVEL");
    Logger log("binder-adjust",
               logLevelFromName(lvl ? lvl : "info", LogLevel::Info));
    std::string base = "binder";
    int basePort = 5301;        /* boot/setup.sh:135 */
    int count = -1;
    std::string dir = "/var/run/binder";
    std::string sockDir;
    std::string cfgFile;
    std::string store;
    int waitSecs = 0;
    int c;
    while ((c = getopt(argc, argv, "hb:B:i:d:s:f:S:w:")) != -1) {
        switch (c) {
        case 'b': base = optarg; break;
        case 'B': basePort = atoi(optarg); break;
        case 'i': count = atoi(optarg); break;
        case 'd': dir = optarg; break;
        case 's': sockDir = optarg; break;
        case 'f': cfgFile = optarg; break;
        case 'S': store = optarg; break;
        case 'w': waitSecs = atoi(optarg); break;
        case 'h':
        default:
            fprintf(stderr,
                    "usage: binder-adjust -i count [-b base] [-B port] "
                    "[-d state-dir] [-s socket-dir] [-f config] "
                    "[-S store] [-w secs]\n");
            return c == 'h' ? 0 : 1;
        }
    }
    if (count < 0 || count > 32) {
        /* same bounds as smf_adjust.c:904-909 / boot/setup.sh:15 */
        fprintf(stderr, "binder-adjust: -i must be 0..32\n");
        return 1;
    }
    if (sockDir.empty()) sockDir = dir + "/sockets";
    mkdir(dir.c_str(), 0755);
    mkdir((dir + "/instances").c_str(), 0755);
    mkdir(sockDir.c_str(), 0770);

    /* plan phase (smf_adjust.c:960-969) */
    std::map<std::string, Json> plan;
    for (int i = 0; i < count; ++i) {
        int port = basePort + i;
        std::string name = base + "-" + std::to_string(port);
        Json cfg = Json::object();
        cfg.set("instance", Json((int64_t)port));
        cfg.set("port", Json((int64_t)port));
        cfg.set("socket_path", Json(sockDir + "/" + std::to_string(port)));
        if (!cfgFile.empty()) cfg.set("config_file", Json(cfgFile));
        if (!store.empty()) cfg.set("store", Json(store));
        cfg.set("enabled", Json(true));
        plan[name] = std::move(cfg);
    }

    /* walk existing instance files (smf_adjust.c:974-1019) */
    std::string idir = dir + "/instances";
    std::set<std::string> existing;
    DIR* d = opendir(idir.c_str());
    if (d != nullptr) {
        struct dirent* ent;
        while ((ent = readdir(d)) != nullptr) {
            std::string name = ent->d_name;
            if (name.size() < 6 ||
                name.substr(name.size() - 5) != ".json")
                continue;
            existing.insert(name.substr(0, name.size() - 5));
        }
        closedir(d);
    }

    int created = 0, removed = 0, updated = 0, kept = 0;
    /* remove pass (smf_adjust.c:1024-1038) — only instances matching
     * our base name, mirroring smf_adjust's per-service scope */
    for (const auto& name : existing) {
        if (plan.count(name) > 0) continue;
        if (name.rfind(base + "-", 0) != 0) continue;
        unlink((idir + "/" + name + ".json").c_str());
        removed++;
        log.info({{"instance", Json(name)}}, "removed unplanned instance");
    }
    /* create/configure pass (smf_adjust.c:1040-1099) */
    for (auto& [name, cfg] : plan) {
        std::string path = idir + "/" + name + ".json";
        Json current = readJson(path);
        if (current.isObject() && current.dump() == cfg.dump()) {
            kept++;  // no-op update skipped (nvlist diff semantics)
            continue;
        }
        std::string tmp = path + ".tmp";
        {
            std::ofstream f(tmp);
            f << cfg.dump() << "\n";
        }
        rename(tmp.c_str(), path.c_str());
        if (existing.count(name) > 0)
            updated++;
        else
            created++;
    }
    log.info({{"created", Json((int64_t)created)},
              {"removed", Json((int64_t)removed)},
              {"updated", Json((int64_t)updated)},
              {"unchanged", Json((int64_t)kept)}},
             "instance convergence complete");

    if (waitSecs > 0) {
        /* wait-for-online (smf_adjust.c:457-544: 60s default there) */
        int64_t deadline = monotonicMillis() + (int64_t)waitSecs * 1000;
        while (monotonicMillis() < deadline) {
            Json status = readJson(dir + "/status.json");
            const Json& insts = status.get("instances");
            int online = 0;
            for (auto& [name, cfg] : plan)
                if (insts.get(name).get("state").asString() == "online")
                    online++;
            if (online == (int)plan.size()) {
                log.info("all instances online");
                return 0;
            }
            usleep(100 * 1000);  /* smf_adjust polls at 100ms too */
        }
        log.error("timed out waiting for instances to come online");
        return 2;
    }
    return 0;
}
