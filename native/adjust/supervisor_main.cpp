/*
 * binder-supervisor: instance manager (the Linux-native replacement for
 * the illumos SMF restarter role in the reference deployment; SURVEY.md
 * §2 rows 6, 15). The reference relies on svc.startd to run N binder
 * SMF instances created by smf_adjust; here a small supervisor daemon
 * owns that role so the stack is self-contained on Linux.
 *
 * State directory layout (shared contract with binder-adjust):
 *   <dir>/instances/<name>.json   {"port": P, "socket_path": "...",
 *                                  "config_file": "...", "enabled": true}
 *   <dir>/status.json             written by the supervisor: per-instance
 *                                 {pid, state, restarts, since}
 *   <dir>/metric_ports            comma-separated metric ports of online
 *                                 instances (metric-ports-updater
 *                                 equivalent, smf/methods/
 *                                 metric-ports-updater.sh:34-80)
 *
 * Behavior: scans instances/ once a second (and on SIGHUP); starts
 * missing processes (`binderd -p P -b SOCK [-f CFG] [-S STORE]`),
 * SIGTERMs processes whose file disappeared or changed (then restarts
 * with the new config), restarts crashed instances with 1s..30s
 * exponential backoff, and reaps children via SIGCHLD.
 */
#include <fcntl.h>
#include <signal.h>
#include <sys/epoll.h>
#include <sys/signalfd.h>
#include <sys/stat.h>
#include <sys/types.h>
#include <sys/wait.h>
#include <unistd.h>

#include <dirent.h>

#include <cstring>
#include <fstream>
#include <map>
#include <sstream>
#include <string>
#include <vector>

#include "../common/json.hpp"
#include "../common/log.hpp"
#include "../common/loop.hpp"

using namespace bamd;

namespace {

struct Instance {
    std::string name;
    Json cfg;
    std::string cfgDump;
    pid_t pid = -1;
    int restarts = 0;
    int64_t backoffMs = 1000;
    int64_t nextStartAt = 0;
    int64_t since = 0;
    bool stopping = false;     // SIGTERM sent, waiting for exit
    bool removed = false;      // file gone: stop and forget
    std::string state = "offline";
};

class Supervisor {
  public:
    Supervisor(EventLoop* loop, Logger log, std::string dir,
               std::string binderd)
        : loop_(loop), log_(std::move(log)), dir_(std::move(dir)),
          binderd_(std::move(binderd)) {}

    bool start();

  private:
    void scan();
    void converge();
    void startInstance(Instance& in);
    void stopInstance(Instance& in);
    void onChild();
    void writeStatus();

    EventLoop* loop_;
    Logger log_;
    std::string dir_;
    std::string binderd_;
    std::map<std::string, Instance> instances_;
    bool shuttingDown_ = false;
};

bool Supervisor::start() {
    mkdir(dir_.c_str(), 0755);
    mkdir((dir_ + "/instances").c_str(), 0755);
    mkdir((dir_ + "/log").c_str(), 0755);

    sigset_t mask;
    sigemptyset(&mask);
    sigaddset(&mask, SIGCHLD);
    sigaddset(&mask, SIGHUP);
    sigaddset(&mask, SIGTERM);
    sigaddset(&mask, SIGINT);
    sigprocmask(SIG_BLOCK, &mask, nullptr);
    int sfd = signalfd(-1, &mask, SFD_NONBLOCK | SFD_CLOEXEC);
    loop_->addFd(sfd, EPOLLIN, [this, sfd](uint32_t) {
        struct signalfd_siginfo si;
        while (read(sfd, &si, sizeof(si)) == sizeof(si)) {
            if (si.ssi_signo == SIGCHLD) {
                onChild();
            } else if (si.ssi_signo == SIGHUP) {
                scan();
                converge();
            } else {
                log_.info("shutting down; stopping all instances");
                shuttingDown_ = true;  /* converge() must not respawn */
                for (auto& [name, in] : instances_)
                    if (in.pid > 0) {
                        in.stopping = true;
                        kill(in.pid, SIGTERM);
                    }
                /* wait for children so our exit means theirs (ports
                 * free for an immediate successor) */
                int64_t deadline = monotonicMillis() + 5000;
                bool anyLeft = true;
                while (anyLeft && monotonicMillis() < deadline) {
                    onChild();
                    anyLeft = false;
                    for (auto& [name, in] : instances_)
                        anyLeft = anyLeft || in.pid > 0;
                    if (anyLeft) usleep(50 * 1000);
                }
                for (auto& [name, in] : instances_)
                    if (in.pid > 0) kill(in.pid, SIGKILL);
                loop_->stop();
            }
        }
    });

    auto tick = std::make_shared<std::function<void()>>();
    *tick = [this, tick]() {
        scan();
        converge();
        writeStatus();
        loop_->addTimer(1000, *tick);
    };
    scan();
    converge();
    writeStatus();
    loop_->addTimer(1000, *tick);
    log_.info({{"dir", Json(dir_)}}, "supervisor started");
    return true;
}

void Supervisor::scan() {
    std::map<std::string, std::pair<Json, std::string>> files;
    std::string idir = dir_ + "/instances";
    DIR* d = opendir(idir.c_str());
    if (d != nullptr) {
        struct dirent* ent;
        while ((ent = readdir(d)) != nullptr) {
            std::string name = ent->d_name;
            if (name.size() < 6 ||
                name.substr(name.size() - 5) != ".json")
                continue;
            std::ifstream f(idir + "/" + name);
            std::stringstream ss;
            ss << f.rdbuf();
            auto parsed = Json::parse(ss.str());
            if (!parsed || !parsed->isObject()) continue;
            std::string iname = name.substr(0, name.size() - 5);
            files[iname] = {*parsed, parsed->dump()};
        }
        closedir(d);
    }

    for (auto& [name, data] : files) {
        auto it = instances_.find(name);
        if (it == instances_.end()) {
            Instance in;
            in.name = name;
            in.cfg = data.first;
            in.cfgDump = data.second;
            instances_[name] = std::move(in);
            log_.info({{"instance", Json(name)}}, "instance added");
        } else if (it->second.cfgDump != data.second &&
                   !it->second.removed) {
            /* config change: restart with new config (smf_adjust's
             * refresh-on-diff, smf_adjust.c:338-455) */
            log_.info({{"instance", Json(name)}},
                      "instance config changed; restarting");
            it->second.cfg = data.first;
            it->second.cfgDump = data.second;
            if (it->second.pid > 0) stopInstance(it->second);
        }
    }
    for (auto& [name, in] : instances_) {
        if (files.count(name) == 0 && !in.removed) {
            in.removed = true;
            log_.info({{"instance", Json(name)}}, "instance removed");
            if (in.pid > 0) stopInstance(in);
        }
    }
}

void Supervisor::converge() {
    if (shuttingDown_) return;
    int64_t now = monotonicMillis();
    for (auto it = instances_.begin(); it != instances_.end();) {
        Instance& in = it->second;
        if (in.removed && in.pid <= 0) {
            it = instances_.erase(it);
            continue;
        }
        if (!in.removed && in.pid <= 0 && now >= in.nextStartAt &&
            in.cfg.get("enabled").asBool(true))
            startInstance(in);
        ++it;
    }
}

void Supervisor::startInstance(Instance& in) {
    std::vector<std::string> argv{binderd_};
    if (in.cfg.get("port").isNumber()) {
        argv.push_back("-p");
        argv.push_back(std::to_string(in.cfg.get("port").asInt()));
    }
    if (in.cfg.get("socket_path").isString()) {
        argv.push_back("-b");
        argv.push_back(in.cfg.get("socket_path").asString());
    }
    if (in.cfg.get("config_file").isString()) {
        argv.push_back("-f");
        argv.push_back(in.cfg.get("config_file").asString());
    }
    if (in.cfg.get("store").isString()) {
        argv.push_back("-S");
        argv.push_back(in.cfg.get("store").asString());
    }

    pid_t pid = fork();
    if (pid < 0) {
        log_.error({{"instance", Json(in.name)}}, "fork failed");
        in.nextStartAt = monotonicMillis() + 5000;
        return;
    }
    if (pid == 0) {
        /* child: redirect output to per-instance log */
        std::string logPath = dir_ + "/log/" + in.name + ".log";
        int lfd = open(logPath.c_str(), O_WRONLY | O_CREAT | O_APPEND,
                       0644);
        if (lfd >= 0) {
            dup2(lfd, 1);
            dup2(lfd, 2);
            close(lfd);
        }
        sigset_t none;
        sigemptyset(&none);
        sigprocmask(SIG_SETMASK, &none, nullptr);
        std::vector<char*> cargv;
        for (auto& a : argv) cargv.push_back(const_cast<char*>(a.c_str()));
        cargv.push_back(nullptr);
        execv(cargv[0], cargv.data());
        _exit(127);
    }
    in.pid = pid;
    in.state = "starting";
    in.since = wallMillis();
    log_.info({{"instance", Json(in.name)},
               {"pid", Json((int64_t)pid)}},
              "instance started");
}

void Supervisor::stopInstance(Instance& in) {
    if (in.pid > 0 && !in.stopping) {
        in.stopping = true;
        in.state = "stopping";
        kill(in.pid, SIGTERM);
    }
}

void Supervisor::onChild() {
    while (true) {
        int status;
        pid_t pid = waitpid(-1, &status, WNOHANG);
        if (pid <= 0) return;
        for (auto& [name, in] : instances_) {
            if (in.pid != pid) continue;
            in.pid = -1;
            bool wasStopping = in.stopping;
            in.stopping = false;
            in.state = "offline";
            if (wasStopping || in.removed) {
                in.backoffMs = 1000;
                in.nextStartAt = 0;
                log_.info({{"instance", Json(name)}}, "instance stopped");
            } else {
                in.restarts++;
                in.nextStartAt = monotonicMillis() + in.backoffMs;
                log_.warn({{"instance", Json(name)},
                           {"status", Json((int64_t)status)},
                           {"backoff_ms", Json(in.backoffMs)}},
                          "instance died; will restart");
                in.backoffMs = in.backoffMs >= 30000 ? 30000
                                                     : in.backoffMs * 2;
            }
            break;
        }
        converge();
        writeStatus();
    }
}

void Supervisor::writeStatus() {
    Json out = Json::object();
    JsonObject insts;
    std::string metricPorts;
    for (auto& [name, in] : instances_) {
        /* online only once the instance is actually serving: binderd
         * creates its balancer socket AFTER binding UDP+TCP, so file
         * presence == ready (the SMF-online equivalent; smf_adjust's
         * -w waits for real online, smf_adjust.c:457-544). Instances
         * without a socket_path count as online once spawned. */
        if (in.pid > 0 && in.state == "starting") {
            const Json& sp = in.cfg.get("socket_path");
            if (!sp.isString() ||
                access(sp.asString().c_str(), F_OK) == 0)
                in.state = "online";
        }
        Json j = Json::object();
        j.set("pid", Json((int64_t)in.pid));
        j.set("state", Json(in.pid > 0 ? in.state : "offline"));
        j.set("restarts", Json((int64_t)in.restarts));
        j.set("since", Json(in.since));
        if (in.cfg.get("port").isNumber())
            j.set("port", in.cfg.get("port"));
        insts[name] = std::move(j);
        if (in.pid > 0 && in.cfg.get("port").isNumber()) {
            /* metric port = service port + 1000 (main.js:144-152;
             * metric-ports-updater.sh:48-60) */
            if (!metricPorts.empty()) metricPorts += ",";
            metricPorts +=
                std::to_string(in.cfg.get("port").asInt() + 1000);
        }
    }
    out.set("instances", Json(std::move(insts)));
    std::string tmp = dir_ + "/status.json.tmp";
    {
        std::ofstream f(tmp);
        f << out.dump() << "\n";
    }
    rename(tmp.c_str(), (dir_ + "/status.json").c_str());
    {
        std::ofstream f(dir_ + "/metric_ports.tmp");
        f << metricPorts << "\n";
    }
    rename((dir_ + "/metric_ports.tmp").c_str(),
           (dir_ + "/metric_ports").c_str());
}

}  // namespace

int main(int argc, char** argv) {
    /* a peer closing mid-write must be an EPIPE errno, not process
     * death */
    signal(SIGPIPE, SIG_IGN);
    const char* lvl = getenv("LOG_LEVEL");
    Logger log("binder-supervisor",
               logLevelFromName(lvl ? lvl : "info", LogLevel::Info));
    std::string dir = "/var/run/binder";
    std::string binderd;
    int c;
    while ((c = getopt(argc, argv, "hd:x:")) != -1) {
        switch (c) {
        case 'd': dir = optarg; break;
        case 'x': binderd = optarg; break;
        case 'h':
        default:
            fprintf(stderr,
                    "usage: binder-supervisor [-d state-dir] "
                    "[-x binderd-path]\n");
            return c == 'h' ? 0 : 1;
        }
    }
    if (binderd.empty()) {
        /* default: binderd next to this executable */
        char self[4096];
        ssize_t n = readlink("/proc/self/exe", self, sizeof(self) - 1);
        if (n > 0) {
            self[n] = '\0';
            std::string s(self);
            size_t slash = s.rfind('/');
            binderd = s.substr(0, slash + 1) + "binderd";
        } else {
            binderd = "binderd";
        }
    }
    EventLoop loop;
    Supervisor sup(&loop, log, dir, binderd);
    if (!sup.start()) return 1;
    loop.run();
    return 0;
}
