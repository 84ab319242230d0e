/*
 * dnsblast: UDP DNS load generator for the benchmark suite
 * (BASELINE.md's qps + p50/p99 measurement).
 *
 * Each worker thread keeps a sliding window of in-flight queries
 * spread over K sockets (-P), cycling through a name list loaded from
 * a file. Query wire images are prebuilt; per-send we patch only the
 * DNS id. Latencies land in log-spaced microsecond buckets for
 * p50/p99 extraction. Output: ONE JSON line per run on stdout.
 *
 * Design notes (each measured on the quota-capped bench boxes;
 * profiles/SCALING.md round 2):
 *  - multiple sockets per thread (-P): the balancer's SO_REUSEPORT
 *    shards hash FLOWS; the flow count must comfortably exceed the
 *    shard count or shard load is balls-in-bins uneven;
 *  - every burst is one UDP_SEGMENT super-packet (equal segment sizes
 *    via per-burst wire size classes, class picked by a weighted die
 *    so the aggregate mix is unchanged) - one kernel loopback
 *    traversal per burst instead of per query; -g disables;
 *  - -r <qps> paces a fixed offered rate (32-packet token quanta,
 *    sub-ms sleeps): the "qps at SLO" protocol, far less noisy than
 *    the chaotic closed-loop saturation equilibrium;
 *  - -D daemon mode keeps the generator (threads, sockets, wires)
 *    alive across runs, reading "RUN <queries> <rate>" lines on
 *    stdin and answering one JSON line each: thread/socket setup is
 *    100-170 ms on the bench boxes and must not be charged to the
 *    timed steps; a start barrier excludes it in one-shot mode too.
 *
 * usage: dnsblast -s <server-ip> -p <port> [-n <queries>] [-c window]
 *        [-t threads] [-P socks/thread] [-r offered-qps]
 *        [-f names-file] [-B bind-ip-base] [-T timeout-ms] [-R] [-g]
 *        [-D]
 *   names-file: lines of "<name> <qtype>"; default a single test name.
 *   -B 127.0.0.x base: thread i binds source ip base+i (gives the
 *      balancer distinct remotes so per-IP affinity spreads load).
 */
#include <signal.h>
#include <arpa/inet.h>
#include <netinet/in.h>
#include <netinet/udp.h>
#include <poll.h>
#include <sys/socket.h>
#include <unistd.h>

#include <algorithm>
#include <array>
#include <atomic>
#include <chrono>
#include <cstdio>
#include <cstring>
#include <fstream>
#include <map>
#include <random>
#include <sstream>
#include <string>
#include <thread>
#include <vector>

#include "../common/log.hpp"
#include "../dns/codec.hpp"

using namespace bamd;

namespace {

struct NameEntry {
    std::string name;
    uint16_t qtype;
};

constexpr int kLatBuckets = 512;

struct ThreadResult {
    uint64_t sent = 0;
    uint64_t received = 0;
    uint64_t timeouts = 0;
    uint64_t rcodeNoerror = 0;
    uint64_t rcodeOther = 0;
    uint64_t answers = 0;
    std::vector<uint64_t> latBuckets = std::vector<uint64_t>(kLatBuckets);

    void reset() {
        sent = received = timeouts = 0;
        rcodeNoerror = rcodeOther = answers = 0;
        std::fill(latBuckets.begin(), latBuckets.end(), 0);
    }
};

/* latency mapping: buckets 0..255: 1us each; 256..495: 16us each
 * (to ~4.1ms); 496..510: 10ms steps; 511: overflow */
inline int latBucket2(int64_t us) {
    if (us < 256) return (int)us;
    int64_t v = (us - 256) / 16;
    if (v < 240) return 256 + (int)v;
    int64_t w = (us - (256 + 240 * 16)) / 10000;  // 10ms steps
    if (w < 15) return 496 + (int)w;
    return 511;
}

double bucketMidUs(int b) {
    if (b < 256) return b + 0.5;
    if (b < 496) return 256 + (b - 256) * 16 + 8;
    if (b < 511) return 256 + 240 * 16 + (b - 496) * 10000 + 5000;
    return 200000;
}

int64_t nowUs() {
    struct timespec ts;
    clock_gettime(CLOCK_MONOTONIC, &ts);
    return (int64_t)ts.tv_sec * 1000000 + ts.tv_nsec / 1000;
}

struct Config {
    std::string server = "127.0.0.1";
    uint16_t port = 1053;
    uint64_t queries = 100000;
    int window = 64;
    int threads = 1;
    int socksPerThread = 1;
    double rateQps = 0;  // total offered rate; 0 = closed loop
    std::string namesFile;
    std::string bindBase;
    int timeoutMs = 2000;
    bool rd = false;   // set RD (exercises recursion paths)
    bool gso = true;   // UDP_SEGMENT batching (auto-fallback)
    bool daemon = false;
};

/* wire indices grouped by identical wire size (GSO super-packets
 * need equal-size segments) */
using SizeClasses = std::vector<std::vector<size_t>>;

/* run control shared between the command loop and the workers */
struct RunCtl {
    std::atomic<uint64_t> gen{0};      // bumped per RUN
    std::atomic<int> setup{0};         // workers finished setup
    std::atomic<int> running{0};       // workers still in a run
    std::atomic<bool> shutdown{false};
    uint64_t queriesPerThread = 0;
    double rateQps = 0;                // total offered rate
};

void worker(const Config& cfg, int tid,
            const std::vector<std::vector<uint8_t>>& wires,
            const SizeClasses& classes, ThreadResult* out,
            RunCtl* ctl) {
    const int K = cfg.socksPerThread < 1 ? 1 : cfg.socksPerThread;
    std::vector<int> fds;
    for (int k = 0; k < K; ++k) {
        int fd = socket(AF_INET, SOCK_DGRAM, 0);
        if (fd < 0) break;
        int sz = 4 << 20;
        setsockopt(fd, SOL_SOCKET, SO_RCVBUF, &sz, sizeof(sz));
        setsockopt(fd, SOL_SOCKET, SO_SNDBUF, &sz, sizeof(sz));
        /* receive balancer reply runs coalesced (one kernel
         * traversal per run); harmless if unsupported */
        int one = 1;
        setsockopt(fd, SOL_UDP, UDP_GRO, &one, sizeof(one));
        if (!cfg.bindBase.empty()) {
            /* bind distinct loopback source ip per thread: base + tid */
            struct in_addr base;
            inet_pton(AF_INET, cfg.bindBase.c_str(), &base);
            uint32_t ip = ntohl(base.s_addr) + (uint32_t)tid;
            struct sockaddr_in src {};
            src.sin_family = AF_INET;
            src.sin_addr.s_addr = htonl(ip);
            src.sin_port = 0;  // distinct ephemeral port per socket
            bind(fd, (struct sockaddr*)&src, sizeof(src));
        }
        struct sockaddr_in dst {};
        dst.sin_family = AF_INET;
        dst.sin_port = htons(cfg.port);
        inet_pton(AF_INET, cfg.server.c_str(), &dst.sin_addr);
        if (connect(fd, (struct sockaddr*)&dst, sizeof(dst)) != 0) {
            close(fd);
            break;
        }
        fds.push_back(fd);
    }
    if (fds.empty()) {
        ctl->setup.fetch_add(1);
        return;
    }
    const int nSock = (int)fds.size();

    /* qid layout: slot in the low bits (window-sized), per-slot
     * sequence in the rest - W up to 1024 with >=6 stale-check bits */
    int W = cfg.window;
    if (W > 1024) W = 1024;
    int slotBits = 1;
    while ((1 << slotBits) < W) slotBits++;
    const uint16_t slotMask = (uint16_t)((1 << slotBits) - 1);
    const uint16_t seqMask = (uint16_t)(0xFFFF >> slotBits);
    std::vector<int64_t> sentAt(W, 0);       // 0 = slot idle
    std::vector<uint16_t> slotSeq(W, 0);
    std::mt19937 rng(12345 + tid);
    const int64_t timeoutUs = (int64_t)cfg.timeoutMs * 1000;

    /* batched RX (recvmmsg) and TX (sendmmsg/GSO): syscall count, not
     * packet handling, bounds the generator at high QPS */
    constexpr int kRxBatch = 64;
    /* 16 KB per rx message: with UDP_GRO one message may be a
     * coalesced run of up to 48 replies */
    std::vector<std::array<uint8_t, 16384>> rxBufs(kRxBatch);
    std::vector<std::array<char, CMSG_SPACE(sizeof(uint16_t))>>
        rxCtrl(kRxBatch);
    std::vector<struct mmsghdr> rxHdrs(kRxBatch);
    std::vector<struct iovec> rxIovs(kRxBatch);
    std::vector<std::array<uint8_t, 2048>> txBufs(kRxBatch);
    std::vector<struct mmsghdr> txHdrs(kRxBatch);
    std::vector<struct iovec> txIovs(kRxBatch);
    std::vector<int> freeSlots;
    freeSlots.reserve(W);
    bool gsoOk = cfg.gso;
    std::vector<uint8_t> gsoBuf(2048 * 64);
    std::vector<struct pollfd> pfds(nSock);
    for (int k = 0; k < nSock; ++k) pfds[k] = {fds[k], POLLIN, 0};

    ctl->setup.fetch_add(1);

    uint64_t seenGen = 0;
    while (!ctl->shutdown.load(std::memory_order_acquire)) {
        /* wait for the next run (or shutdown) */
        if (ctl->gen.load(std::memory_order_acquire) == seenGen) {
            std::this_thread::sleep_for(std::chrono::microseconds(200));
            continue;
        }
        seenGen = ctl->gen.load(std::memory_order_acquire);

        const uint64_t target = ctl->queriesPerThread;
        const double rate = ctl->rateQps > 0
                                ? ctl->rateQps / (double)cfg.threads
                                : 0;
        uint64_t launched = 0, completed = 0;
        std::fill(sentAt.begin(), sentAt.end(), 0);
        freeSlots.clear();

        /* drain stragglers from a previous run so they cannot collide
         * with fresh slots */
        for (int k = 0; k < nSock; ++k) {
            uint8_t junk[2048];
            while (recv(fds[k], junk, sizeof(junk), MSG_DONTWAIT) > 0) {
            }
        }

        const int64_t tStart = nowUs();

        /* How many more queries may launch right now (rate pacing).
         * Launches are quantized to full GSO bursts (32 segments, or
         * 300 us of tokens at low rates) - smaller quanta shrink the
         * super-packets and the per-packet kernel cost comes back. */
        constexpr uint64_t kQuantum = 32;
        int64_t lastLaunchUs = 0;
        auto allowance = [&]() -> uint64_t {
            if (launched >= target) return 0;
            uint64_t left = target - launched;
            if (rate <= 0) return left;
            int64_t now = nowUs();
            uint64_t paced =
                (uint64_t)((double)(now - tStart) * rate / 1e6);
            if (paced <= launched) return 0;
            uint64_t a = paced - launched;
            if (a < left && a < kQuantum && now - lastLaunchUs < 300)
                return 0;
            lastLaunchUs = now;
            return a < left ? a : left;
        };

        /* Send queries for free slots, grouped per socket
         * (slot % nSock). GSO path: the whole burst is equal-size
         * segments of one super-packet => ONE trip through the
         * kernel's send path instead of one per query. */
        auto batchSend = [&](std::vector<int>& slots) {
            uint64_t budget = allowance();
            size_t used = 0;
            for (int k = 0; k < nSock && budget > 0; ++k) {
                const std::vector<size_t>* cls = nullptr;
                size_t wireSize = 0;
                if (gsoOk) {
                    size_t ni = rng() % wires.size();
                    for (const auto& c : classes)
                        if (wires[c[0]].size() == wires[ni].size()) {
                            cls = &c;
                            break;
                        }
                    wireSize = wires[(*cls)[0]].size();
                }
                int nTx = 0;
                size_t gsoLen = 0;
                auto flush = [&]() {
                    if (nTx == 0) return;
                    if (gsoOk) {
                        struct msghdr mh {};
                        struct iovec iov {gsoBuf.data(), gsoLen};
                        char cbuf[CMSG_SPACE(sizeof(uint16_t))] = {0};
                        mh.msg_iov = &iov;
                        mh.msg_iovlen = 1;
                        if (nTx > 1) {
                            mh.msg_control = cbuf;
                            mh.msg_controllen = sizeof(cbuf);
                            struct cmsghdr* cm = CMSG_FIRSTHDR(&mh);
                            cm->cmsg_level = SOL_UDP;
                            cm->cmsg_type = UDP_SEGMENT;
                            cm->cmsg_len = CMSG_LEN(sizeof(uint16_t));
                            uint16_t seg = (uint16_t)wireSize;
                            memcpy(CMSG_DATA(cm), &seg, sizeof(seg));
                        }
                        ssize_t rv = sendmsg(fds[k], &mh, 0);
                        if (rv < 0) {
                            if (errno == EINVAL || errno == EIO ||
                                errno == ENOTSUP)
                                gsoOk = false;  /* no kernel GSO */
                            /* any failure: re-send the burst as
                             * discrete packets so nothing is lost */
                            for (int m = 0; m < nTx; ++m) {
                                ssize_t r2 = send(
                                    fds[k],
                                    gsoBuf.data() + m * wireSize,
                                    wireSize, 0);
                                (void)r2;
                            }
                        }
                        gsoLen = 0;
                        nTx = 0;
                        return;
                    }
                    int done = 0;
                    while (done < nTx) {
                        int rv = sendmmsg(fds[k], txHdrs.data() + done,
                                          nTx - done, 0);
                        if (rv <= 0) break;
                        done += rv;
                    }
                    nTx = 0;
                };
                /* <= 32 segments per GSO packet (kernel caps at 64),
                 * and never past the ~64KB UDP datagram limit */
                int maxBurst = gsoOk ? 32 : kRxBatch;
                if (gsoOk && wireSize > 0 &&
                    (size_t)maxBurst * wireSize > 60000)
                    maxBurst = (int)(60000 / wireSize);
                if (maxBurst < 1) maxBurst = 1;
                for (size_t si = 0;
                     si < slots.size() && budget > 0; ++si) {
                    int slot = slots[si];
                    if (slot < 0 || slot % nSock != k) continue;
                    size_t ni = cls != nullptr
                                    ? (*cls)[rng() % cls->size()]
                                    : rng() % wires.size();
                    const auto& w = wires[ni];
                    slotSeq[slot]++;
                    uint16_t qid = (uint16_t)(
                        (slot & slotMask) |
                        ((slotSeq[slot] & seqMask) << slotBits));
                    uint8_t* dst;
                    if (gsoOk) {
                        dst = gsoBuf.data() + gsoLen;
                        gsoLen += w.size();
                    } else {
                        dst = txBufs[nTx].data();
                        txIovs[nTx] = {dst, w.size()};
                        memset(&txHdrs[nTx], 0, sizeof(txHdrs[nTx]));
                        txHdrs[nTx].msg_hdr.msg_iov = &txIovs[nTx];
                        txHdrs[nTx].msg_hdr.msg_iovlen = 1;
                    }
                    memcpy(dst, w.data(), w.size());
                    dst[0] = (uint8_t)(qid >> 8);
                    dst[1] = (uint8_t)qid;
                    sentAt[slot] = nowUs();
                    out->sent++;
                    launched++;
                    budget--;
                    used++;
                    slots[si] = -1;  // consumed
                    if (++nTx == maxBurst) flush();
                }
                flush();
            }
            if (used == slots.size() || budget == 0) {
                /* compact: drop consumed entries */
                size_t w = 0;
                for (size_t i = 0; i < slots.size(); ++i)
                    if (slots[i] >= 0) slots[w++] = slots[i];
                slots.resize(w);
            } else {
                slots.clear();
            }
        };

        /* prime the window */
        for (int s = 0; s < W && (uint64_t)s < target; ++s)
            freeSlots.push_back(s);
        batchSend(freeSlots);

        int64_t lastSweep = nowUs();
        while (completed < target && !ctl->shutdown.load()) {
            /* Paced mode: wait only until the next launch quantum
             * accrues. The box runs under a CPU quota, so spinning
             * generator threads steal quota from the server chain and
             * CFS throttling puts whole-group freezes straight into
             * p99; sub-quantum sleeps keep launches smooth without
             * burning idle cycles. */
            int rv;
            if (rate > 0) {
                uint64_t backlog = allowance();
                long waitNs;
                if (backlog >= kQuantum) {
                    waitNs = 0;
                } else {
                    double need =
                        (double)(kQuantum - backlog) / rate * 1e9;
                    waitNs = need < 250000 ? (long)need : 250000;
                }
                struct timespec ts {0, waitNs};
                rv = ppoll(pfds.data(), (nfds_t)nSock, &ts, nullptr);
            } else {
                rv = poll(pfds.data(), (nfds_t)nSock, 50);
            }
            if (rv > 0) {
                for (int k = 0; k < nSock; ++k) {
                    if (!(pfds[k].revents & POLLIN)) continue;
                    while (true) {
                        for (int i = 0; i < kRxBatch; ++i) {
                            rxIovs[i] = {rxBufs[i].data(),
                                         rxBufs[i].size()};
                            memset(&rxHdrs[i], 0, sizeof(rxHdrs[i]));
                            rxHdrs[i].msg_hdr.msg_iov = &rxIovs[i];
                            rxHdrs[i].msg_hdr.msg_iovlen = 1;
                            rxHdrs[i].msg_hdr.msg_control =
                                rxCtrl[i].data();
                            rxHdrs[i].msg_hdr.msg_controllen =
                                rxCtrl[i].size();
                        }
                        int nr = recvmmsg(fds[k], rxHdrs.data(),
                                          kRxBatch, MSG_DONTWAIT,
                                          nullptr);
                        if (nr <= 0) break;
                        int64_t now = nowUs();
                        for (int i = 0; i < nr; ++i) {
                            /* GRO: split a coalesced message into its
                             * equal-size segments (cmsg carries the
                             * segment size) */
                            size_t seg = rxHdrs[i].msg_len;
                            for (struct cmsghdr* cm = CMSG_FIRSTHDR(
                                     &rxHdrs[i].msg_hdr);
                                 cm != nullptr;
                                 cm = CMSG_NXTHDR(&rxHdrs[i].msg_hdr,
                                                  cm)) {
                                if (cm->cmsg_level == SOL_UDP &&
                                    cm->cmsg_type == UDP_GRO) {
                                    uint16_t g;
                                    memcpy(&g, CMSG_DATA(cm),
                                           sizeof(g));
                                    if (g > 0) seg = g;
                                }
                            }
                            size_t total = rxHdrs[i].msg_len;
                            for (size_t off = 0; off < total;
                                 off += seg) {
                            const uint8_t* rb = rxBufs[i].data() + off;
                            size_t len = std::min(seg, total - off);
                            if (len < 12) continue;
                            uint16_t qid =
                                (uint16_t)((rb[0] << 8) | rb[1]);
                            int slot = qid & slotMask;
                            uint16_t seq = (uint16_t)(qid >> slotBits);
                            if (slot >= W || sentAt[slot] == 0 ||
                                (uint16_t)(slotSeq[slot] & seqMask) !=
                                    seq)
                                continue;  // stale/duplicate
                            int64_t lat = now - sentAt[slot];
                            out->latBuckets[latBucket2(lat)]++;
                            out->received++;
                            uint8_t rcode = rb[3] & 0x0F;
                            if (rcode == 0)
                                out->rcodeNoerror++;
                            else
                                out->rcodeOther++;
                            out->answers +=
                                (uint64_t)((rb[6] << 8) | rb[7]);
                            sentAt[slot] = 0;
                            completed++;
                            freeSlots.push_back(slot);
                            }
                        }
                        if (nr < kRxBatch) break;
                    }
                }
            }
            if (launched < target) batchSend(freeSlots);
            else freeSlots.clear();
            int64_t now = nowUs();
            if (now - lastSweep > 100000) {  // timeout sweep / 100ms
                lastSweep = now;
                for (int s = 0; s < W; ++s) {
                    if (sentAt[s] != 0 && now - sentAt[s] > timeoutUs) {
                        out->timeouts++;
                        completed++;
                        sentAt[s] = 0;
                        if (launched < target) freeSlots.push_back(s);
                    }
                }
                if (launched < target) batchSend(freeSlots);
                else freeSlots.clear();
            }
        }
        ctl->running.fetch_sub(1);
    }
    for (int fd : fds) close(fd);
}

}  // namespace

int main(int argc, char** argv) {
    /* a peer closing mid-write must be an EPIPE errno, not process
     * death */
    signal(SIGPIPE, SIG_IGN);
    Config cfg;
    int c;
    while ((c = getopt(argc, argv, "hs:p:n:c:t:P:r:f:B:T:RgD")) != -1) {
        switch (c) {
        case 'R': cfg.rd = true; break;
        case 'g': cfg.gso = false; break;
        case 'D': cfg.daemon = true; break;
        case 's': cfg.server = optarg; break;
        case 'p': cfg.port = (uint16_t)atoi(optarg); break;
        case 'n': cfg.queries = strtoull(optarg, nullptr, 10); break;
        case 'c': cfg.window = atoi(optarg); break;
        case 't': cfg.threads = atoi(optarg); break;
        case 'P': cfg.socksPerThread = atoi(optarg); break;
        case 'r': cfg.rateQps = atof(optarg); break;
        case 'f': cfg.namesFile = optarg; break;
        case 'B': cfg.bindBase = optarg; break;
        case 'T': cfg.timeoutMs = atoi(optarg); break;
        case 'h':
        default:
            fprintf(stderr,
                    "usage: dnsblast -s server -p port [-n queries] "
                    "[-c window] [-t threads] [-P socks/thread] "
                    "[-r offered-qps] [-f names-file] "
                    "[-B bind-base-ip] [-T timeout-ms] [-R] "
                    "[-g no-gso] [-D daemon]\n");
            return c == 'h' ? 0 : 1;
        }
    }

    std::vector<NameEntry> names;
    if (!cfg.namesFile.empty()) {
        std::ifstream f(cfg.namesFile);
        std::string line;
        while (std::getline(f, line)) {
            if (line.empty()) continue;
            std::istringstream ss(line);
            NameEntry e;
            std::string t;
            ss >> e.name >> t;
            e.qtype = dns::typeFromName(t.empty() ? "A" : t);
            if (e.qtype == 0) e.qtype = dns::TYPE_A;
            names.push_back(std::move(e));
        }
    }
    if (names.empty()) names.push_back({"test.foo.com", dns::TYPE_A});

    /* prebuild wire images */
    std::vector<std::vector<uint8_t>> wires;
    wires.reserve(names.size());
    for (const auto& e : names) {
        dns::Message q;
        q.header.id = 0;
        q.header.rd = cfg.rd;
        dns::Question qq;
        qq.name = e.name;
        qq.qtype = e.qtype;
        q.questions.push_back(qq);
        wires.push_back(q.encode(0));
    }

    /* group wires by size for the GSO burst classes */
    SizeClasses classes;
    {
        std::map<size_t, std::vector<size_t>> bySize;
        for (size_t i = 0; i < wires.size(); ++i)
            bySize[wires[i].size()].push_back(i);
        for (auto& [sz, v] : bySize) classes.push_back(std::move(v));
    }

    std::vector<ThreadResult> results(cfg.threads);
    RunCtl ctl;
    std::vector<std::thread> threads;
    for (int i = 0; i < cfg.threads; ++i) {
        threads.emplace_back(
            [&cfg, i, &wires, &classes, &results, &ctl]() {
                worker(cfg, i, wires, classes, &results[i], &ctl);
            });
    }
    /* barrier: socket/buffer setup must not be charged to any run */
    while (ctl.setup.load() < cfg.threads)
        std::this_thread::sleep_for(std::chrono::milliseconds(1));

    auto doRun = [&](uint64_t queries, double rate) {
        for (auto& r : results) r.reset();
        ctl.queriesPerThread = queries / (uint64_t)cfg.threads;
        ctl.rateQps = rate;
        ctl.running.store(cfg.threads);
        int64_t t0 = nowUs();
        ctl.gen.fetch_add(1, std::memory_order_release);
        while (ctl.running.load() > 0)
            std::this_thread::sleep_for(std::chrono::microseconds(200));
        int64_t elapsedUs = nowUs() - t0;

        ThreadResult total;
        for (const auto& r : results) {
            total.sent += r.sent;
            total.received += r.received;
            total.timeouts += r.timeouts;
            total.rcodeNoerror += r.rcodeNoerror;
            total.rcodeOther += r.rcodeOther;
            total.answers += r.answers;
            for (int b = 0; b < kLatBuckets; ++b)
                total.latBuckets[b] += r.latBuckets[b];
        }
        auto pct = [&](double q) {
            uint64_t n = total.received;
            if (n == 0) return 0.0;
            uint64_t want = (uint64_t)(q * (double)n);
            uint64_t cum = 0;
            for (int b = 0; b < kLatBuckets; ++b) {
                cum += total.latBuckets[b];
                if (cum > want) return bucketMidUs(b);
            }
            return bucketMidUs(kLatBuckets - 1);
        };
        double secs = (double)elapsedUs / 1e6;
        double qps = secs > 0 ? (double)total.received / secs : 0;
        printf("{\"sent\": %llu, \"received\": %llu, "
               "\"timeouts\": %llu, \"noerror\": %llu, "
               "\"other_rcode\": %llu, \"answers\": %llu, "
               "\"elapsed_s\": %.6f, \"qps\": %.1f, "
               "\"p50_us\": %.1f, \"p90_us\": %.1f, "
               "\"p99_us\": %.1f}\n",
               (unsigned long long)total.sent,
               (unsigned long long)total.received,
               (unsigned long long)total.timeouts,
               (unsigned long long)total.rcodeNoerror,
               (unsigned long long)total.rcodeOther,
               (unsigned long long)total.answers, secs, qps,
               pct(0.50), pct(0.90), pct(0.99));
        fflush(stdout);
    };

    if (cfg.daemon) {
        /* "RUN <queries> <rate>" per line; EOF = shutdown */
        char line[256];
        while (fgets(line, sizeof(line), stdin) != nullptr) {
            unsigned long long q = 0;
            double r = 0;
            if (sscanf(line, "RUN %llu %lf", &q, &r) >= 1 && q > 0)
                doRun(q, r);
        }
    } else {
        doRun(cfg.queries, cfg.rateQps);
    }

    ctl.shutdown.store(true, std::memory_order_release);
    for (auto& t : threads) t.join();
    return 0;
}
