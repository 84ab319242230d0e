/*
 * zklogcat: ZooKeeper replicated-transaction-log decoder (capability
 * parity with the reference's src/zklog.c forensic tool, SURVEY.md §2
 * row 10; fresh implementation of the public FileTxnLog format).
 *
 * Format (ZooKeeper FileTxnLog, version 2):
 *   file header:  magic 'ZKLG' (0x5A4B4C47) i32, version i32, dbid i64
 *   per txn:      checksum i64 (adler32 of the txn bytes),
 *                 txnlen i32, txn bytes, EOR byte 0x42
 *   txn bytes:    TxnHeader{clientId i64, cxid i32, zxid i64,
 *                 time i64, type i32} + type-specific jute record
 *   zero txnlen / zero checksum => preallocated tail, stop.
 *
 * Output: one JSON object per transaction on stdout.
 * Flags (mirroring the reference tool's):
 *   -d           include node data as hex
 *   -S           after decoding, dump sessions still open + durations
 *   -t <secs>    only txns in the last <secs> window of the log
 *   -s <hex>     only txns for session id (hex)
 *   -z <id>      only sessions created by server <id> (top byte of
 *                session id)
 */
#include <fcntl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>

#include <cinttypes>
#include <cstdio>
#include <cstring>
#include <map>
#include <string>
#include <vector>

#include "../common/json.hpp"
#include "../zk/jute.hpp"

using namespace bamd;

namespace {

constexpr uint32_t kMagic = 0x5A4B4C47;  // 'ZKLG'
constexpr uint8_t kEor = 0x42;

/* txn op types (ZooKeeper ZooDefs + Txn types) */
const char* txnTypeName(int32_t t) {
    switch (t) {
    case -10: return "createSession";
    case -11: return "closeSession";
    case -1: return "error";
    case 1: return "create";
    case 2: return "delete";
    case 3: return "exists";
    case 4: return "getData";
    case 5: return "setData";
    case 7: return "setACL";
    case 9: return "sync";
    case 13: return "check";
    case 14: return "multi";
    case 15: return "create2";
    default: return "unknown";
    }
}

struct Options {
    bool hexData = false;
    bool dumpSessions = false;
    int64_t windowSecs = 0;
    uint64_t sessionFilter = 0;
    int serverFilter = -1;
};

struct SessionInfo {
    int64_t createdMs = 0;
    int64_t closedMs = 0;
    int timeoutMs = 0;
    uint64_t txns = 0;
};

std::string hexDump(const std::string& data) {
    static const char* kHex = "0123456789abcdef";
    std::string out;
    out.reserve(data.size() * 2);
    for (unsigned char c : data) {
        out.push_back(kHex[c >> 4]);
        out.push_back(kHex[c & 0xF]);
    }
    return out;
}

void decodeRecord(int32_t type, zk::Reader& r, Json& out,
                  const Options& opt) {
    switch (type) {
    case 1:    /* create */
    case 15: { /* create2 */
        out.set("path", Json(r.str()));
        std::string data = r.str();
        if (opt.hexData) out.set("data", Json(hexDump(data)));
        out.set("data_len", Json((int64_t)data.size()));
        int32_t nacl = r.i32();
        for (int32_t i = 0; i < nacl && r.ok; ++i) {
            r.i32();
            r.str();
            r.str();
        }
        out.set("ephemeral", Json(r.boolean()));
        if (!r.atEnd()) out.set("parent_cversion", Json((int64_t)r.i32()));
        break;
    }
    case 2:  /* delete */
        out.set("path", Json(r.str()));
        break;
    case 5: { /* setData */
        out.set("path", Json(r.str()));
        std::string data = r.str();
        if (opt.hexData) out.set("data", Json(hexDump(data)));
        out.set("data_len", Json((int64_t)data.size()));
        out.set("version", Json((int64_t)r.i32()));
        break;
    }
    case 13:  /* check */
        out.set("path", Json(r.str()));
        out.set("version", Json((int64_t)r.i32()));
        break;
    case 7: { /* setACL */
        out.set("path", Json(r.str()));
        int32_t nacl = r.i32();
        JsonArray acls;
        for (int32_t i = 0; i < nacl && r.ok; ++i) {
            Json a = Json::object();
            a.set("perms", Json((int64_t)r.i32()));
            a.set("scheme", Json(r.str()));
            a.set("id", Json(r.str()));
            acls.push_back(std::move(a));
        }
        out.set("acl", Json(std::move(acls)));
        out.set("version", Json((int64_t)r.i32()));
        break;
    }
    case -1:  /* error */
        out.set("err", Json((int64_t)r.i32()));
        break;
    case -10:  /* createSession */
        out.set("timeout_ms", Json((int64_t)r.i32()));
        break;
    case -11:  /* closeSession (may carry ephemeral list in 3.6+) */
        break;
    case 14: { /* multi: vector<Txn{type, data buffer}>, recursive */
        int32_t n = r.i32();
        JsonArray inner;
        for (int32_t i = 0; i < n && r.ok; ++i) {
            int32_t itype = r.i32();
            std::string data = r.str();
            Json sub = Json::object();
            sub.set("type", Json(txnTypeName(itype)));
            zk::Reader ir{(const uint8_t*)data.data(), data.size()};
            decodeRecord(itype, ir, sub, opt);
            inner.push_back(std::move(sub));
        }
        out.set("ops", Json(std::move(inner)));
        break;
    }
    default:
        break;
    }
}

int doFile(const char* path, const Options& opt,
           std::map<uint64_t, SessionInfo>& sessions) {
    int fd = open(path, O_RDONLY);
    if (fd < 0) {
        fprintf(stderr, "zklogcat: cannot open %s: %s\n", path,
                strerror(errno));
        return 1;
    }
    struct stat st;
    fstat(fd, &st);
    if (st.st_size < 16) {
        close(fd);
        fprintf(stderr, "zklogcat: %s: too short\n", path);
        return 1;
    }
    const uint8_t* base = (const uint8_t*)mmap(
        nullptr, (size_t)st.st_size, PROT_READ, MAP_PRIVATE, fd, 0);
    close(fd);
    if (base == MAP_FAILED) {
        fprintf(stderr, "zklogcat: mmap %s failed\n", path);
        return 1;
    }

    zk::Reader r{base, (size_t)st.st_size};
    uint32_t magic = (uint32_t)r.i32();
    int32_t version = r.i32();
    int64_t dbid = r.i64();
    if (magic != kMagic || version != 2) {
        fprintf(stderr,
                "zklogcat: %s: bad magic/version (%08x v%d)\n", path,
                magic, version);
        munmap((void*)base, (size_t)st.st_size);
        return 1;
    }

    /* first pass for -t: find the newest txn time */
    int64_t newestMs = 0;
    if (opt.windowSecs > 0) {
        zk::Reader rr = r;
        while (rr.ok && rr.pos + 12 <= rr.len) {
            int64_t crc = rr.i64();
            int32_t len = rr.i32();
            if (crc == 0 || len <= 0 || !rr.need((size_t)len + 1)) break;
            zk::Reader tr{rr.p + rr.pos, (size_t)len};
            tr.i64();
            tr.i32();
            tr.i64();
            int64_t timeMs = tr.i64();
            if (timeMs > newestMs) newestMs = timeMs;
            rr.pos += (size_t)len + 1;
        }
    }

    while (r.ok && r.pos + 12 <= r.len) {
        int64_t crc = r.i64();
        int32_t len = r.i32();
        if (crc == 0 || len <= 0) break;  // preallocated tail
        if (!r.need((size_t)len + 1)) break;
        const uint8_t* txn = r.p + r.pos;
        uint8_t eor = r.p[r.pos + (size_t)len];
        r.pos += (size_t)len + 1;
        if (eor != kEor) {
            fprintf(stderr, "zklogcat: %s: missing EOR, stopping\n",
                    path);
            break;
        }
        zk::Reader tr{txn, (size_t)len};
        int64_t clientId = tr.i64();
        int32_t cxid = tr.i32();
        int64_t zxid = tr.i64();
        int64_t timeMs = tr.i64();
        int32_t type = tr.i32();
        if (!tr.ok) break;

        /* session bookkeeping (reference tracks open sessions, -S) */
        SessionInfo& si = sessions[(uint64_t)clientId];
        si.txns++;
        if (type == -10) {
            si.createdMs = timeMs;
            zk::Reader peek = tr;
            si.timeoutMs = peek.i32();
        } else if (type == -11) {
            si.closedMs = timeMs;
        }

        if (opt.sessionFilter != 0 &&
            (uint64_t)clientId != opt.sessionFilter)
            continue;
        if (opt.serverFilter >= 0 &&
            (int)(((uint64_t)clientId >> 56) & 0xFF) != opt.serverFilter)
            continue;
        if (opt.windowSecs > 0 &&
            timeMs < newestMs - opt.windowSecs * 1000)
            continue;

        Json out = Json::object();
        char buf[32];
        snprintf(buf, sizeof(buf), "%" PRIx64, (uint64_t)clientId);
        out.set("session", Json(std::string(buf)));
        out.set("cxid", Json((int64_t)cxid));
        out.set("zxid", Json(zxid));
        out.set("time_ms", Json(timeMs));
        out.set("type", Json(txnTypeName(type)));
        out.set("dbid", Json(dbid));
        decodeRecord(type, tr, out, opt);
        printf("%s\n", out.dump().c_str());
    }
    munmap((void*)base, (size_t)st.st_size);
    return 0;
}

}  // namespace

int main(int argc, char** argv) {
    Options opt;
    int c;
    while ((c = getopt(argc, argv, "hdSt:s:z:")) != -1) {
        switch (c) {
        case 'd': opt.hexData = true; break;
        case 'S': opt.dumpSessions = true; break;
        case 't': opt.windowSecs = atoll(optarg); break;
        case 's': opt.sessionFilter = strtoull(optarg, nullptr, 16); break;
        case 'z': opt.serverFilter = atoi(optarg); break;
        case 'h':
        default:
            fprintf(stderr,
                    "usage: zklogcat [-d] [-S] [-t secs] [-s sessionhex] "
                    "[-z serverid] <txnlog> [...]\n");
            return c == 'h' ? 0 : 1;
        }
    }
    if (optind >= argc) {
        fprintf(stderr, "zklogcat: no input files\n");
        return 1;
    }
    std::map<uint64_t, SessionInfo> sessions;
    int rc = 0;
    for (int i = optind; i < argc; ++i)
        rc |= doFile(argv[i], opt, sessions);

    if (opt.dumpSessions) {
        for (const auto& [sid, si] : sessions) {
            if (opt.serverFilter >= 0 &&
                (int)((sid >> 56) & 0xFF) != opt.serverFilter)
                continue;
            Json out = Json::object();
            char buf[32];
            snprintf(buf, sizeof(buf), "%" PRIx64, sid);
            out.set("session", Json(std::string(buf)));
            out.set("open", Json(si.closedMs == 0));
            out.set("timeout_ms", Json((int64_t)si.timeoutMs));
            out.set("txns", Json((int64_t)si.txns));
            if (si.createdMs != 0 && si.closedMs != 0)
                out.set("duration_ms", Json(si.closedMs - si.createdMs));
            printf("%s\n", out.dump().c_str());
        }
    }
    return rc;
}
