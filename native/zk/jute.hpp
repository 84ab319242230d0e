/*
 * binder-amd: jute (ZooKeeper wire) primitives.
 *
 * The ZooKeeper client protocol frames every packet as u32be length +
 * body; ints/longs big-endian; strings/buffers are i32 length (-1 null)
 * + bytes; bools one byte. The reference consumed this via the zkstream
 * npm package; this is a from-scratch implementation of
 * the same protocol (ZooKeeper 3.4-compatible subset).
 */
#pragma once

#include <cstdint>
#include <cstring>
#include <optional>
#include <string>
#include <vector>

namespace bamd::zk {

/* op codes */
enum : int32_t {
    OP_NOTIFICATION = 0,
    OP_CREATE = 1,
    OP_DELETE = 2,
    OP_EXISTS = 3,
    OP_GETDATA = 4,
    OP_SETDATA = 5,
    OP_GETACL = 6,
    OP_SETACL = 7,
    OP_GETCHILDREN = 8,
    OP_SYNC = 9,
    OP_PING = 11,
    OP_GETCHILDREN2 = 12,
    OP_CLOSE = -11,
};

/* special xids */
enum : int32_t {
    XID_NOTIFICATION = -1,
    XID_PING = -2,
    XID_AUTH = -4,
    XID_SET_WATCHES = -8,
};

/* error codes (subset) */
enum : int32_t {
    ZOK = 0,
    ZSYSTEMERROR = -1,
    ZCONNECTIONLOSS = -4,
    ZMARSHALLINGERROR = -5,
    ZNONODE = -101,
    ZNOAUTH = -102,
    ZBADVERSION = -103,
    ZNOCHILDRENFOREPHEMERALS = -108,
    ZNODEEXISTS = -110,
    ZNOTEMPTY = -111,
    ZSESSIONEXPIRED = -112,
};

/* watcher event types */
enum : int32_t {
    EV_NODE_CREATED = 1,
    EV_NODE_DELETED = 2,
    EV_NODE_DATA_CHANGED = 3,
    EV_NODE_CHILDREN_CHANGED = 4,
};
/* keeper states */
enum : int32_t {
    STATE_DISCONNECTED = 0,
    STATE_SYNC_CONNECTED = 3,
    STATE_EXPIRED = -112,
};

/* create flags */
enum : int32_t {
    CREATE_PERSISTENT = 0,
    CREATE_EPHEMERAL = 1,
    CREATE_SEQUENTIAL = 2,
};

struct Stat {
    int64_t czxid = 0, mzxid = 0, ctime = 0, mtime = 0;
    int32_t version = 0, cversion = 0, aversion = 0;
    int64_t ephemeralOwner = 0;
    int32_t dataLength = 0, numChildren = 0;
    int64_t pzxid = 0;
};

struct Writer {
    std::string buf;
    void i32(int32_t v) {
        uint32_t u = (uint32_t)v;
        char b[4] = {(char)(u >> 24), (char)(u >> 16), (char)(u >> 8),
                     (char)u};
        buf.append(b, 4);
    }
    void i64(int64_t v) {
        uint64_t u = (uint64_t)v;
        for (int s = 56; s >= 0; s -= 8) buf.push_back((char)(u >> s));
    }
    void boolean(bool b) { buf.push_back(b ? 1 : 0); }
    void str(const std::string& s) {
        i32((int32_t)s.size());
        buf += s;
    }
    void buffer(const std::string& s) { str(s); }
    void nullBuffer() { i32(-1); }
};

struct Reader {
    const uint8_t* p;
    size_t len;
    size_t pos = 0;
    bool ok = true;

    bool need(size_t n) {
        if (pos + n > len) {
            ok = false;
            return false;
        }
        return true;
    }
    int32_t i32() {
        if (!need(4)) return 0;
        int32_t v = (int32_t)(((uint32_t)p[pos] << 24) |
                              ((uint32_t)p[pos + 1] << 16) |
                              ((uint32_t)p[pos + 2] << 8) |
                              (uint32_t)p[pos + 3]);
        pos += 4;
        return v;
    }
    int64_t i64() {
        if (!need(8)) return 0;
        uint64_t v = 0;
        for (int i = 0; i < 8; ++i) v = (v << 8) | p[pos + i];
        pos += 8;
        return (int64_t)v;
    }
    bool boolean() {
        if (!need(1)) return false;
        return p[pos++] != 0;
    }
    std::string str() {
        int32_t n = i32();
        if (n < 0) return "";
        if (!need((size_t)n)) return "";
        std::string s((const char*)p + pos, (size_t)n);
        pos += (size_t)n;
        return s;
    }
    std::vector<std::string> strVec() {
        int32_t n = i32();
        std::vector<std::string> out;
        if (n < 0) return out;
        if (n > 1 << 22) {  // sanity
            ok = false;
            return out;
        }
        out.reserve((size_t)n);
        for (int32_t i = 0; i < n && ok; ++i) out.push_back(str());
        return out;
    }
    Stat stat() {
        Stat s;
        s.czxid = i64();
        s.mzxid = i64();
        s.ctime = i64();
        s.mtime = i64();
        s.version = i32();
        s.cversion = i32();
        s.aversion = i32();
        s.ephemeralOwner = i64();
        s.dataLength = i32();
        s.numChildren = i32();
        s.pzxid = i64();
        return s;
    }
    bool atEnd() const { return pos >= len; }
};

/* Append the world:anyone ACL vector used for all our creates. */
inline void writeOpenAcl(Writer& w) {
    w.i32(1);    // one ACL
    w.i32(31);   // perms: ALL
    w.str("world");
    w.str("anyone");
}

}  // namespace bamd::zk
