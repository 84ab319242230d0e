"""Spec-derived ZooKeeper jute wire helpers (shared by the golden
vector tests and the native-zkd tests). Hand-written from the
published jute IDL — intentionally NOT reusing stubzk's or the native
code's packers, so tests pin both implementations to the spec."""
import struct


# --- independent jute packers (do NOT reuse stubzk's) ---------------

def be32(v):
    return struct.pack(">i", v)


def be64(v):
    return struct.pack(">q", v)


def jstr(s):
    b = s.encode() if isinstance(s, str) else s
    return be32(len(b)) + b


def packet(body):
    """Every ZK packet is a 4-byte BE length prefix + body."""
    return be32(len(body)) + body


def read_packet(sock):
    hdr = b""
    while len(hdr) < 4:
        chunk = sock.recv(4 - len(hdr))
        assert chunk, "peer closed"
        hdr += chunk
    (n,) = struct.unpack(">i", hdr)
    body = b""
    while len(body) < n:
        chunk = sock.recv(n - len(body))
        assert chunk, "peer closed mid-packet"
        body += chunk
    return body


# ConnectRequest (zookeeper.jute proto.ConnectRequest):
#   int protocolVersion; long lastZxidSeen; int timeOut;
#   long sessionId; buffer passwd;  [+ optional boolean readOnly 3.4.6+]
def connect_request(timeout_ms=30000, session_id=0,
                    passwd=b"\x00" * 16, read_only=None):
    body = be32(0) + be64(0) + be32(timeout_ms) + be64(session_id) + \
        jstr(passwd)
    if read_only is not None:
        body += bytes([1 if read_only else 0])
    return packet(body)


# RequestHeader: int xid; int type  (fixed opcodes: create=1, delete=2,
# getData=4, setData=5, getChildren2=12, ping=11 w/ xid -2)
def req(xid, op, payload=b""):
    return packet(be32(xid) + be32(op) + payload)


# default ACL world:anyone with ALL perms (0x1f); jute:
# vector<ACL>{ int perms; Id id { string scheme; string id; } }
ACL_OPEN = be32(1) + be32(0x1F) + jstr("world") + jstr("anyone")


def parse_reply_header(body):
    xid, zxid, err = struct.unpack(">iqi", body[:16])
    return xid, zxid, err, body[16:]


def parse_stat(b):
    names = ("czxid", "mzxid", "ctime", "mtime", "version", "cversion",
             "aversion", "ephemeralOwner", "dataLength", "numChildren",
             "pzxid")
    vals = struct.unpack(">qqqqiiiqiiq", b[:68])
    return dict(zip(names, vals)), b[68:]
