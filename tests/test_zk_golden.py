"""Golden ZooKeeper wire-protocol byte vectors (interop pinning).

No JVM exists in this image, so a genuine ZooKeeper cannot run here
(the reference bakes zookeeper-3.4.12 into its image,
/root/reference/Makefile:74-77, and tests against it). To keep the
from-scratch client (native/zk/client.cpp) and server (zkd,
binder_amd/stubzk.py) pinned to the REAL protocol rather than merely
to each other, these tests hand-construct wire images from the
published jute IDL (zookeeper.jute: ConnectRequest/ConnectResponse,
RequestHeader/ReplyHeader, GetDataRequest/Response, Create, GetChildren2,
WatcherEvent, Stat) and the fixed opcode table, byte for byte, with the
derivation of every field inline — and then require:

  * the server accepts and answers EXACTLY these spec-built requests
    (including a 3.4.0-era ConnectRequest WITHOUT the trailing
    readOnly flag, which 3.4.6+ clients append);
  * the client EMITS exactly the spec bytes for its handshake and
    requests, and correctly consumes spec-built responses, when spoken
    to by a fake endpoint that is nothing but these vectors.

All jute integers are big-endian; `string`/`buffer` are int32 length +
bytes (length -1 = null); `vector<T>` is int32 count + elements;
`boolean` is one byte. Stat is 8 fields of int64/int32 in IDL order:
czxid, mzxid, ctime, mtime (int64), version, cversion, aversion
(int32), ephemeralOwner (int64), dataLength, numChildren (int32),
pzxid (int64) — 68 bytes.
"""
import socket
import struct
import subprocess
import time

import pytest

from zkwire import (ACL_OPEN, be32, be64, connect_request, jstr, packet,
                    parse_reply_header, parse_stat, read_packet, req)

from binder_amd import REPO_ROOT
from binder_amd.harness import NativeZkd
from binder_amd.stubzk import StubZk

ZKTOOL = REPO_ROOT / "bin" / "zktool"


# --- server side: zkd must accept/answer the spec bytes -------------

@pytest.fixture(params=["stub", "native"])
def zk(request, tmp_path):
    """Both registry implementations must accept/answer the spec
    bytes: the in-proc Python stub (CI fixture) AND the native zkd
    (the supported single-node registry)."""
    if request.param == "stub":
        z = StubZk().start()
        z.expire_all = z.expire_sessions
    else:
        # short session timeout so natural expiry is testable
        z = NativeZkd(session_timeout_ms=800).start()
        z.expire_all = lambda: time.sleep(2.2)
    yield z
    z.stop()


def connect_raw(zk_obj, **kw):
    s = socket.socket()
    s.settimeout(5)
    s.connect(("127.0.0.1", zk_obj.port))
    s.sendall(connect_request(**kw))
    body = read_packet(s)
    # ConnectResponse: int protocolVersion; int timeOut; long sessionId;
    # buffer passwd; [boolean readOnly]
    proto, timeout, sid = struct.unpack(">iiq", body[:16])
    (plen,) = struct.unpack(">i", body[16:20])
    passwd = body[20:20 + plen]
    return s, proto, timeout, sid, passwd


def test_server_accepts_classic_connect_without_readonly(zk):
    """A 3.4.0-era ConnectRequest (44-byte body, no readOnly flag)
    must be accepted — zkstream and old clients send this form."""
    s, proto, timeout, sid, passwd = connect_raw(zk)
    try:
        assert proto == 0
        assert timeout > 0
        assert sid != 0
        assert len(passwd) == 16
    finally:
        s.close()


def test_server_connect_with_readonly_flag(zk):
    s, proto, timeout, sid, passwd = connect_raw(zk, read_only=False)
    try:
        assert proto == 0 and sid != 0
    finally:
        s.close()


def test_server_session_resume_and_expiry_bytes(zk):
    s1, _, _, sid, passwd = connect_raw(zk)
    s1.close()
    # resume: same sessionId+passwd => server must return the SAME sid
    s2, proto, timeout, sid2, _ = connect_raw(zk, session_id=sid,
                                              passwd=passwd)
    assert sid2 == sid
    s2.close()
    # expiry: after the server expires the session, the spec response
    # is sessionId=0 AND timeOut=0 (clients detect expiry by that)
    zk.expire_all()
    s3, proto, timeout, sid3, _ = connect_raw(zk, session_id=sid,
                                              passwd=passwd)
    assert (timeout, sid3) == (0, 0)
    s3.close()


def test_server_create_getdata_children_golden(zk):
    s, *_ = connect_raw(zk)
    try:
        # create(xid=1): path, data, acl, flags(persistent=0)
        s.sendall(req(1, 1, jstr("/g") + jstr(b"hi") + ACL_OPEN +
                      be32(0)))
        xid, zxid, err, rest = parse_reply_header(read_packet(s))
        assert (xid, err) == (1, 0)
        assert zxid > 0, "create must carry the txn zxid"
        # CreateResponse: string path
        (plen,) = struct.unpack(">i", rest[:4])
        assert rest[4:4 + plen] == b"/g"

        # getData(xid=2, watch=0)
        s.sendall(req(2, 4, jstr("/g") + b"\x00"))
        xid, zxid, err, rest = parse_reply_header(read_packet(s))
        assert (xid, err) == (2, 0)
        (dlen,) = struct.unpack(">i", rest[:4])
        assert rest[4:4 + dlen] == b"hi"
        stat, tail = parse_stat(rest[4 + dlen:])
        assert stat["dataLength"] == 2
        assert stat["numChildren"] == 0
        assert stat["version"] == 0
        assert tail == b""

        # getChildren2(xid=3, watch=0) on / must list "g" + stat
        s.sendall(req(3, 12, jstr("/") + b"\x00"))
        xid, zxid, err, rest = parse_reply_header(read_packet(s))
        assert (xid, err) == (3, 0)
        (count,) = struct.unpack(">i", rest[:4])
        kids, off = [], 4
        for _ in range(count):
            (klen,) = struct.unpack(">i", rest[off:off + 4])
            kids.append(rest[off + 4:off + 4 + klen].decode())
            off += 4 + klen
        assert "g" in kids
        stat, tail = parse_stat(rest[off:])
        assert tail == b""

        # setData(xid=4): path, data, version(-1)
        s.sendall(req(4, 5, jstr("/g") + jstr(b"bye") + be32(-1)))
        xid, zxid, err, rest = parse_reply_header(read_packet(s))
        assert (xid, err) == (4, 0)
        stat, tail = parse_stat(rest)
        assert stat["version"] == 1
        assert stat["dataLength"] == 3

        # delete(xid=5): path, version(-1) => empty response body
        s.sendall(req(5, 2, jstr("/g") + be32(-1)))
        xid, zxid, err, rest = parse_reply_header(read_packet(s))
        assert (xid, err) == (5, 0)
        assert rest == b""

        # getData on the deleted node => err ZNONODE (-101)
        s.sendall(req(6, 4, jstr("/g") + b"\x00"))
        xid, zxid, err, rest = parse_reply_header(read_packet(s))
        assert (xid, err) == (6, -101)
    finally:
        s.close()


def test_server_ping_golden(zk):
    s, *_ = connect_raw(zk)
    try:
        # ping: xid=-2, type=11, empty payload; reply echoes xid -2
        s.sendall(req(-2, 11))
        xid, zxid, err, rest = parse_reply_header(read_packet(s))
        assert (xid, err) == (-2, 0)
    finally:
        s.close()


def test_server_watch_event_golden_bytes(zk):
    """The notification packet is ReplyHeader{xid=-1, err=0} +
    WatcherEvent{int type; int state; string path} — NodeDataChanged=3,
    SyncConnected=3 (fixed constants in the ZK protocol)."""
    a, *_ = connect_raw(zk)
    b, *_ = connect_raw(zk)
    try:
        a.sendall(req(1, 1, jstr("/w") + jstr(b"x") + ACL_OPEN +
                      be32(0)))
        read_packet(a)
        # register a data watch from conn A
        a.sendall(req(2, 4, jstr("/w") + b"\x01"))
        read_packet(a)
        # mutate from conn B
        b.sendall(req(1, 5, jstr("/w") + jstr(b"y") + be32(-1)))
        read_packet(b)
        # A's next packet must be the notification
        body = read_packet(a)
        xid, zxid, err, rest = parse_reply_header(body)
        assert (xid, err) == (-1, 0)
        ev_type, ev_state = struct.unpack(">ii", rest[:8])
        (plen,) = struct.unpack(">i", rest[8:12])
        assert ev_type == 3        # NodeDataChanged
        assert ev_state == 3       # SyncConnected
        assert rest[12:12 + plen] == b"/w"
        assert rest[12 + plen:] == b""
    finally:
        a.close()
        b.close()


# --- client side: zktool must EMIT the spec bytes -------------------

def test_client_emits_golden_bytes(tmp_path):
    """Drive bin/zktool (the native ZkClient) against a fake endpoint
    made of nothing but spec-built vectors: its ConnectRequest and
    GetDataRequest must match the golden bytes EXACTLY, and it must
    consume a spec-built ConnectResponse + GetDataResponse."""
    srv = socket.socket()
    srv.bind(("127.0.0.1", 0))
    srv.listen(1)
    port = srv.getsockname()[1]

    proc = subprocess.Popen(
        [str(ZKTOOL), "-s", f"127.0.0.1:{port}", "get", "/com"],
        stdout=subprocess.PIPE, stderr=subprocess.PIPE)
    try:
        srv.settimeout(10)
        conn, _ = srv.accept()
        conn.settimeout(10)

        # --- handshake: must be the exact 3.4.6+ golden form
        got = read_packet(conn)
        golden_connect = connect_request(timeout_ms=10000,  # zktool cfg
                                         read_only=False)[4:]
        assert got == golden_connect, \
            f"ConnectRequest drifted from spec:\n {got.hex()}\n " \
            f"{golden_connect.hex()}"

        # golden ConnectResponse: proto 0, negotiated 30000, a sid,
        # 16-byte passwd, readOnly 0
        sid = 0x0100000000000042
        conn.sendall(packet(be32(0) + be32(30000) + be64(sid) +
                            jstr(b"\xab" * 16) + b"\x00"))

        # --- first request: GetData xid=1 type=4 "/com" watch=0
        got = read_packet(conn)
        assert got == be32(1) + be32(4) + jstr("/com") + b"\x00", \
            f"GetDataRequest drifted from spec: {got.hex()}"

        # golden GetDataResponse: header(xid=1, zxid=7, err=0) +
        # buffer "hello" + Stat(czxid=5, mzxid=7, dataLength=5, ...)
        stat = be64(5) + be64(7) + be64(1000) + be64(2000) + \
            be32(1) + be32(0) + be32(0) + be64(0) + be32(5) + \
            be32(0) + be64(7)
        conn.sendall(packet(be32(1) + be64(7) + be32(0) +
                            jstr(b"hello") + stat))

        out, err = proc.communicate(timeout=10)
        assert proc.returncode == 0, err.decode()
        assert out == b"hello\n"
    finally:
        proc.kill()
        srv.close()


def test_client_handles_golden_expiry_response(tmp_path):
    """A spec-built expiry ConnectResponse (timeOut=0, sessionId=0)
    must make the client drop the dead session and reconnect with a
    FRESH handshake (sessionId=0, zeroed passwd) — the rebuild path
    real ZooKeeper triggers on session expiry."""
    srv = socket.socket()
    srv.bind(("127.0.0.1", 0))
    srv.listen(2)
    port = srv.getsockname()[1]
    proc = subprocess.Popen(
        [str(ZKTOOL), "-s", f"127.0.0.1:{port}", "get", "/com"],
        stdout=subprocess.PIPE, stderr=subprocess.PIPE)
    try:
        srv.settimeout(10)
        conn, _ = srv.accept()
        conn.settimeout(10)
        read_packet(conn)
        # spec expiry response: proto 0, timeOut 0, sessionId 0
        conn.sendall(packet(be32(0) + be32(0) + be64(0) +
                            jstr(b"\x00" * 16) + b"\x00"))
        # the client must come back with a brand-new session handshake
        conn2, _ = srv.accept()
        conn2.settimeout(10)
        got = read_packet(conn2)
        golden_fresh = connect_request(timeout_ms=10000,
                                       read_only=False)[4:]
        assert got == golden_fresh, \
            f"post-expiry handshake drifted: {got.hex()}"
        conn2.close()
        conn.close()
    finally:
        proc.kill()
        srv.close()


def test_server_exists_getchildren_and_error_codes(zk):
    """Remaining spec vectors: exists (op 3, Stat-only response),
    plain getChildren (op 8, no trailing Stat), ZNODEEXISTS (-110) on
    duplicate create, ZNONODE (-101) on exists/delete of a missing
    node, ZNOTEMPTY (-111) on deleting a parent."""
    s, *_ = connect_raw(zk)
    try:
        s.sendall(req(1, 1, jstr("/p") + jstr(b"d") + ACL_OPEN +
                      be32(0)))
        assert parse_reply_header(read_packet(s))[2] == 0
        s.sendall(req(2, 1, jstr("/p/k") + jstr(b"") + ACL_OPEN +
                      be32(0)))
        assert parse_reply_header(read_packet(s))[2] == 0

        # exists: ReplyHeader + bare Stat (68 bytes, no data buffer)
        s.sendall(req(3, 3, jstr("/p") + b"\x00"))
        xid, zxid, err, rest = parse_reply_header(read_packet(s))
        assert (xid, err) == (3, 0)
        stat, tail = parse_stat(rest)
        assert tail == b""
        assert stat["numChildren"] == 1
        assert stat["dataLength"] == 1

        # exists on a missing node => ZNONODE, empty body
        s.sendall(req(4, 3, jstr("/nope") + b"\x00"))
        xid, zxid, err, rest = parse_reply_header(read_packet(s))
        assert (xid, err) == (4, -101)
        assert rest == b""

        # plain getChildren (op 8): vector<string> only, NO Stat
        s.sendall(req(5, 8, jstr("/p") + b"\x00"))
        xid, zxid, err, rest = parse_reply_header(read_packet(s))
        assert (xid, err) == (5, 0)
        (count,) = struct.unpack(">i", rest[:4])
        assert count == 1
        (klen,) = struct.unpack(">i", rest[4:8])
        assert rest[8:8 + klen] == b"k"
        assert rest[8 + klen:] == b"", "op 8 must not append a Stat"

        # duplicate create => ZNODEEXISTS
        s.sendall(req(6, 1, jstr("/p") + jstr(b"x") + ACL_OPEN +
                      be32(0)))
        assert parse_reply_header(read_packet(s))[2] == -110

        # delete non-empty parent => ZNOTEMPTY
        s.sendall(req(7, 2, jstr("/p") + be32(-1)))
        assert parse_reply_header(read_packet(s))[2] == -111

        # delete missing => ZNONODE
        s.sendall(req(8, 2, jstr("/gone") + be32(-1)))
        assert parse_reply_header(read_packet(s))[2] == -101
    finally:
        s.close()


def test_server_child_watch_fires_on_create(zk):
    """getChildren with watch=1: a subsequent child create must push
    ReplyHeader{xid=-1} + WatcherEvent{NodeChildrenChanged=4,
    SyncConnected=3, parent path}."""
    a, *_ = connect_raw(zk)
    b, *_ = connect_raw(zk)
    try:
        a.sendall(req(1, 1, jstr("/cw") + jstr(b"") + ACL_OPEN +
                      be32(0)))
        read_packet(a)
        a.sendall(req(2, 12, jstr("/cw") + b"\x01"))  # watch
        read_packet(a)
        b.sendall(req(1, 1, jstr("/cw/kid") + jstr(b"") + ACL_OPEN +
                      be32(0)))
        read_packet(b)
        body = read_packet(a)
        xid, zxid, err, rest = parse_reply_header(body)
        assert (xid, err) == (-1, 0)
        ev_type, ev_state = struct.unpack(">ii", rest[:8])
        (plen,) = struct.unpack(">i", rest[8:12])
        assert ev_type == 4       # NodeChildrenChanged
        assert ev_state == 3      # SyncConnected
        assert rest[12:12 + plen] == b"/cw"
    finally:
        a.close()
        b.close()
