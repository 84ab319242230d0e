"""Native zkd (bin/zkd): the supported single-node registry.

Covers what the golden wire vectors (test_zk_golden.py, which already
runs against this server) do not: restart durability via the
FileTxnLog, torn-tail truncation, compaction, session-timeout expiry
reaping ephemerals (with watches fired), sequential creates, version
checks, four-letter words, zklogcat decoding zkd's log, and the full
binderd-mirror-over-zkd end-to-end path.
"""
import json
import socket
import struct
import subprocess
import time

import pytest

from zkwire import (ACL_OPEN, be32, connect_request, jstr,
                    parse_reply_header, parse_stat, read_packet, req)

from binder_amd import REPO_ROOT
from binder_amd.harness import BinderProcess, NativeZkd
from binder_amd.zkclient import ZkConn

ZKLOGCAT = REPO_ROOT / "bin" / "zklogcat"


def raw_connect(port, timeout_ms=30000, session_id=0,
                passwd=b"\x00" * 16):
    s = socket.socket()
    s.settimeout(5)
    s.connect(("127.0.0.1", port))
    s.sendall(connect_request(timeout_ms=timeout_ms,
                              session_id=session_id, passwd=passwd))
    body = read_packet(s)
    proto, neg, sid = struct.unpack(">iiq", body[:16])
    return s, sid


def test_restart_durability(tmp_path):
    d = tmp_path / "data"
    z = NativeZkd(data_dir=str(d)).start()
    try:
        c = ZkConn("127.0.0.1", z.port)
        c.mkdirp("/com/foo")
        c.create("/com/foo/web", b'{"a": 1}')
        c.set("/com/foo/web", b'{"a": 2}')
        c.close()
    finally:
        z.stop()
    z2 = NativeZkd(data_dir=str(d)).start()
    try:
        assert z2.nodes_restored == 3
        c = ZkConn("127.0.0.1", z2.port)
        assert c.get("/com/foo/web") == b'{"a": 2}'
        assert c.children("/com/foo") == ["web"]
        c.close()
    finally:
        z2.stop()


def test_torn_tail_truncated_on_restart(tmp_path):
    d = tmp_path / "data"
    z = NativeZkd(data_dir=str(d)).start()
    try:
        c = ZkConn("127.0.0.1", z.port)
        c.mkdirp("/a")
        c.create("/a/keep", b"yes")
        c.close()
    finally:
        z.stop()
    # simulate a crash mid-append: garbage tail after valid entries
    log = d / "log.1"
    with open(log, "ab") as f:
        f.write(b"\x00\x01garbage-torn-tail")
    z2 = NativeZkd(data_dir=str(d)).start()
    try:
        c = ZkConn("127.0.0.1", z2.port)
        assert c.get("/a/keep") == b"yes"
        # new writes after the repair must survive ANOTHER restart
        c.create("/a/after", b"ok")
        c.close()
    finally:
        z2.stop()
    z3 = NativeZkd(data_dir=str(d)).start()
    try:
        c = ZkConn("127.0.0.1", z3.port)
        assert c.get("/a/after") == b"ok"
        c.close()
    finally:
        z3.stop()


def test_compaction_bounds_log(tmp_path):
    d = tmp_path / "data"
    z = NativeZkd(data_dir=str(d)).start()
    try:
        c = ZkConn("127.0.0.1", z.port)
        c.create("/hot", b"0")
        for i in range(300):
            c.set("/hot", str(i).encode())
        c.close()
    finally:
        z.stop()
    size_before = (d / "log.1").stat().st_size
    z2 = NativeZkd(data_dir=str(d)).start()  # restart compacts
    z2.stop()
    size_after = (d / "log.1").stat().st_size
    assert size_after < size_before / 4, \
        f"no compaction: {size_before} -> {size_after}"
    z3 = NativeZkd(data_dir=str(d)).start()
    try:
        c = ZkConn("127.0.0.1", z3.port)
        assert c.get("/hot") == b"299"
        c.close()
    finally:
        z3.stop()
    # the compacted log remains valid FileTxnLog v2 for zklogcat
    out = subprocess.run([str(ZKLOGCAT), str(d / "log.1")],
                         capture_output=True, text=True, check=True)
    txns = [json.loads(line) for line in out.stdout.splitlines()]
    assert any(t["type"] == "create" and t["path"] == "/hot"
               for t in txns)


def test_session_timeout_reaps_ephemerals_and_fires_watches():
    z = NativeZkd(session_timeout_ms=800).start()
    try:
        # session A creates an ephemeral (flags=1)
        a, sid_a = raw_connect(z.port, timeout_ms=800)
        a.sendall(req(1, 1, jstr("/eph") + jstr(b"x") + ACL_OPEN +
                      be32(1)))
        xid, zxid, err, rest = parse_reply_header(read_packet(a))
        assert err == 0

        # session B watches it
        b, sid_b = raw_connect(z.port)
        b.sendall(req(1, 3, jstr("/eph") + b"\x01"))  # exists+watch
        xid, zxid, err, rest = parse_reply_header(read_packet(b))
        assert err == 0
        stat, _ = parse_stat(rest)
        assert stat["ephemeralOwner"] == sid_a

        # A's connection dies without close; after the session timeout
        # the ephemeral must be reaped and B's watch must fire
        # DELETED. B keeps ITS session alive by pinging (liveness is
        # packet-based, like real ZooKeeper), so reads interleave ping
        # replies (xid -2) with the notification (xid -1).
        a.close()
        b.settimeout(0.3)
        deadline = time.time() + 8
        body = None
        while time.time() < deadline:
            b.sendall(req(-2, 11))  # ping
            try:
                pkt = read_packet(b)
            except (TimeoutError, socket.timeout):
                continue
            if struct.unpack(">i", pkt[:4])[0] == -1:
                body = pkt
                break
        assert body is not None, "notification never arrived"
        xid, zxid, err, rest = parse_reply_header(body)
        assert xid == -1
        ev_type, ev_state = struct.unpack(">ii", rest[:8])
        assert ev_type == 2  # NodeDeleted
        b.settimeout(5)
        b.sendall(req(2, 3, jstr("/eph") + b"\x00"))
        while True:
            pkt = read_packet(b)
            if struct.unpack(">i", pkt[:4])[0] == 2:
                break
        xid, zxid, err, rest = parse_reply_header(pkt)
        assert err == -101  # ZNONODE
        b.close()
    finally:
        z.stop()


def test_sequential_create():
    z = NativeZkd().start()
    try:
        s, sid = raw_connect(z.port)
        s.sendall(req(1, 1, jstr("/q") + jstr(b"") + ACL_OPEN +
                      be32(0)))
        read_packet(s)
        names = []
        for i in range(3):
            s.sendall(req(10 + i, 1, jstr("/q/n-") + jstr(b"") +
                          ACL_OPEN + be32(2)))  # SEQUENTIAL
            xid, zxid, err, rest = parse_reply_header(read_packet(s))
            assert err == 0
            (plen,) = struct.unpack(">i", rest[:4])
            names.append(rest[4:4 + plen].decode())
        assert names == sorted(names)
        assert len(set(names)) == 3
        for n in names:
            base = n.rsplit("/", 1)[1]
            assert base.startswith("n-") and len(base) == 12, n
            int(base[2:])  # 10-digit numeric suffix
        s.close()
    finally:
        z.stop()


def test_version_checks():
    z = NativeZkd().start()
    try:
        c = ZkConn("127.0.0.1", z.port)
        c.create("/v", b"a")
        # setData with wrong explicit version => ZBADVERSION
        with pytest.raises(Exception):
            c.set("/v", b"b", version=7)
        c.set("/v", b"b", version=0)
        assert c.get("/v") == b"b"
        c.close()
    finally:
        z.stop()


def test_four_letter_words():
    z = NativeZkd().start()
    try:
        for cmd, expect in (("ruok", b"imok"), ("srvr", b"Mode:")):
            s = socket.socket()
            s.settimeout(3)
            s.connect(("127.0.0.1", z.port))
            s.sendall(cmd.encode())
            buf = b""
            while True:
                chunk = s.recv(4096)
                if not chunk:
                    break
                buf += chunk
            assert expect in buf, (cmd, buf)
            s.close()
    finally:
        z.stop()


def test_zklogcat_decodes_zkd_log(tmp_path):
    d = tmp_path / "data"
    z = NativeZkd(data_dir=str(d)).start()
    try:
        c = ZkConn("127.0.0.1", z.port)
        c.create("/x", b"hello")
        c.set("/x", b"world")
        c.delete("/x")
        c.close()
    finally:
        z.stop()
    out = subprocess.run([str(ZKLOGCAT), str(d / "log.1")],
                         capture_output=True, text=True, check=True)
    txns = [json.loads(line) for line in out.stdout.splitlines()]
    types = [t["type"] for t in txns]
    assert "createSession" in types
    assert "create" in types and "setData" in types and \
        "delete" in types
    create = next(t for t in txns if t["type"] == "create")
    assert create["path"] == "/x"


@pytest.mark.timeout(120)
def test_binderd_mirror_over_native_zkd(tmp_path):
    """The full production story on the native registry: binderd
    mirrors zkd, a registrar-style ephemeral registration appears in
    DNS, and vanishes when the registrar's session dies."""
    z = NativeZkd(data_dir=str(tmp_path / "data"),
                  session_timeout_ms=1500).start()
    b = None
    try:
        c = ZkConn("127.0.0.1", z.port)
        c.mkdirp("/com/foo")
        c.create("/com/foo/web", json.dumps(
            {"type": "host", "host": {"address": "10.0.0.9"}}).encode())

        b = BinderProcess(dns_domain="foo.com", store="zk",
                          zk_host="127.0.0.1", zk_port=z.port,
                          workdir=tmp_path,
                          log_path=str(tmp_path / "b.log"))
        b.start()
        r = b.wait_ready("web.foo.com", timeout=20)
        assert r.answers[0]["address"] == "10.0.0.9"

        # registrar-style ephemeral: another client session registers
        # a host and then dies without closing
        reg = ZkConn("127.0.0.1", z.port, session_timeout_ms=1500)
        reg.create("/com/foo/eph", json.dumps(
            {"type": "host", "host": {"address": "10.0.0.77"}}).encode(),
            flags=1)
        r = b.wait_ready("eph.foo.com", timeout=10)
        assert r.answers[0]["address"] == "10.0.0.77"
        reg.sock.close()  # vanish without CloseSession

        deadline = time.time() + 15
        while time.time() < deadline:
            if b.dig("eph.foo.com").status == "REFUSED":
                break
            time.sleep(0.25)
        assert b.dig("eph.foo.com").status == "REFUSED", \
            "ephemeral registration never reaped from DNS"
        # the persistent record is unaffected
        assert b.dig("web.foo.com").answers[0]["address"] == "10.0.0.9"
        c.close()
    finally:
        if b:
            b.stop()
        z.stop()


@pytest.mark.timeout(120)
def test_binderd_survives_zkd_restart(tmp_path):
    """Registry crash+restart: binderd serves stale answers during the
    outage, and when zkd comes back on the SAME data the
    mirror rebuilds (zkd restarts have no live sessions, so clients go
    through the expiry->fresh-session path) and new writes propagate."""
    d = tmp_path / "data"
    port = None
    z = NativeZkd(data_dir=str(d)).start()
    port = z.port
    b = None
    try:
        c = ZkConn("127.0.0.1", z.port)
        c.mkdirp("/com/foo")
        c.create("/com/foo/web", json.dumps(
            {"type": "host", "host": {"address": "10.0.0.9"}}).encode())
        c.close()

        b = BinderProcess(dns_domain="foo.com", store="zk",
                          zk_host="127.0.0.1", zk_port=z.port,
                          workdir=tmp_path,
                          log_path=str(tmp_path / "b.log"))
        b.start()
        b.wait_ready("web.foo.com", timeout=20)

        # hard-kill the registry
        z.proc.kill()
        z.proc.wait()
        time.sleep(0.5)
        # stale-serve during the outage
        assert b.dig("web.foo.com").answers[0]["address"] == "10.0.0.9"

        # restart on the same data and the SAME port
        z = NativeZkd(port=port, data_dir=str(d)).start()
        assert z.nodes_restored == 3
        # new write must reach the (rebuilt) mirror
        c = ZkConn("127.0.0.1", z.port)
        c.create("/com/foo/neu", json.dumps(
            {"type": "host",
             "host": {"address": "10.0.0.10"}}).encode())
        r = b.wait_ready("neu.foo.com", timeout=40)
        assert r.answers[0]["address"] == "10.0.0.10"
        assert b.dig("web.foo.com").answers[0]["address"] == "10.0.0.9"
        c.close()
    finally:
        if b:
            b.stop()
        z.stop()


@pytest.mark.timeout(120)
def test_mirror_converges_under_zkd_churn(tmp_path):
    """Sustained mutations against the native registry: the binderd
    mirror must converge to the final addresses (the stubzk churn
    suite's semantics, against zkd)."""
    z = NativeZkd().start()
    b = None
    try:
        c = ZkConn("127.0.0.1", z.port)
        c.mkdirp("/com/foo")
        n_hosts = 300
        for i in range(n_hosts):
            c.create(f"/com/foo/h{i}", json.dumps(
                {"type": "host",
                 "host": {"address": f"10.1.{i // 250}.{i % 250}"}}
            ).encode())

        b = BinderProcess(dns_domain="foo.com", store="zk",
                          zk_host="127.0.0.1", zk_port=z.port,
                          workdir=tmp_path)
        b.start()
        b.wait_ready(f"h{n_hosts - 1}.foo.com", timeout=30)

        # churn: rewrite every host several times, last write wins
        import random
        rng = random.Random(3)
        final = {}
        for _ in range(4 * n_hosts):
            i = rng.randrange(n_hosts)
            addr = f"10.9.{rng.randrange(250)}.{rng.randrange(1, 250)}"
            c.set(f"/com/foo/h{i}", json.dumps(
                {"type": "host", "host": {"address": addr}}).encode())
            final[i] = addr

        deadline = time.time() + 30
        pendingkeys = sorted(final)
        while time.time() < deadline and pendingkeys:
            still = []
            for i in pendingkeys:
                r = b.dig(f"h{i}.foo.com")
                if not (r.status == "NOERROR" and
                        r.answers[0]["address"] == final[i]):
                    still.append(i)
            pendingkeys = still
            if pendingkeys:
                time.sleep(0.3)
        assert not pendingkeys, \
            f"{len(pendingkeys)} hosts never converged"
        c.close()
    finally:
        if b:
            b.stop()
        z.stop()
