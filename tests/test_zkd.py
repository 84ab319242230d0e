"""zkd: the built-in durable single-node registry.

Restart durability via txn-log replay; ephemeral registrations do NOT
survive a restart (their sessions are gone — real-ZK semantics); a
binderd can serve from it across the restart.
"""
import json
import subprocess
import sys
import time

import pytest

from binder_amd.harness import BinderProcess, free_port
from binder_amd.stubzk import StubZk
from binder_amd.zkclient import ZkConn


def test_replay_restores_tree(tmp_path):
    data = tmp_path / "zkdata"
    zk = StubZk(txnlog_dir=str(data)).start()
    zk.mkdirp("/com/foo")
    zk.put("/com/foo/web", json.dumps(
        {"type": "host", "host": {"address": "1.2.3.4"}}).encode())
    zk.put("/com/foo/web", json.dumps(
        {"type": "host", "host": {"address": "1.2.3.5"}}).encode())
    zk.put("/com/foo/tmp", b"null")
    zk.delete("/com/foo/tmp")
    # ephemeral via wire session
    reg = ZkConn("127.0.0.1", zk.port)
    reg.create("/com/foo/eph", b'{"type":"host","host":{}}', flags=1)
    reg.close()
    zk.stop()

    # restart from the same log
    zk2 = StubZk(txnlog_dir=str(data)).start()
    try:
        assert zk2.get("/com/foo/web") == json.dumps(
            {"type": "host", "host": {"address": "1.2.3.5"}}).encode()
        assert not zk2.exists("/com/foo/tmp")
        assert not zk2.exists("/com/foo/eph")  # ephemerals don't survive
        assert "web" in zk2.children("/com/foo")
        # and it continues journaling after restart
        zk2.put("/com/foo/more", b"null")
    finally:
        zk2.stop()
    zk3 = StubZk(txnlog_dir=str(data)).start()
    try:
        assert zk3.exists("/com/foo/more")
    finally:
        zk3.stop()


@pytest.mark.timeout(120)
def test_zkd_daemon_serves_binderd(tmp_path):
    data = tmp_path / "zkdata"
    port = free_port()
    proc = subprocess.Popen(
        [sys.executable, "-m", "binder_amd", "zkd", "-p", str(port),
         "-d", str(data)],
        stdout=subprocess.PIPE, text=True)
    try:
        deadline = time.time() + 15
        while time.time() < deadline:
            try:
                with ZkConn("127.0.0.1", port, timeout=1) as c:
                    c.mkdirp("/com/foo")
                    c.mkdirp("/com/foo/web")
                    c.set("/com/foo/web", json.dumps(
                        {"type": "host",
                         "host": {"address": "9.9.9.9"}}).encode())
                break
            except OSError:
                time.sleep(0.2)
        else:
            pytest.fail("zkd never came up")

        srv = BinderProcess(store="zk", zk_host="127.0.0.1",
                            zk_port=port, workdir=tmp_path)
        srv.start()
        try:
            r = srv.wait_ready("web.foo.com")
            assert r.answers[0]["address"] == "9.9.9.9"
        finally:
            srv.stop()
    finally:
        proc.terminate()
        proc.wait(timeout=10)


def test_replay_survives_corrupt_tail(tmp_path):
    """Crash mid-append: the log ends in a truncated/corrupt entry.
    Restart must recover everything before the corruption, truncate the
    bad tail, and KEEP journaling durably — entries written after the
    recovery must survive the next restart (without truncation they
    would be appended after the garbage and lost)."""
    data = tmp_path / "zkdata"
    zk = StubZk(txnlog_dir=str(data)).start()
    zk.mkdirp("/com/foo")
    zk.put("/com/foo/a", b'{"type":"host","host":{"address":"1.1.1.1"}}')
    zk.put("/com/foo/b", b'{"type":"host","host":{"address":"2.2.2.2"}}')
    zk.stop()

    log = data / "log.1"
    with open(log, "ab") as f:
        f.write(b"\x00\x00\x00\x07\xde\xad")  # truncated garbage entry

    zk2 = StubZk(txnlog_dir=str(data)).start()
    try:
        assert zk2.exists("/com/foo/a")
        assert zk2.exists("/com/foo/b")
        zk2.put("/com/foo/c", b"null")  # journaled after recovery
    finally:
        zk2.stop()

    zk3 = StubZk(txnlog_dir=str(data)).start()
    try:
        assert zk3.exists("/com/foo/c"), "post-recovery write lost"
        assert zk3.exists("/com/foo/b")
    finally:
        zk3.stop()


def test_replay_stops_at_bad_crc(tmp_path):
    """A flipped bit inside an entry's payload: replay must stop at
    that entry (adler32 mismatch) rather than apply corrupt data."""
    import struct as _s
    data = tmp_path / "zkdata"
    zk = StubZk(txnlog_dir=str(data)).start()
    zk.mkdirp("/com/foo")
    zk.put("/com/foo/good", b"null")
    zk.put("/com/foo/late", b"null")
    zk.stop()

    log = data / "log.1"
    raw = bytearray(log.read_bytes())
    # find the LAST entry and flip a payload bit: walk entries
    off = 16
    last_payload = None
    while off + 12 <= len(raw):
        crc, tlen = _s.unpack_from(">qi", raw, off)
        if crc == 0 or tlen <= 0 or off + 12 + tlen + 1 > len(raw):
            break
        last_payload = off + 12
        off += 12 + tlen + 1
    assert last_payload is not None
    raw[last_payload + 20] ^= 0xFF
    log.write_bytes(bytes(raw))

    zk2 = StubZk(txnlog_dir=str(data)).start()
    try:
        assert zk2.exists("/com/foo/good")
        # the corrupted trailing entry was dropped (we can't know which
        # op it held; only that the server is consistent and serving)
        zk2.put("/com/foo/after", b"null")
    finally:
        zk2.stop()
    zk3 = StubZk(txnlog_dir=str(data)).start()
    try:
        assert zk3.exists("/com/foo/after")
    finally:
        zk3.stop()


def test_log_compaction_on_restart(tmp_path):
    """History-dominated logs are compacted at restart into one create
    per live node (still valid FileTxnLog v2); state is preserved and
    journaling continues."""
    data = tmp_path / "zkdata"
    zk = StubZk(txnlog_dir=str(data)).start()
    zk.mkdirp("/com/foo")
    for i in range(500):
        zk.put("/com/foo/hot", json.dumps(
            {"type": "host",
             "host": {"address": f"10.0.0.{i % 250}"}}).encode())
    zk.put("/com/foo/cold", b"null")
    zk.stop()
    big = (data / "log.1").stat().st_size

    zk2 = StubZk(txnlog_dir=str(data)).start()
    try:
        small = (data / "log.1").stat().st_size
        assert small < big / 5, (big, small)
        assert zk2.get("/com/foo/hot").endswith(b'"10.0.0.249"}}')
        assert zk2.exists("/com/foo/cold")
        zk2.put("/com/foo/after", b"null")
    finally:
        zk2.stop()

    # compacted log is valid FileTxnLog: zklogcat decodes it and a
    # third restart still sees everything
    from binder_amd import REPO_ROOT
    out = subprocess.run(
        [str(REPO_ROOT / "bin" / "zklogcat"), str(data / "log.1")],
        capture_output=True, text=True, check=True)
    types = {json.loads(l)["type"] for l in out.stdout.splitlines()
             if l.strip()}
    assert types <= {"create", "setData", "delete"}
    zk3 = StubZk(txnlog_dir=str(data)).start()
    try:
        assert zk3.exists("/com/foo/after")
        assert zk3.exists("/com/foo/hot")
    finally:
        zk3.stop()
