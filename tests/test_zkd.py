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
