"""CLI subcommands + Python ZK client against the stub server."""
import json

import pytest

from binder_amd import cli
from binder_amd.stubzk import StubZk
from binder_amd.zkclient import ZkConn, ZkError


@pytest.fixture()
def zk():
    z = StubZk().start()
    yield z
    z.stop()


def test_zkclient_crud(zk):
    with ZkConn("127.0.0.1", zk.port) as c:
        c.mkdirp("/com/foo/bar")
        assert c.exists("/com/foo/bar")
        c.set("/com/foo/bar", b'{"x":1}')
        assert c.get("/com/foo/bar") == b'{"x":1}'
        c.create("/com/foo/baz", b"null")
        assert sorted(c.children("/com/foo")) == ["bar", "baz"]
        c.rmr("/com")
        assert not c.exists("/com")


def test_zkclient_errors(zk):
    with ZkConn("127.0.0.1", zk.port) as c:
        with pytest.raises(ZkError) as e:
            c.get("/nope")
        assert e.value.code == -101
        c.create("/a")
        with pytest.raises(ZkError) as e:
            c.create("/a")
        assert e.value.code == -110


def test_register_then_resolve(zk, tmp_path):
    """`binder-amd register` writes the registrar layout; a binderd
    mirror must then serve the _dns._udp SRV + A for it."""
    rc = cli.main(["register", "binder.coal.foo.com", "10.77.77.1",
                   "-i", "binder0", "--zk-host", "127.0.0.1",
                   "--zk-port", str(zk.port)])
    assert rc == 0
    data = json.loads(zk.get("/com/foo/coal/binder").decode())
    assert data["type"] == "service"
    assert data["service"]["srvce"] == "_dns"

    from binder_amd.harness import BinderProcess
    srv = BinderProcess(store="zk", zk_host="127.0.0.1",
                        zk_port=zk.port, workdir=tmp_path)
    srv.start()
    try:
        r = srv.wait_ready("_dns._udp.binder.coal.foo.com", qtype="SRV")
        assert r.answers[0]["port"] == 53
        assert r["additionals"][0]["address"] == "10.77.77.1"
        r = srv.dig("binder0.binder.coal.foo.com")
        assert r.answers[0]["address"] == "10.77.77.1"
    finally:
        srv.stop()


def test_cli_zk_ops(zk, capsys):
    cli.main(["zk", "mkdirp", "/x/y", "--zk-port", str(zk.port)])
    cli.main(["zk", "set", "/x/y", '{"k":1}', "--zk-port", str(zk.port)])
    cli.main(["zk", "get", "/x/y", "--zk-port", str(zk.port)])
    out = capsys.readouterr().out
    assert '{"k":1}' in out
    cli.main(["zk", "ls", "/x", "--zk-port", str(zk.port)])
    assert "y" in capsys.readouterr().out


def test_cli_dig_and_balstat_and_status(tmp_path, capsys):
    """The operator CLI verbs end-to-end: dig against binderd, balstat
    against a live balancer, status against a supervisor state dir."""
    import json as _json
    import os
    import subprocess
    import time

    from binder_amd.harness import BALANCERD, BinderProcess, free_port

    sockdir = tmp_path / "socks"
    sockdir.mkdir()
    store = tmp_path / "tree.json"
    store.write_text(_json.dumps(
        {"foo.com": None,
         "web.foo.com": {"type": "host",
                         "host": {"address": "1.2.3.4"}}}))
    srv = BinderProcess(store=f"file:{store}", workdir=tmp_path,
                        balancer_socket=str(sockdir / "b0"))
    srv.start()
    stats = tmp_path / "stats.sock"
    port = free_port()
    bal = subprocess.Popen(
        [str(BALANCERD), "-p", str(port), "-H", "127.0.0.1",
         "-s", str(sockdir), "-S", str(stats), "-r", "100"],
        env=dict(os.environ, LOG_LEVEL="warn"),
        stdout=subprocess.DEVNULL, stderr=subprocess.STDOUT)
    try:
        deadline = time.time() + 15
        while time.time() < deadline and not stats.exists():
            time.sleep(0.1)
        time.sleep(0.5)

        assert cli.main(["dig", "web.foo.com", "-s", "127.0.0.1",
                         "-p", str(srv.port)]) == 0
        out = capsys.readouterr().out
        assert "NOERROR" in out and "1.2.3.4" in out

        assert cli.main(["dig", "web.foo.com", "A", "-s", "127.0.0.1",
                         "-p", str(port), "--tcp"]) == 0
        assert "NOERROR" in capsys.readouterr().out

        assert cli.main(["balstat", str(stats)]) == 0
        out = capsys.readouterr().out
        assert "PATH" in out and "b0" in out
        assert cli.main(["balstat", str(stats), "--json"]) == 0
        st = _json.loads(capsys.readouterr().out)
        assert st["backends"][0]["ok"] is True

        # status against a fabricated state dir
        state = tmp_path / "state"
        state.mkdir()
        (state / "status.json").write_text(_json.dumps(
            {"instances": {"binder-5301": {
                "pid": 42, "state": "online", "restarts": 0,
                "since": 0, "port": 5301}}}))
        assert cli.main(["status", "-d", str(state)]) == 0
        out = capsys.readouterr().out
        assert "online" in out and "binder-5301" in out
    finally:
        bal.terminate()
        bal.wait(timeout=5)
        srv.stop()


def test_register_hold_is_ephemeral(zk, tmp_path):
    """`register --hold` keeps an ephemeral registration alive via
    session pings; killing the holder deregisters it."""
    import json as _json
    import signal
    import subprocess
    import sys
    import time

    proc = subprocess.Popen(
        [sys.executable, "-m", "binder_amd", "register",
         "api.coal.foo.com", "10.77.0.9", "-i", "h0", "--hold",
         "--zk-host", "127.0.0.1", "--zk-port", str(zk.port)],
        stdout=subprocess.PIPE, text=True)
    try:
        deadline = time.time() + 10
        while time.time() < deadline:
            if zk.exists("/com/foo/coal/api/h0"):
                break
            time.sleep(0.1)
        else:
            pytest.fail("ephemeral registration never appeared")
        data = _json.loads(zk.get("/com/foo/coal/api/h0").decode())
        assert data["rr_host"]["address"] == "10.77.0.9"
        # holder dies => session closes => node reaped... our stub
        # reaps on session close/expiry; SIGKILL leaves the session
        # until timeout, so use expire to model it deterministically
        proc.send_signal(signal.SIGKILL)
        proc.wait(timeout=5)
        # TCP close makes the stub drop the conn; expire the sessions
        # (as ZK would at session timeout) and the node must go
        zk.expire_sessions()
        deadline = time.time() + 5
        while time.time() < deadline:
            if not zk.exists("/com/foo/coal/api/h0"):
                break
            time.sleep(0.1)
        else:
            pytest.fail("ephemeral registration survived expiry")
    finally:
        if proc.poll() is None:
            proc.kill()
