"""Production-shaped end-to-end: the whole framework wired together.

StubZk <- 3x binderd (zk mirror, via supervisor+adjust) <- balancer
with: registrar-style service registration, crash recovery, drain on
scale-down, and the operator CLI — the full SURVEY.md §3.1/§3.6 story
on Linux-native pieces.
"""
import json
import os
import socket
import subprocess
import time

import pytest

from binder_amd import REPO_ROOT, cli
from binder_amd.digclient import dig
from binder_amd.harness import free_port, BALANCERD, SUPERVISORD, ADJUST
from binder_amd.stubzk import StubZk
from binder_amd.zkclient import ZkConn

BIN = REPO_ROOT / "bin"
BASE = 26401


def balstat(path):
    with socket.socket(socket.AF_UNIX) as s:
        s.settimeout(2)
        s.connect(str(path))
        return json.loads(s.recv(1 << 20).decode())


@pytest.mark.timeout(180)
def test_full_stack(tmp_path):
    zk = StubZk().start()
    statedir = tmp_path / "state"
    statedir.mkdir()
    procs = []
    try:
        zk.mkdirp("/com/foo")

        # shared binderd config (zk store)
        cfg = tmp_path / "binder.json"
        cfg.write_text(json.dumps({
            "dnsDomain": "foo.com", "datacenterName": "coal",
            "host": "127.0.0.1"}))

        # supervisor + adjust converge 3 instances
        sup = subprocess.Popen(
            [str(SUPERVISORD), "-d", str(statedir),
             "-x", str(BIN / "binderd")],
            env=dict(os.environ, LOG_LEVEL="warn",
                     ZK_HOST="127.0.0.1", ZK_PORT=str(zk.port)),
            stdout=open(tmp_path / "sup.log", "ab"),
            stderr=subprocess.STDOUT)
        procs.append(sup)
        rc = subprocess.run(
            [str(ADJUST), "-i", "3", "-B", str(BASE),
             "-d", str(statedir), "-f", str(cfg), "-S", "zk",
             "-w", "30"],
            capture_output=True, text=True)
        assert rc.returncode == 0, rc.stderr

        # balancer fronts the socket dir
        bport = free_port()
        stats = tmp_path / "stats.sock"
        bal = subprocess.Popen(
            [str(BALANCERD), "-p", str(bport),
             "-H", "127.0.0.1", "-s", str(statedir / "sockets"),
             "-S", str(stats), "-r", "100"],
            env=dict(os.environ, LOG_LEVEL="warn"),
            stdout=open(tmp_path / "bal.log", "ab"),
            stderr=subprocess.STDOUT)
        procs.append(bal)

        deadline = time.time() + 15
        while time.time() < deadline:
            try:
                if sum(1 for b in balstat(stats)["backends"]
                       if b["ok"]) == 3:
                    break
            except (OSError, ValueError):
                pass
            time.sleep(0.2)
        else:
            pytest.fail("balancer never saw 3 healthy backends")

        # registrar-style registration through the CLI
        assert cli.main(["register", "api.coal.foo.com", "10.88.0.1",
                         "-i", "api0", "-p", "8080",
                         "--zk-host", "127.0.0.1",
                         "--zk-port", str(zk.port)]) == 0

        # ... plus a raw ephemeral host like a real registrar would
        reg = ZkConn("127.0.0.1", zk.port)
        reg.mkdirp("/com/foo/coal")
        reg.create("/com/foo/coal/worker", json.dumps(
            {"type": "host", "host": {"address": "10.88.0.2"}}).encode(),
            flags=1)

        # resolution through the balancer
        deadline = time.time() + 15
        while time.time() < deadline:
            r = dig("api0.api.coal.foo.com", port=bport, timeout=1)
            if r.status == "NOERROR":
                break
            time.sleep(0.2)
        assert r.answers[0]["address"] == "10.88.0.1"
        r = dig("_dns._udp.api.coal.foo.com", "SRV", port=bport,
                timeout=2)
        assert r.status == "NOERROR"
        assert r.answers[0]["port"] == 8080
        r = dig("worker.coal.foo.com", port=bport, timeout=2)
        assert r.answers[0]["address"] == "10.88.0.2"

        # crash a backend: supervisor restarts it; service continues
        st = json.loads((statedir / "status.json").read_text())
        victim = st["instances"][f"binder-{BASE}"]["pid"]
        os.kill(victim, 9)
        # A query in flight at the crash instant can land in the dying
        # process's socket buffer before the kernel closes its fds and
        # is legitimately lost (UDP semantics; the reference loses it
        # identically) - allow at most one such loss, then service
        # must be continuous.
        lost = 0
        for i in range(30):
            try:
                assert dig("worker.coal.foo.com", port=bport,
                           timeout=2).status == "NOERROR"
            except (TimeoutError, OSError):
                lost += 1
                assert i == 0 and lost <= 1, \
                    f"query lost outside the crash instant (i={i})"
        deadline = time.time() + 20
        while time.time() < deadline:
            st = json.loads((statedir / "status.json").read_text())
            inst = st["instances"][f"binder-{BASE}"]
            if inst["state"] == "online" and inst["pid"] != victim:
                break
            time.sleep(0.3)
        else:
            pytest.fail("crashed instance never came back")

        # ephemeral registration vanishes when the registrar dies
        zk.expire_session(reg.session_id)
        deadline = time.time() + 10
        while time.time() < deadline:
            if dig("worker.coal.foo.com", port=bport,
                   timeout=2).status == "REFUSED":
                break
            time.sleep(0.2)
        else:
            pytest.fail("dead registrar's node still served")

        # scale down to 1: drained instances leave the balancer
        rc = subprocess.run(
            [str(ADJUST), "-i", "1", "-B", str(BASE),
             "-d", str(statedir), "-f", str(cfg), "-S", "zk"],
            capture_output=True, text=True)
        assert rc.returncode == 0
        deadline = time.time() + 20
        while time.time() < deadline:
            try:
                if len(balstat(stats)["backends"]) == 1:
                    break
            except (OSError, ValueError):
                pass
            time.sleep(0.3)
        else:
            pytest.fail("balancer kept drained backends")
        assert dig("api0.api.coal.foo.com", port=bport,
                   timeout=2).status == "NOERROR"
    finally:
        for p in reversed(procs):
            p.terminate()
        for p in procs:
            try:
                p.wait(timeout=10)
            except subprocess.TimeoutExpired:
                p.kill()
        zk.stop()


@pytest.mark.timeout(120)
def test_sdc_single_process_flavor(tmp_path):
    """The Triton/SDC deployment flavor (reference boot/setup.sh:146-178):
    deploy/setup.sh FLAVOR=sdc boots ONE binderd on $PORT, writes the
    single metric port (port+1000, the `mdata-put metricPorts 1053`
    equivalent), and self-registers in the registry as an rr_host +
    _dns._udp SRV service — then answers for its own domain."""
    import urllib.request

    zk = StubZk().start()
    statedir = tmp_path / "state"
    statedir.mkdir()
    port = free_port()
    cfg = tmp_path / "binder.json"
    cfg.write_text(json.dumps({
        "dnsDomain": "foo.com", "datacenterName": "coal",
        "host": "127.0.0.1", "store": "zk",
        "metricsPort": free_port()}))
    pid = None
    try:
        zk.mkdirp("/com/foo")
        env = dict(os.environ,
                   FLAVOR="sdc", PORT=str(port),
                   STATEDIR=str(statedir), PREFIX=str(REPO_ROOT),
                   CONFIG=str(cfg), REGISTER_ADDR="127.0.0.1",
                   ZK_HOST="127.0.0.1", ZK_PORT=str(zk.port),
                   LOG_LEVEL="info")
        rc = subprocess.run(
            ["sh", str(REPO_ROOT / "deploy" / "setup.sh")],
            env=env, cwd=str(REPO_ROOT), capture_output=True, text=True,
            timeout=60)
        assert rc.returncode == 0, rc.stdout + rc.stderr

        # single metric port written for cmon-agent discovery
        mp = (statedir / "metric_ports").read_text().strip()
        assert mp == str(port + 1000)
        pid = int((statedir / "binderd.pid").read_text().strip())

        # self-registration: binder answers A + SRV for its own domain
        deadline = time.time() + 20
        r = None
        while time.time() < deadline:
            try:
                r = dig("binder.coal.foo.com", "A", server="127.0.0.1",
                        port=port, timeout=0.5)
                if r.status == "NOERROR":
                    break
            except (socket.timeout, OSError):
                pass
            time.sleep(0.2)
        assert r is not None and r.status == "NOERROR", \
            (statedir / "log" / "binder.log").read_text()[-2000:]
        assert r.answers[0]["address"] == "127.0.0.1"
        rs = dig("_dns._udp.binder.coal.foo.com", "SRV",
                 server="127.0.0.1", port=port, timeout=2)
        assert rs.status == "NOERROR"
        assert any(a.get("port") == port for a in rs.answers)

        # metrics exposition is up on the configured port
        mport = json.loads(cfg.read_text())["metricsPort"]
        with urllib.request.urlopen(
                f"http://127.0.0.1:{mport}/metrics", timeout=3) as resp:
            assert b"binder_requests_completed" in resp.read()
    finally:
        if pid is not None:
            try:
                os.kill(pid, 15)
            except ProcessLookupError:
                pass
        zk.stop()
