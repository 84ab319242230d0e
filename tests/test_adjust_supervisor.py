"""binder-adjust + binder-supervisor convergence tests.

Mirrors src/smf_adjust.c semantics: plan/diff/converge
instance set, idempotency (no-op when nothing changed), scale up/down,
crash restart, and the metric-ports registry
(smf/methods/metric-ports-updater.sh equivalent).
"""
import json
import os
import signal
import subprocess
import time
from pathlib import Path

import pytest

from binder_amd import REPO_ROOT
from binder_amd.digclient import dig
from binder_amd.harness import ADJUST, SUPERVISORD

BIN = REPO_ROOT / "bin"


def run_adjust(statedir, count, base_port, cfg, tree, wait=0):
    cmd = [str(ADJUST), "-i", str(count),
           "-B", str(base_port), "-d", str(statedir),
           "-f", str(cfg), "-S", f"file:{tree}"]
    if wait:
        cmd += ["-w", str(wait)]
    return subprocess.run(cmd, capture_output=True, text=True)


def read_status(statedir):
    p = Path(statedir) / "status.json"
    if not p.exists():
        return {}
    return json.loads(p.read_text()).get("instances", {})


@pytest.fixture()
def sup(tmp_path):
    statedir = tmp_path / "state"
    statedir.mkdir()
    tree = tmp_path / "tree.json"
    tree.write_text(json.dumps({
        "foo.com": None,
        "web.foo.com": {"type": "host", "host": {"address": "1.2.3.4"}},
    }))
    # shared config: metricsPort must differ per instance; binderd
    # derives it as port+1000 when not pinned, so leave it unset
    cfg = tmp_path / "binder.json"
    cfg.write_text(json.dumps({
        "dnsDomain": "foo.com", "datacenterName": "coal",
        "host": "127.0.0.1"}))
    proc = subprocess.Popen(
        [str(SUPERVISORD), "-d", str(statedir),
         "-x", str(BIN / "binderd")],
        env=dict(os.environ, LOG_LEVEL="info"),
        stdout=open(tmp_path / "sup.log", "ab"),
        stderr=subprocess.STDOUT)
    time.sleep(0.3)
    yield {"statedir": statedir, "cfg": cfg, "tree": tree,
           "proc": proc, "tmp": tmp_path}
    proc.terminate()
    try:
        proc.wait(timeout=5)
    except subprocess.TimeoutExpired:
        proc.kill()


BASE = 25801


def wait_online(statedir, names, timeout=15):
    deadline = time.time() + timeout
    while time.time() < deadline:
        st = read_status(statedir)
        if all(st.get(n, {}).get("state") == "online" for n in names):
            return st
        time.sleep(0.2)
    raise TimeoutError(f"instances never online: {read_status(statedir)}")


def test_scale_up_down_and_idempotency(sup):
    statedir, cfg, tree = sup["statedir"], sup["cfg"], sup["tree"]
    # converge to 3
    r = run_adjust(statedir, 3, BASE, cfg, tree, wait=20)
    assert r.returncode == 0, r.stderr
    names = [f"binder-{BASE + i}" for i in range(3)]
    st = wait_online(statedir, names)
    assert len(st) == 3

    # each instance actually serves DNS on its port
    for i in range(3):
        resp = dig("web.foo.com", port=BASE + i, timeout=3)
        assert resp.status == "NOERROR"

    # idempotent: run again, instance files untouched (same pids)
    pids = {n: st[n]["pid"] for n in names}
    r = run_adjust(statedir, 3, BASE, cfg, tree)
    assert r.returncode == 0
    assert '"unchanged":3' in r.stderr or '"unchanged":3' in r.stdout
    time.sleep(1.5)
    st2 = read_status(statedir)
    assert {n: st2[n]["pid"] for n in names} == pids, "restarted on no-op"

    # scale down to 1: extra instances drained
    r = run_adjust(statedir, 1, BASE, cfg, tree)
    assert r.returncode == 0
    deadline = time.time() + 10
    while time.time() < deadline:
        st3 = read_status(statedir)
        if len(st3) == 1 and st3.get(names[0], {}).get("state") == \
                "online":
            break
        time.sleep(0.2)
    else:
        pytest.fail(f"scale-down never converged: {read_status(statedir)}")


def test_crash_restart_with_backoff(sup):
    statedir, cfg, tree = sup["statedir"], sup["cfg"], sup["tree"]
    assert run_adjust(statedir, 1, BASE + 50, cfg, tree,
                      wait=20).returncode == 0
    name = f"binder-{BASE + 50}"
    st = wait_online(statedir, [name])
    pid = st[name]["pid"]
    os.kill(pid, signal.SIGKILL)
    deadline = time.time() + 15
    while time.time() < deadline:
        st = read_status(statedir)
        if st.get(name, {}).get("state") == "online" and \
                st[name]["pid"] != pid:
            break
        time.sleep(0.2)
    else:
        pytest.fail("instance never restarted after crash")
    assert st[name]["restarts"] >= 1


def test_metric_ports_registry(sup):
    statedir, cfg, tree = sup["statedir"], sup["cfg"], sup["tree"]
    assert run_adjust(statedir, 2, BASE + 60, cfg, tree,
                      wait=20).returncode == 0
    names = [f"binder-{BASE + 60 + i}" for i in range(2)]
    wait_online(statedir, names)
    time.sleep(1.5)  # let the registry tick
    ports = (Path(statedir) / "metric_ports").read_text().strip()
    got = set(ports.split(","))
    assert got == {str(BASE + 60 + 1000), str(BASE + 61 + 1000)}


def test_balancer_sockets_created(sup):
    """Instances get per-port socket paths under <dir>/sockets — the
    balancer discovers them (smf_adjust.c:44 socket_path contract)."""
    statedir, cfg, tree = sup["statedir"], sup["cfg"], sup["tree"]
    assert run_adjust(statedir, 2, BASE + 70, cfg, tree,
                      wait=20).returncode == 0
    wait_online(statedir, [f"binder-{BASE + 70 + i}" for i in range(2)])
    sockdir = Path(statedir) / "sockets"
    deadline = time.time() + 5
    while time.time() < deadline:
        socks = sorted(p.name for p in sockdir.iterdir())
        if socks == [str(BASE + 70), str(BASE + 71)]:
            return
        time.sleep(0.2)
    pytest.fail(f"sockets missing: {list(sockdir.iterdir())}")


def test_supervisor_sighup_rescans(sup):
    """SIGHUP triggers an immediate rescan (faster than the 1s tick is
    hard to assert; assert the instance appears and serves)."""
    statedir, cfg, tree = sup["statedir"], sup["cfg"], sup["tree"]
    port = BASE + 90
    inst = {"port": port, "instance": port, "enabled": True,
            "config_file": str(cfg), "store": f"file:{tree}",
            "socket_path": str(statedir / "sockets" / str(port))}
    (statedir / "instances").mkdir(exist_ok=True)
    (statedir / "sockets").mkdir(exist_ok=True)
    (statedir / "instances" / f"binder-{port}.json").write_text(
        json.dumps(inst))
    sup["proc"].send_signal(signal.SIGHUP)
    wait_online(statedir, [f"binder-{port}"])
    assert dig("web.foo.com", port=port, timeout=3).status == "NOERROR"


def test_full_scale_32_instances(sup):
    """The reference caps at 32 processes per zone (boot/setup.sh:15);
    converge to the cap, verify serving, then to zero."""
    statedir, cfg, tree = sup["statedir"], sup["cfg"], sup["tree"]
    base = BASE + 100
    r = run_adjust(statedir, 32, base, cfg, tree, wait=60)
    assert r.returncode == 0, r.stderr
    names = [f"binder-{base + i}" for i in range(32)]
    st = wait_online(statedir, names, timeout=60)
    assert len(st) == 32
    # spot-check serving across the range
    for port in (base, base + 15, base + 31):
        assert dig("web.foo.com", port=port, timeout=3).status == \
            "NOERROR"
    # beyond the cap is rejected (smf_adjust.c:904-909 bounds)
    r = run_adjust(statedir, 33, base, cfg, tree)
    assert r.returncode != 0
    # converge to zero: all instances drained
    r = run_adjust(statedir, 0, base, cfg, tree)
    assert r.returncode == 0
    deadline = time.time() + 30
    while time.time() < deadline:
        if not read_status(statedir):
            break
        time.sleep(0.3)
    else:
        pytest.fail(f"instances left: {read_status(statedir)}")
