"""Full-stack verification tier (runs on the driver's real box).

This project has no GPU compute (BASELINE.json north_star: the
reference is a Node.js DNS server); the `gpu` marker here selects the
heavier full-stack integration checks the driver runs on real hardware
at round end: build artifacts present, stub-ZK -> mirror -> balancer ->
load-generator pipeline, and a short sustained-load run.
"""
import json
import os
import subprocess
import time

import pytest

from binder_amd import REPO_ROOT, require_native
from binder_amd.harness import BinderProcess, free_port, BALANCERD
from binder_amd.stubzk import StubZk

pytestmark = pytest.mark.gpu


def test_native_artifacts_present():
    n = require_native()
    assert n.domain_to_path("foo.com") == "/com/foo"
    for exe in ("binderd", "binder-balancer", "dnsblast"):
        assert (REPO_ROOT / "bin" / exe).exists(), exe


def test_full_stack_sustained_load(tmp_path):
    zk = StubZk().start()
    try:
        zk.mkdirp("/com/foo")
        names = []
        for i in range(1000):
            zk.put(f"/com/foo/h{i}".encode().decode(), json.dumps(
                {"type": "host",
                 "host": {"address": f"10.1.{i // 250}.{i % 250}"}}
            ).encode())
            names.append(f"h{i}.foo.com A")
        (tmp_path / "names.txt").write_text("\n".join(names))

        sockdir = tmp_path / "socks"
        sockdir.mkdir()
        b = BinderProcess(store="zk", zk_host="127.0.0.1",
                          zk_port=zk.port, workdir=tmp_path,
                          log_level="warn",
                          balancer_socket=str(sockdir / "b0"),
                          log_path=str(tmp_path / "b0.log"))
        b.start()
        try:
            b.wait_ready("h999.foo.com", timeout=30)
            port = free_port()
            bal = subprocess.Popen(
                [str(BALANCERD),
                 "-p", str(port), "-H", "127.0.0.1",
                 "-s", str(sockdir), "-r", "100"],
                env=dict(os.environ, LOG_LEVEL="warn"),
                stdout=subprocess.DEVNULL, stderr=subprocess.STDOUT)
            try:
                time.sleep(0.5)
                out = subprocess.run(
                    [str(REPO_ROOT / "bin" / "dnsblast"),
                     "-s", "127.0.0.1", "-p", str(port),
                     "-n", "50000", "-c", "32", "-t", "2",
                     "-f", str(tmp_path / "names.txt")],
                    capture_output=True, text=True, timeout=60,
                    check=True)
                stats = json.loads(out.stdout.strip())
                assert stats["received"] >= 49500
                assert stats["noerror"] == stats["received"]
                assert stats["qps"] > 5000, stats
            finally:
                bal.terminate()
                bal.wait(timeout=5)
        finally:
            b.stop()
    finally:
        zk.stop()


def test_bench_entrypoint_quick():
    out = subprocess.run(
        ["python3", str(REPO_ROOT / "bench.py"), "--steps", "1",
         "--warmup", "0", "--queries-per-proc", "30000"],
        capture_output=True, text=True, timeout=300, cwd=REPO_ROOT)
    assert out.returncode == 0, out.stderr[-2000:]
    line = json.loads(out.stdout.strip().splitlines()[-1])
    assert line["metric"] == "dns_queries_per_sec"
    assert line["value"] > 1000


def test_native_zkd_registry_chain(tmp_path):
    """The native registry on the real box: zkd-backed mirror serves,
    survives a zkd hard-kill + restart on the same data, and new
    writes propagate (exercises bin/zkd in the recorded gpu tier)."""
    from binder_amd.harness import NativeZkd
    from binder_amd.zkclient import ZkConn

    d = tmp_path / "data"
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
                          workdir=tmp_path)
        b.start()
        r = b.wait_ready("web.foo.com", timeout=30)
        assert r.answers[0]["address"] == "10.0.0.9"

        z.proc.kill()
        z.proc.wait()
        assert b.dig("web.foo.com").answers[0]["address"] == "10.0.0.9"

        z = NativeZkd(port=port, data_dir=str(d)).start()
        assert z.nodes_restored == 3
        c = ZkConn("127.0.0.1", z.port)
        c.create("/com/foo/neu", json.dumps(
            {"type": "host",
             "host": {"address": "10.0.0.10"}}).encode())
        r = b.wait_ready("neu.foo.com", timeout=40)
        assert r.answers[0]["address"] == "10.0.0.10"
        c.close()
    finally:
        if b:
            b.stop()
        z.stop()
