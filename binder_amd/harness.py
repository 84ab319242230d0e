"""Process harness: start/stop binderd (and friends) for tests and bench.

The reference test fixture starts an in-process server against a real
ZooKeeper (/root/reference/test/helper.js:47-96); binderd is a separate
native process, so the fixture here manages child processes and waits
for readiness via the wire (first successful DNS response) or the
metrics endpoint.
"""
from __future__ import annotations

import json
import os
import signal
import socket
import subprocess
import time
from pathlib import Path
from typing import Optional

from . import REPO_ROOT
from .digclient import dig

# BINDERD_BIN overrides the server binary (e.g. an ASan build)
BINDERD = Path(os.environ.get("BINDERD_BIN",
                              REPO_ROOT / "bin" / "binderd"))
BALANCERD = Path(os.environ.get("BALANCER_BIN",
                                REPO_ROOT / "bin" / "binder-balancer"))
SUPERVISORD = Path(os.environ.get("SUPERVISOR_BIN",
                                  REPO_ROOT / "bin" / "binder-supervisor"))
ADJUST = Path(os.environ.get("ADJUST_BIN",
                             REPO_ROOT / "bin" / "binder-adjust"))


def free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


class BinderProcess:
    """One binderd instance."""

    def __init__(self, dns_domain="foo.com", datacenter="coal",
                 port: Optional[int] = None, host="127.0.0.1",
                 store: Optional[str] = None,
                 config: Optional[dict] = None,
                 balancer_socket: Optional[str] = None,
                 zk_host: Optional[str] = None, zk_port: Optional[int] = None,
                 log_path: Optional[str] = None, log_level="info",
                 workdir: Optional[Path] = None, verbose=0):
        self.port = port or free_port()
        self.host = host
        self.dns_domain = dns_domain
        self.workdir = Path(workdir or "/tmp")
        cfg = {
            "dnsDomain": dns_domain,
            "datacenterName": datacenter,
            "port": self.port,
            "host": host,
            "metricsPort": free_port(),
        }
        if config:
            cfg.update(config)
        self.config = cfg
        self.config_path = self.workdir / f"binder-{self.port}.json"
        self.config_path.write_text(json.dumps(cfg))
        self.cmd = [str(BINDERD), "-f", str(self.config_path)]
        if store:
            self.cmd += ["-S", store]
        if balancer_socket:
            self.cmd += ["-b", balancer_socket]
        self.cmd += ["-v"] * verbose
        self.env = dict(os.environ)
        self.env["LOG_LEVEL"] = log_level
        if zk_host:
            self.env["ZK_HOST"] = zk_host
        if zk_port:
            self.env["ZK_PORT"] = str(zk_port)
        self.log_path = log_path
        self.proc: Optional[subprocess.Popen] = None

    @property
    def metrics_port(self) -> int:
        return self.config["metricsPort"]

    def start(self, wait_ready=True, timeout=20.0):
        logf = open(self.log_path, "ab") if self.log_path \
            else subprocess.DEVNULL
        self.proc = subprocess.Popen(
            self.cmd, env=self.env, stdout=logf, stderr=subprocess.STDOUT)
        if wait_ready:
            self.wait_listening(timeout)
        return self

    def wait_listening(self, timeout=20.0):
        """Wait until the server answers DNS at all (any rcode)."""
        deadline = time.time() + timeout
        last_err = None
        while time.time() < deadline:
            if self.proc.poll() is not None:
                raise RuntimeError(
                    f"binderd exited early with {self.proc.returncode}")
            try:
                dig("ready-probe.invalid", "A", server=self.host,
                    port=self.port, timeout=0.25)
                return
            except (socket.timeout, OSError, ValueError) as e:
                last_err = e
                time.sleep(0.05)
        raise TimeoutError(f"binderd not listening: {last_err}")

    def wait_ready(self, name, timeout=15.0, rcode="NOERROR", qtype="A"):
        """Wait until `name` resolves with the given rcode."""
        deadline = time.time() + timeout
        while time.time() < deadline:
            try:
                r = dig(name, qtype, server=self.host, port=self.port,
                        timeout=0.25)
                if r.status == rcode:
                    return r
            except (socket.timeout, OSError):
                pass
            time.sleep(0.05)
        raise TimeoutError(f"{name} never reached {rcode}")

    def metrics(self) -> str:
        import urllib.request
        with urllib.request.urlopen(
                f"http://127.0.0.1:{self.metrics_port}/metrics",
                timeout=2) as r:
            return r.read().decode()

    def dig(self, name, qtype="A", **kw):
        kw.setdefault("server", self.host)
        kw.setdefault("port", self.port)
        return dig(name, qtype, **kw)

    def sigterm(self):
        if self.proc:
            self.proc.send_signal(signal.SIGTERM)

    def stop(self):
        if self.proc and self.proc.poll() is None:
            self.proc.terminate()
            try:
                self.proc.wait(timeout=5)
            except subprocess.TimeoutExpired:
                self.proc.kill()
                self.proc.wait()
        self.proc = None

    def __enter__(self):
        return self.start()

    def __exit__(self, *exc):
        self.stop()


ZKD = Path(os.environ.get("ZKD_BIN", REPO_ROOT / "bin" / "zkd"))


class NativeZkd:
    """bin/zkd (the native single-node registry) as a test fixture."""

    def __init__(self, port=0, data_dir=None, session_timeout_ms=30000,
                 host="127.0.0.1"):
        self.host = host
        self.port = port
        self.data_dir = data_dir
        self.session_timeout_ms = session_timeout_ms
        self.proc: Optional[subprocess.Popen] = None
        self.nodes_restored = 0

    def start(self, timeout=10.0):
        cmd = [str(ZKD), "-H", self.host, "-p", str(self.port),
               "-t", str(self.session_timeout_ms)]
        if self.data_dir:
            cmd += ["-d", str(self.data_dir)]
        self.proc = subprocess.Popen(
            cmd, stdout=subprocess.PIPE, stderr=subprocess.DEVNULL,
            text=True)
        # "zkd listening on H:P, N nodes restored"
        deadline = time.time() + timeout
        line = ""
        while time.time() < deadline:
            line = self.proc.stdout.readline()
            if "listening on" in line:
                break
        if "listening on" not in line:
            raise RuntimeError("zkd never reported listening")
        self.port = int(line.split(":")[1].split(",")[0])
        self.nodes_restored = int(line.split(",")[1].split()[0])
        return self

    def stop(self):
        if self.proc and self.proc.poll() is None:
            self.proc.terminate()
            try:
                self.proc.wait(timeout=5)
            except subprocess.TimeoutExpired:
                self.proc.kill()
                self.proc.wait()
        self.proc = None
