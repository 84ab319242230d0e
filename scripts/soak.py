#!/usr/bin/env python3
"""Soak test: sustained load + churn while sampling process RSS.

Catches slow leaks (pending-map growth, watch bookkeeping, buffer
bloat) that short benches miss. Exits nonzero if RSS keeps climbing
after warmup or if service degrades.

usage: soak.py [seconds] [n_backends] (default 60 2)
"""
import json
import subprocess
import sys
import tempfile
import threading
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

from binder_amd.harness import (  # noqa: E402
    BALANCERD, BinderProcess, free_port)
from binder_amd.stubzk import StubZk  # noqa: E402


def rss_mb(pid):
    try:
        with open(f"/proc/{pid}/status") as f:
            for line in f:
                if line.startswith("VmRSS:"):
                    return int(line.split()[1]) / 1024.0
    except OSError:
        return 0.0
    return 0.0


def main():
    seconds = int(sys.argv[1]) if len(sys.argv) > 1 else 60
    n_backends = int(sys.argv[2]) if len(sys.argv) > 2 else 2
    tmp = Path(tempfile.mkdtemp(prefix="soak-"))
    zk = StubZk().start()
    zk.mkdirp("/com/foo")
    names = []
    for i in range(5000):
        zk.put(f"/com/foo/h{i}", json.dumps(
            {"type": "host",
             "host": {"address": f"10.7.{(i >> 8) & 255}.{i & 255}"}}
        ).encode())
        names.append(f"h{i}.foo.com A")
    (tmp / "names.txt").write_text("\n".join(names))

    sockdir = tmp / "socks"
    sockdir.mkdir()
    backends = []
    for i in range(n_backends):
        b = BinderProcess(store="zk", zk_host="127.0.0.1",
                          zk_port=zk.port, workdir=tmp,
                          log_level="warn",
                          balancer_socket=str(sockdir / f"b{i}"))
        b.start()
        backends.append(b)
    for b in backends:
        b.wait_ready("h4999.foo.com", timeout=60)

    bport = free_port()
    bal = subprocess.Popen(
        [str(BALANCERD), "-p", str(bport),
         "-H", "127.0.0.1", "-s", str(sockdir), "-r", "200",
         "-w", str(min(8, n_backends))],
        stdout=subprocess.DEVNULL, stderr=subprocess.STDOUT)
    time.sleep(1.0)

    stop = threading.Event()

    def churner():
        i = 0
        while not stop.is_set():
            h = i % 5000
            zk.put(f"/com/foo/h{h}", json.dumps(
                {"type": "host",
                 "host": {"address": f"10.8.{i % 200}.{h % 250}"}}
            ).encode())
            i += 1
            time.sleep(0.005)

    t = threading.Thread(target=churner, daemon=True)
    t.start()

    # TCP pokes too (connection churn exercises idle sweeps)
    def tcp_poker():
        from binder_amd.digclient import dig
        while not stop.is_set():
            try:
                dig("h1.foo.com", port=bport, tcp=True, timeout=2)
            except OSError:
                pass
            time.sleep(0.05)

    t2 = threading.Thread(target=tcp_poker, daemon=True)
    t2.start()

    blast = subprocess.Popen(
        [str(REPO / "bin" / "dnsblast"), "-s", "127.0.0.1",
         "-p", str(bport), "-n", str(10_000_000_000), "-c", "64",
         "-t", str(2 * n_backends), "-f", str(tmp / "names.txt"),
         "-B", "127.0.1.1"],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT)

    samples = []
    pids = [b.proc.pid for b in backends] + [bal.pid]
    t0 = time.time()
    ok = True
    try:
        while time.time() - t0 < seconds:
            time.sleep(5)
            row = [round(time.time() - t0)] + \
                [round(rss_mb(p), 1) for p in pids]
            samples.append(row)
            print(json.dumps({"t": row[0], "rss_mb": row[1:]}),
                  flush=True)
        # leak check: compare steady-state to early sample
        if len(samples) >= 4:
            early = samples[1]
            late = samples[-1]
            for i in range(1, len(early)):
                if late[i] > early[i] * 1.5 + 20:
                    print(f"LEAK SUSPECT pid[{i}]: "
                          f"{early[i]} -> {late[i]} MiB")
                    ok = False
        # service still healthy?
        from binder_amd.digclient import dig
        r = dig("h1.foo.com", port=bport, timeout=2)
        if r.status != "NOERROR":
            print(f"DEGRADED: {r.status}")
            ok = False
    finally:
        stop.set()
        blast.terminate()
        bal.terminate()
        for b in backends:
            b.stop()
        zk.stop()
    print("SOAK", "OK" if ok else "FAIL")
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
