#!/usr/bin/env python3
"""One-off scaling probe: where does the N-process chain saturate?

Starts N binderd (file store for fast startup) + balancer with W
workers, runs dnsblast at several window/thread settings, and dumps
per-backend query distribution from the balancer stats socket.

usage: scale_probe.py [N] [workers]
"""
import json
import os
import socket
import subprocess
import sys
import tempfile
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

from binder_amd.harness import (  # noqa: E402
    BALANCERD, BinderProcess, free_port)


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 8
    workers = int(sys.argv[2]) if len(sys.argv) > 2 else min(8, n)
    tmp = Path(tempfile.mkdtemp(prefix="scale-probe-"))
    tree = {"foo.com": None}
    names = []
    for i in range(5000):
        tree[f"h{i}.foo.com"] = {
            "type": "host",
            "host": {"address": f"10.{(i >> 8) & 255}.{i & 255}.1"}}
        names.append(f"h{i}.foo.com A")
    # same A+SRV mix as bench.py
    for i in range(1000):
        tree[f"s{i}.foo.com"] = {
            "type": "service",
            "service": {"srvce": "_x", "proto": "_tcp", "port": 80,
                        "ttl": 60}}
        for j in range(4):
            tree[f"m{j}.s{i}.foo.com"] = {
                "type": "rr_host",
                "rr_host": {"address": f"10.9.{i % 250}.{j + 1}"}}
        names.append(f"s{i}.foo.com A")
        names.append(f"_x._tcp.s{i}.foo.com SRV")
    (tmp / "tree.json").write_text(json.dumps(tree))
    (tmp / "names.txt").write_text("\n".join(names))

    sockdir = tmp / "socks"
    sockdir.mkdir()
    backends = []
    for i in range(n):
        b = BinderProcess(store=f"file:{tmp/'tree.json'}", workdir=tmp,
                          log_level="warn",
                          balancer_socket=str(sockdir / f"b{i}"))
        b.start()
        backends.append(b)

    bal_port = free_port()
    bal = None

    def restart_balancer(w):
        nonlocal bal, bal_port
        if bal is not None:
            bal.terminate()
            bal.wait(timeout=5)
        bal_port = free_port()
        bal = subprocess.Popen(
            [str(BALANCERD), "-p", str(bal_port),
             "-H", "127.0.0.1", "-s", str(sockdir),
             "-S", str(tmp / "stats.sock"), "-r", "200",
             "-w", str(w)],
            env=dict(os.environ, LOG_LEVEL="warn"),
            stdout=subprocess.DEVNULL, stderr=subprocess.STDOUT)
        time.sleep(1.2)

    def stats():
        with socket.socket(socket.AF_UNIX) as s:
            s.connect(str(tmp / "stats.sock"))
            return json.loads(s.recv(1 << 20).decode())

    def blast(q, window, threads):
        out = subprocess.run(
            [str(REPO / "bin" / "dnsblast"), "-s", "127.0.0.1",
             "-p", str(bal_port), "-n", str(q), "-c", str(window),
             "-t", str(threads), "-f", str(tmp / "names.txt"),
             "-B", "127.0.1.1", "-T", "5000"],
            capture_output=True, text=True, check=True)
        return json.loads(out.stdout.strip())

    try:
        # direct-to-one-backend ceiling
        d = subprocess.run(
            [str(REPO / "bin" / "dnsblast"), "-s", "127.0.0.1",
             "-p", str(backends[0].port), "-n", "300000", "-c", "64",
             "-t", "4", "-f", str(tmp / "names.txt")],
            capture_output=True, text=True, check=True)
        print("direct-1-backend:", d.stdout.strip())

        if len(sys.argv) > 3:
            configs = [tuple(int(x) for x in a.split(":"))
                       for a in sys.argv[3:]]
        else:
            configs = [(n, 64, 4 * n), (2 * n, 64, 4 * n),
                       (2 * n, 64, 6 * n), (2 * n, 128, 4 * n),
                       (16, 64, 32)]
        for bal_workers, window, threads in configs:
            restart_balancer(bal_workers)
            per_point = int(os.environ.get("PROBE_QUERIES",
                                            str(150000 * n)))
            r = blast(per_point, window, threads)
            st = stats()
            qtot = sum(b["queries"] for b in st["backends"])
            dist = sorted(round(b["queries"] / max(qtot, 1), 3)
                          for b in st["backends"])
            print(f"bw={bal_workers} w={window} t={threads}: "
                  f"qps={r['qps']:.0f} p50={r['p50_us']} "
                  f"p99={r['p99_us']} to={r['timeouts']} dist={dist}")
    finally:
        bal.terminate()
        for b in backends:
            b.stop()


if __name__ == "__main__":
    main()
