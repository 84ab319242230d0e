#!/usr/bin/env python3
"""Isolate the rare ~10 ms stall events that flake p99 probes
(profiles/SCALING.md round-2 'remaining limiter'): run the same paced
load under one-variable-at-a-time variants and print each probe's
p90/p99 so the stall's presence/absence per variant is visible.
"""
import argparse
import json
import subprocess
import sys
import tempfile
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

import bench  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--procs", type=int, default=1)
    ap.add_argument("--rate", type=int, default=1_500_000)
    ap.add_argument("--probes", type=int, default=6)
    args = ap.parse_args()

    from binder_amd.harness import free_port, NativeZkd, BALANCERD
    from binder_amd.zkclient import ZkConn
    import os

    tmp = Path(tempfile.mkdtemp(prefix="stall-"))
    names_file = tmp / "names.txt"
    zkd = NativeZkd().start()
    conn = ZkConn("127.0.0.1", zkd.port)
    bench.build_tree(conn, names_file, 10000)
    backends, sockdir = bench.start_backends(args.procs, tmp, zkd.port)

    variants = [
        ("default", {"workers": 12, "rescan": 200, "stats": True,
                     "gso": True}),
        ("no-gso", {"workers": 12, "rescan": 200, "stats": True,
                    "gso": False}),
        ("rescan-2000", {"workers": 12, "rescan": 2000, "stats": True,
                         "gso": True}),
        ("workers-4", {"workers": 4, "rescan": 200, "stats": True,
                       "gso": True}),
        ("no-stats", {"workers": 12, "rescan": 200, "stats": False,
                      "gso": True}),
    ]

    for tag, v in variants:
        port = free_port()
        cmd = [str(BALANCERD), "-p", str(port), "-H", "127.0.0.1",
               "-s", str(sockdir), "-r", str(v["rescan"]),
               "-w", str(v["workers"])]
        if v["stats"]:
            cmd += ["-S", str(tmp / "stats.sock")]
        bal = subprocess.Popen(
            cmd, env=dict(os.environ, LOG_LEVEL="warn"),
            stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
        time.sleep(1.5)
        try:
            # one warm run, then the probes
            extra = [] if v["gso"] else ["-g"]
            def blast(q, rate):
                c = [str(REPO / "bin" / "dnsblast"), "-s", "127.0.0.1",
                     "-p", str(port), "-n", str(q), "-c", "512",
                     "-t", "10", "-P", "8", "-f", str(names_file),
                     "-B", "127.0.1.1", "-T", "10000"] + extra
                if rate:
                    c += ["-r", str(rate)]
                out = subprocess.run(c, capture_output=True, text=True,
                                     check=True)
                return json.loads(out.stdout.strip())
            blast(500_000, 0)
            p99s, p90s = [], []
            for _ in range(args.probes):
                r = blast(int(args.rate * 2.0), args.rate)
                p99s.append(r["p99_us"])
                p90s.append(r["p90_us"])
            print(json.dumps({"variant": tag, **v,
                              "p90s": p90s, "p99s": p99s,
                              "stalls": sum(1 for p in p99s
                                            if p > 4000)}),
                  flush=True)
        finally:
            bal.terminate()
            bal.wait(timeout=5)

    for b in backends:
        b.stop()
    conn.close()
    zkd.stop()


if __name__ == "__main__":
    main()
