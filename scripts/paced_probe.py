#!/usr/bin/env python3
"""Paced-mode diagnosis probe: where do the multi-ms tails at N=8 come
from? Builds the bench topology once, then sweeps offered rate, client
threads/sockets, and balancer workers, printing one JSON line per
experiment plus balancer drop/overwrite deltas.

usage: paced_probe.py [--procs 8] [--workers 16] [--tree 10000]
"""
import argparse
import json
import socket
import sys
import tempfile
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

import bench  # noqa: E402  (reuse the harness pieces)


def cpustat():
    try:
        d = {}
        for line in open("/sys/fs/cgroup/cpu.stat"):
            k, v = line.split()
            d[k] = int(v)
        return d
    except OSError:
        return {}


def balstat(path):
    with socket.socket(socket.AF_UNIX) as s:
        s.settimeout(2)
        s.connect(str(path))
        return json.loads(s.recv(1 << 20).decode())


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--procs", type=int, default=8)
    ap.add_argument("--workers", type=int, default=16)
    ap.add_argument("--tree", type=int, default=10000)
    ap.add_argument("--quick", action="store_true")
    ap.add_argument("--sweep", action="store_true",
                    help="sweep balancer workers x client threads "
                         "(closed loop) for the quota-bound operating "
                         "point, then pace the best")
    args = ap.parse_args()

    from binder_amd.harness import free_port
    from binder_amd.stubzk import StubZk

    tmp = Path(tempfile.mkdtemp(prefix="paced-probe-"))
    names_file = tmp / "names.txt"
    zk = StubZk().start()
    bench.build_tree(zk, names_file, args.tree)
    backends, sockdir = bench.start_backends(
        args.procs, tmp, zk.port,
        last_name=f"h{args.tree // 2 - 1}.foo.com")
    port = free_port()
    bal = bench.start_balancer(tmp, sockdir, port, workers=args.workers)
    bench.wait_balancer_ready(port, args.procs, tmp)
    stats_path = tmp / "stats.sock"

    def stat_delta(before, after):
        return {k: after[k] - before[k]
                for k in ("udp_queries", "udp_replies", "drops")} | {
            "overwrites": sum(b.get("overwrites", 0)
                              for b in after["backends"]) -
            sum(b.get("overwrites", 0) for b in before["backends"])}

    def run(tag, queries, threads, window, socks, rate):
        b0 = balstat(stats_path)
        r = bench.run_blast(port, queries, names_file, threads, window,
                            socks=socks, rate=rate)
        b1 = balstat(stats_path)
        out = {"tag": tag, "threads": threads, "window": window,
               "socks": socks, "rate": rate,
               "qps": r["qps"], "p50_us": r["p50_us"],
               "p90_us": r["p90_us"], "p99_us": r["p99_us"],
               "timeouts": r["timeouts"],
               "bal": stat_delta(b0, b1)}
        print(json.dumps(out), flush=True)
        return r

    q = 2_000_000 if not args.quick else 400_000
    try:
        if not args.sweep:
            run("ramp1", q, 24, 128, 8, 0)
            cap = run("closed-P8", q, 24, 128, 8, 0)["qps"]
            for t in (8, 12, 16, 24):
                rate = int(cap * 0.85)
                run(f"paced-0.85-t{t}-P8-w256", q, t, 256, 8, rate)
            for frac in (0.95, 0.9, 1.0):
                rate = int(cap * frac)
                run(f"paced-{frac}-t16-P8-w256", q, 16, 256, 8, rate)
            run("closed-t16-P8-w256", q, 16, 256, 8, 0)
            return
        # quota-aware sweep: restart the balancer per worker count
        nonlocal_best = {"qps": 0}
        for w in (8, 12):
            bal.terminate()
            bal.wait(timeout=5)
            port2 = free_port()
            bal = bench.start_balancer(tmp, sockdir, port2,
                                       workers=w)
            bench.wait_balancer_ready(port2, args.procs, tmp)
            port = port2

            def run2(tag, threads, window, socks, rate):
                b0 = balstat(stats_path)
                c0 = cpustat()
                r = bench.run_blast(port, q, names_file, threads,
                                    window, socks=socks, rate=rate)
                c1 = cpustat()
                b1 = balstat(stats_path)
                out = {"tag": tag, "workers": w, "threads": threads,
                       "window": window, "socks": socks, "rate": rate,
                       "qps": r["qps"], "p50_us": r["p50_us"],
                       "p90_us": r["p90_us"], "p99_us": r["p99_us"],
                       "timeouts": r["timeouts"],
                       "throttled": c1.get("nr_throttled", 0) -
                       c0.get("nr_throttled", 0),
                       "throttled_ms": (c1.get("throttled_usec", 0) -
                                        c0.get("throttled_usec", 0))
                       // 1000,
                       "bal": stat_delta(b0, b1)}
                print(json.dumps(out), flush=True)
                return r

            run2(f"w{w}-ramp", 10, 512, 8, 0)
            cap = run2(f"w{w}-closed", 10, 512, 8, 0)["qps"]
            for frac in (0.55, 0.75, 0.9):
                r = run2(f"w{w}-paced{frac}", 10, 512, 8,
                         int(cap * frac))
                if r["qps"] > nonlocal_best["qps"]:
                    nonlocal_best = {"qps": r["qps"], "w": w,
                                     "frac": frac}
        print(json.dumps({"best": nonlocal_best}), flush=True)
    finally:
        bal.terminate()
        for b in backends:
            b.stop()
        zk.stop()


if __name__ == "__main__":
    main()
