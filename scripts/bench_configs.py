#!/usr/bin/env python3
"""Measure all five BASELINE.json benchmark configs; one JSON line each.

  1 single A lookup against the in-process stub store (no sockets:
    pure codec+engine, binder_amd._native.StubEngine.query_wire)
  2 1k host records in a (stub) ZooKeeper, one binderd, wire-driven
  3 10k SRV service records, warm cache steady state
  4 recursion enabled, all queries are misses forwarded upstream
  5 4 binderd behind the balancer, 100k mixed records, 1000 mut/s churn

usage: bench_configs.py [--quick] [--out FILE]
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

from binder_amd import require_native            # noqa: E402
from binder_amd.harness import BinderProcess  # noqa: E402
from binder_amd.stubzk import StubZk             # noqa: E402

RESULTS = []


def report(config, metric, value, unit, **extra):
    row = {"config": config, "metric": metric,
           "value": round(value, 1), "unit": unit, **extra}
    RESULTS.append(row)
    print(json.dumps(row), flush=True)


def blast(port, queries, names, threads=4, window=64, rd=False,
          bind_base=None):
    cmd = [str(REPO / "bin" / "dnsblast"), "-s", "127.0.0.1",
           "-p", str(port), "-n", str(queries), "-c", str(window),
           "-t", str(threads), "-f", str(names), "-T", "10000"]
    if rd:
        cmd.append("-R")
    if bind_base:
        cmd += ["-B", bind_base]
    out = subprocess.run(cmd, capture_output=True, text=True, check=True)
    return json.loads(out.stdout.strip())


def write_names(path, names):
    Path(path).write_text("\n".join(names))


def config1(n_queries):
    n = require_native()
    e = n.StubEngine("foo.com", "coal", False)
    e.put("foo.com", "null")
    e.put("web.foo.com", json.dumps(
        {"type": "host", "host": {"address": "10.0.0.1"}}))
    wire = n.encode_message(
        {"id": 1, "questions": [{"name": "web.foo.com", "type": "A"}]})
    # warmup
    for _ in range(10000):
        e.query_wire(wire, 512)
    t0 = time.perf_counter()
    for _ in range(n_queries):
        e.query_wire(wire, 512)
    dt = time.perf_counter() - t0
    report(1, "in_proc_lookups_per_sec", n_queries / dt, "queries/s",
           note="single A vs in-process stub store, one thread, "
                "includes Python call overhead")
    qps = e.bench_wire(wire, n_queries)
    report(1, "in_proc_lookups_per_sec_native_loop", qps, "queries/s",
           note="same path, C++ loop (decode+resolve+encode only)")


def config2(tmp, n_queries):
    zk = StubZk().start()
    try:
        zk.mkdirp("/com/foo")
        names = []
        for i in range(1000):
            zk.put(f"/com/foo/h{i}", json.dumps(
                {"type": "host",
                 "host": {"address": f"10.4.{i // 250}.{i % 250}"}}
            ).encode())
            names.append(f"h{i}.foo.com A")
        nf = tmp / "c2names.txt"
        write_names(nf, names)
        srv = BinderProcess(store="zk", zk_host="127.0.0.1",
                            zk_port=zk.port, workdir=tmp,
                            log_level="warn")
        srv.start()
        try:
            srv.wait_ready("h999.foo.com", timeout=30)
            blast(srv.port, n_queries // 5, nf)  # warm
            r = blast(srv.port, n_queries, nf)
            report(2, "qps_1k_tree_one_proc", r["qps"], "queries/s",
                   p50_us=r["p50_us"], p99_us=r["p99_us"])
        finally:
            srv.stop()
    finally:
        zk.stop()


def config3(tmp, n_queries):
    zk = StubZk().start()
    try:
        zk.mkdirp("/com/foo")
        names = []
        for i in range(2000):
            zk.put(f"/com/foo/s{i}", json.dumps(
                {"type": "service",
                 "service": {"srvce": "_x", "proto": "_tcp",
                             "port": 80, "ttl": 60}}).encode())
            for j in range(4):
                zk.put(f"/com/foo/s{i}/m{j}", json.dumps(
                    {"type": "rr_host",
                     "rr_host": {"address": f"10.5.{i % 250}.{j+1}"}}
                ).encode())
            names.append(f"_x._tcp.s{i}.foo.com SRV")
        nf = tmp / "c3names.txt"
        write_names(nf, names)
        srv = BinderProcess(store="zk", zk_host="127.0.0.1",
                            zk_port=zk.port, workdir=tmp,
                            log_level="warn")
        srv.start()
        try:
            srv.wait_ready("_x._tcp.s1999.foo.com", qtype="SRV",
                           timeout=60)
            blast(srv.port, n_queries // 5, nf)
            r = blast(srv.port, n_queries, nf)
            report(3, "qps_10k_srv_warm", r["qps"], "queries/s",
                   p50_us=r["p50_us"], p99_us=r["p99_us"],
                   answers_per_query=round(r["answers"] /
                                           max(r["received"], 1), 2))
        finally:
            srv.stop()
    finally:
        zk.stop()


def config4(tmp, n_queries):
    # upstream on 127.0.0.2 answers; local binder forwards every miss
    up_tree = tmp / "c4up.json"
    upstream_names = []
    up = {"dc2.foo.com": None}
    for i in range(500):
        up[f"h{i}.dc2.foo.com"] = {
            "type": "host", "host": {"address": f"10.6.0.{i % 250}"}}
        upstream_names.append(f"h{i}.dc2.foo.com A")
    up_tree.write_text(json.dumps(up))
    upstream = BinderProcess(dns_domain="dc2.foo.com", datacenter="dc2",
                             host="127.0.0.2", store=f"file:{up_tree}",
                             workdir=tmp, log_level="warn")
    upstream.start()
    local_tree = tmp / "c4local.json"
    local_tree.write_text(json.dumps({"foo.com": None}))
    local = BinderProcess(
        dns_domain="foo.com", datacenter="dc1",
        store=f"file:{local_tree}", workdir=tmp, log_level="warn",
        config={"recursion": {
            "source": "static", "regionName": "r1",
            "dnsDomain": "foo.com", "upstreamPort": upstream.port,
            "dcs": {"dc2": ["127.0.0.2"]}}})
    local.start()
    try:
        nf = tmp / "c4names.txt"
        write_names(nf, upstream_names)
        deadline = time.time() + 20
        while time.time() < deadline:
            try:
                if local.dig("h0.dc2.foo.com", rd=True,
                             timeout=2).status == "NOERROR":
                    break
            except OSError:
                pass
            time.sleep(0.2)
        else:
            raise TimeoutError("recursion never warmed up")
        blast(local.port, max(n_queries // 50, 2000), nf, rd=True,
              threads=2, window=16)
        r = blast(local.port, max(n_queries // 10, 5000), nf, rd=True,
                  threads=2, window=16)
        report(4, "qps_recursive_forwarding", r["qps"], "queries/s",
               p50_us=r["p50_us"], p99_us=r["p99_us"],
               noerror_frac=round(r["noerror"] /
                                  max(r["received"], 1), 3))
    finally:
        local.stop()
        upstream.stop()


def config5(tmp, n_queries):
    out = subprocess.run(
        [sys.executable, str(REPO / "bench.py"), "--gpus", "4",
         "--tree-records", "100000", "--churn-qps", "1000",
         "--steps", "2", "--warmup", "1",
         "--queries-per-proc", str(n_queries // 4)],
        capture_output=True, text=True, timeout=600, cwd=REPO)
    if out.returncode != 0:
        report(5, "qps_4proc_100k_churn", 0, "queries/s",
               error=out.stderr[-400:])
        return
    d = json.loads(out.stdout.strip().splitlines()[-1])
    report(5, "qps_4proc_100k_churn", d["value"], "queries/s",
           p50_us=d["config"]["p50_us"], p99_us=d["config"]["p99_us"],
           churn_qps=1000, tree_records=100000)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--quick", action="store_true")
    ap.add_argument("--out", default=None)
    ap.add_argument("--skip", default="",
                    help="comma-separated config numbers to skip")
    args = ap.parse_args()
    nq = 50_000 if args.quick else 400_000
    skip = {int(x) for x in args.skip.split(",") if x}

    with tempfile.TemporaryDirectory(prefix="bench-cfg-") as td:
        tmp = Path(td)
        for i, fn in ((1, lambda: config1(nq)),
                      (2, lambda: config2(tmp, nq)),
                      (3, lambda: config3(tmp, nq)),
                      (4, lambda: config4(tmp, nq)),
                      (5, lambda: config5(tmp, nq * 4))):
            if i in skip:
                continue
            try:
                fn()
            except Exception as e:  # keep going; report the failure
                report(i, "error", 0, "", error=str(e)[:300])

    if args.out:
        Path(args.out).write_text(
            "\n".join(json.dumps(r) for r in RESULTS) + "\n")


if __name__ == "__main__":
    main()
