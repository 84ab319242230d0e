#!/usr/bin/env python3
"""binder-amd flagship benchmark (driver contract entrypoint).

Measures the BASELINE.json headline metric: DNS queries/sec (+ p50/p99
latency) served from a 10,000-record ZooKeeper-backed tree, with N
server processes behind the native balancer.

Topology per run (all on one node; "GPU" count N maps to N binderd
server processes, the reference's only parallelism axis — SURVEY.md
§2.3: replicated processes behind one balancer, no sharding):

    zkd (native registry, 10k records)
      <-- N x binderd (zk mirror) <-- binder-balancer
                                         ^
         dnsblast daemon (C++ load generator, GSO bursts, paced)

Protocol: fixed-rate qps@SLO — untimed calibration binary-searches the
highest offered rate sustaining p99 <= 2 ms with zero timeouts and
>=97% delivery (2-of-3 probes per level + margin), then the timed
steps run at that rate on a persistent generator, so a "step" is a
fixed batch of Q = 1M*N queries of pure query work (weak scaling).
Queries are a uniform A+SRV mix over the tree; replies are verified
NOERROR (noerror_frac reported). Note: this workload has no tensor compute — the reference is a
Node.js DNS server (BASELINE.json "north_star" records the tier
mismatch); torch is used only for the multi-rank barrier contract.
"""
import argparse
import json
import os
import shutil
import subprocess
import sys
import tempfile
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent
sys.path.insert(0, str(REPO))


def log(msg):
    print(f"# bench: {msg}", file=sys.stderr, flush=True)


def effective_cpus():
    """CPUs actually available to this cgroup: the bench boxes expose
    256 hw threads but cap cpu.max at 16 cores — sizing against
    os.cpu_count() oversubscribes the quota and CFS throttling puts
    10-100 ms whole-group freezes straight into p99 (measured,
    profiles/SCALING.md round 2)."""
    try:  # cgroup v2
        parts = Path("/sys/fs/cgroup/cpu.max").read_text().split()
        if parts[0] != "max":
            return max(2, int(parts[0]) // int(parts[1]))
    except (OSError, ValueError, IndexError):
        pass
    try:  # cgroup v1
        q = int(Path("/sys/fs/cgroup/cpu/cpu.cfs_quota_us")
                .read_text())
        p = int(Path("/sys/fs/cgroup/cpu/cpu.cfs_period_us")
                .read_text())
        if q > 0 and p > 0:
            return max(2, q // p)
    except (OSError, ValueError):
        pass
    return os.cpu_count() or 8


def build_tree(zk, names_path, records=10000):
    """R records: R/2 hosts + R/10 services x 4 members (=R nodes with
    payloads), uniform A+SRV query mix. R=10000 is the BASELINE
    headline config; --tree-records 100000 with --churn-qps exercises
    BASELINE config 5."""
    names = []
    zk.mkdirp("/com/foo")
    batch = []
    n_hosts = records // 2
    n_svcs = records // 10
    for i in range(n_hosts):
        batch.append((f"/com/foo/h{i}",
                      {"type": "host",
                       "host": {"address": f"10.{(i >> 8) & 255}.{i & 255}.1"}}))
        names.append(f"h{i}.foo.com A")
    for i in range(n_svcs):
        batch.append((f"/com/foo/s{i}",
                      {"type": "service",
                       "service": {"srvce": "_x", "proto": "_tcp",
                                   "port": 80, "ttl": 60}}))
        for j in range(4):
            batch.append((f"/com/foo/s{i}/m{j}",
                          {"type": "rr_host",
                           "rr_host": {"address": f"10.9.{i % 250}.{j + 1}"}}))
        names.append(f"s{i}.foo.com A")
        names.append(f"_x._tcp.s{i}.foo.com SRV")
    for path, obj in batch:
        zk.put(path, json.dumps(obj).encode())
    names_path.write_text("\n".join(names))
    return len(batch)


def start_backends(n, tmp, zk_port, last_name="h4999.foo.com"):
    from binder_amd.harness import BinderProcess, BALANCERD
    sockdir = tmp / "socks"
    sockdir.mkdir(exist_ok=True)
    backends = []
    for i in range(n):
        b = BinderProcess(dns_domain="foo.com", datacenter="coal",
                          store="zk", zk_host="127.0.0.1", zk_port=zk_port,
                          workdir=tmp,
                          log_level=os.environ.get("BENCH_LOG_LEVEL",
                                                   "warn"),
                          balancer_socket=str(sockdir / f"b{i}"),
                          log_path=str(tmp / f"binderd-{i}.log"))
        b.start(wait_ready=False)
        backends.append(b)
    for b in backends:
        b.wait_listening(timeout=30)
        b.wait_ready(last_name, timeout=120)
    return backends, sockdir


def start_balancer(tmp, sockdir, port, workers=1):
    from binder_amd.harness import BALANCERD
    env = dict(os.environ, LOG_LEVEL="warn")
    proc = subprocess.Popen(
        [str(BALANCERD), "-p", str(port),
         "-H", "127.0.0.1", "-s", str(sockdir),
         "-S", str(tmp / "stats.sock"), "-r", "200",
         "-w", str(workers)],
        env=env, stdout=open(tmp / "balancer.log", "ab"),
        stderr=subprocess.STDOUT)
    return proc


def wait_balancer_ready(port, n_backends, tmp, timeout=30):
    import socket as pysock
    from binder_amd.digclient import dig
    deadline = time.time() + timeout
    while time.time() < deadline:
        try:
            with pysock.socket(pysock.AF_UNIX) as s:
                s.settimeout(1)
                s.connect(str(tmp / "stats.sock"))
                stats = json.loads(s.recv(1 << 20).decode())
            ok = sum(1 for b in stats["backends"] if b["ok"])
            if ok >= n_backends:
                r = dig("h0.foo.com", server="127.0.0.1", port=port,
                        timeout=0.5)
                if r.status == "NOERROR":
                    return
        except (OSError, ValueError):
            pass
        time.sleep(0.2)
    raise TimeoutError("balancer never became ready")


def run_blast(port, queries, names_file, threads, window, socks=8,
              rate=0, timeout_ms=10000):
    cmd = [str(REPO / "bin" / "dnsblast"), "-s", "127.0.0.1",
           "-p", str(port), "-n", str(queries), "-c", str(window),
           "-t", str(threads), "-P", str(socks),
           "-f", str(names_file), "-B", "127.0.1.1",
           "-T", str(timeout_ms)]
    if rate > 0:
        cmd += ["-r", str(int(rate))]
    out = subprocess.run(cmd, capture_output=True, text=True, check=True)
    if out.stderr:
        sys.stderr.write(out.stderr)  # diagnostics from dnsblast
    return json.loads(out.stdout.strip().splitlines()[-1])


class BlastDaemon:
    """Persistent dnsblast (-D): the generator's threads/sockets/wires
    survive across steps, so the timed region contains only query
    work (process + socket setup measured 100-400 ms per invocation
    on the bench boxes)."""

    def __init__(self, port, names_file, threads, window, socks):
        self.proc = subprocess.Popen(
            [str(REPO / "bin" / "dnsblast"), "-D", "-s", "127.0.0.1",
             "-p", str(port), "-c", str(window), "-t", str(threads),
             "-P", str(socks), "-f", str(names_file),
             "-B", "127.0.1.1", "-T", "10000"],
            stdin=subprocess.PIPE, stdout=subprocess.PIPE, text=True,
            bufsize=1)

    def step(self, queries, rate=0):
        self.proc.stdin.write(f"RUN {int(queries)} {int(rate)}\n")
        self.proc.stdin.flush()
        line = self.proc.stdout.readline()
        if not line:
            raise RuntimeError("dnsblast daemon died")
        return json.loads(line)

    def close(self):
        try:
            self.proc.stdin.close()
            self.proc.wait(timeout=10)
        except (OSError, subprocess.TimeoutExpired):
            self.proc.kill()


def calibrate_rate(blast, capacity, slo_us):
    """qps@SLO discovery (untimed): binary-search the highest offered
    rate that sustains p99 <= SLO with zero timeouts and >=97%
    delivery. Near the edge single probes are flaky (rare ~10 ms
    stall events land in p99 or miss it), so a rate must pass TWO
    consecutive 2-second probes, and the chosen point keeps a 3%
    margin + one final confirmation. Fixed-rate steps at that point
    are comparable round over round, unlike the chaotic closed-loop
    saturation equilibrium (profiles/SCALING.md)."""
    def probe(rate, tag):
        q = max(400_000, int(rate * 2.0))
        r = blast.step(q, rate=rate)
        good = (r["timeouts"] == 0 and r["p99_us"] <= slo_us and
                r["qps"] >= 0.97 * rate)
        log(f"calibrate {tag}: offered {rate:.0f} -> "
            f"{r['qps']:.0f} qps, p99 {r['p99_us']}us "
            f"{'OK' if good else 'FAIL'}")
        return good

    lo, hi = 0.45 * capacity, 1.02 * capacity
    best = 0.0
    for i in range(5):
        mid = (lo + hi) / 2
        a = probe(mid, f"bisect{i}a")
        b = probe(mid, f"bisect{i}b") if a else False
        if a and not b:
            # one transient flake must not collapse the search:
            # 2-of-3 consecutive probes decide the level
            b = probe(mid, f"bisect{i}c")
        if a and b:
            best = mid
            lo = mid
        else:
            hi = mid
    rate = 0.97 * (best if best > 0 else lo)
    if not probe(rate, "confirm"):
        rate *= 0.93
    return rate


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1,
                    help="server process count (one per rank/GPU slot)")
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--queries-per-proc", type=int, default=1_000_000)
    ap.add_argument("--window", type=int, default=64)
    ap.add_argument("--tree-records", type=int, default=10_000)
    ap.add_argument("--churn-qps", type=int, default=0,
                    help="ZK mutations/sec during timed steps "
                         "(BASELINE config 5)")
    args = ap.parse_args()

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    n = max(args.gpus, world_size)

    dist = None
    if world_size > 1:
        import torch.distributed as dist_mod
        dist = dist_mod
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29511")
        dist.init_process_group(backend="gloo", rank=rank,
                                world_size=world_size)

    def barrier():
        if dist is not None:
            dist.barrier()

    def cuda_sync():
        try:
            import torch
            if torch.cuda.is_available():
                torch.cuda.synchronize()
        except Exception:
            pass

    q_step = args.queries_per_proc * n
    ncpu = min(os.cpu_count() or 8, effective_cpus())
    # Size the harness to the CPU actually available (cgroup quota, not
    # hw threads): operating points are probe-swept on the driver box
    # class (16-core quota; scripts/paced_probe.py --sweep, committed
    # under profiles/).
    if ncpu >= 12:
        # probe-swept best on the 16-core-quota box class with the
        # GSO/GRO ingress (profiles/, sweep3/sweep4_n8.jsonl): the
        # surface is flat at ~4.3M across w12-24 x t10-14; w12/t10 is
        # the lowest-footprint point on the plateau
        workers = max(2, min(12, (ncpu * 3) // 4))     # 12 @ 16 cpus
        threads = max(4, min(10, (ncpu * 5) // 8))     # 10 @ 16 cpus
        # deep windows so the paced generator can absorb RTT
        # excursions + sleep jitter without delivery deficit (the
        # calibration requires >=97% delivery)
        window = 512
        socks = 8  # flows = threads*socks >> balancer reuseport shards
        # experiment overrides (profiling/tuning only)
        workers = int(os.environ.get("BENCH_WORKERS", workers))
        threads = int(os.environ.get("BENCH_THREADS", threads))
        window = int(os.environ.get("BENCH_WINDOW", window))
        socks = int(os.environ.get("BENCH_SOCKS", socks))
    else:
        workers = 1
        threads = min(4 * n, max(2, ncpu // 2))
        window = args.window
        socks = 4
    slo_us = int(os.environ.get("BENCH_SLO_US", "2000"))
    closed_loop = os.environ.get("BENCH_CLOSED_LOOP", "0") == "1"
    result = {}

    tmp = None
    stack = []
    try:
        if rank == 0:
            from binder_amd.harness import free_port, NativeZkd
            from binder_amd.zkclient import ZkConn
            tmp = Path(tempfile.mkdtemp(prefix="binder-bench-"))
            names_file = tmp / "names.txt"
            log(f"starting native zkd + building "
                f"{args.tree_records}-record tree")
            zkd = NativeZkd().start()
            stack.append(zkd.stop)
            zk = ZkConn("127.0.0.1", zkd.port)
            stack.append(zk.close)
            nrec = build_tree(zk, names_file, args.tree_records)
            log(f"{nrec} records; starting {n} binderd process(es)")
            backends, sockdir = start_backends(
                n, tmp, zkd.port,
                last_name=f"h{args.tree_records // 2 - 1}.foo.com")
            stack.append(lambda: [b.stop() for b in backends])
            bal_port = free_port()
            bal = start_balancer(tmp, sockdir, bal_port, workers=workers)
            def stop_bal():
                bal.terminate()
                try:
                    bal.wait(timeout=5)
                except subprocess.TimeoutExpired:
                    bal.kill()
            stack.append(stop_bal)
            wait_balancer_ready(bal_port, n, tmp)
            # fixed chain ramp (setup, untimed): the chain reaches its
            # steady operating point over the first several million
            # queries (affinity pinning, scheduler placement, turbo) —
            # measured +11% at N=8 vs a single warmup step
            log(f"balancer ready on :{bal_port}; ramp + warmup "
                f"{args.warmup} x {q_step} queries")
            blast = BlastDaemon(bal_port, names_file, threads,
                                window, socks)
            stack.append(blast.close)
            capacity = 0.0
            for _ in range(2):
                r = blast.step(q_step)
                capacity = max(capacity, r["qps"])
            rate = 0
            fixed = float(os.environ.get("BENCH_FIXED_RATE", "0"))
            if fixed > 0:
                rate = fixed  # diagnostics: skip calibration
            elif not closed_loop:
                # qps@SLO protocol: fixed offered rate for the timed
                # steps, discovered against the SLO (untimed)
                rate = calibrate_rate(blast, capacity, slo_us)
                log(f"operating point: {rate:.0f} qps offered "
                    f"(capacity {capacity:.0f}, SLO p99<={slo_us}us)")
            for _ in range(args.warmup):
                blast.step(q_step, rate=rate)

        churn_stop = None
        if rank == 0 and args.churn_qps > 0:
            import random
            import threading
            churn_stop = threading.Event()
            n_hosts = args.tree_records // 2
            churn_zk = ZkConn("127.0.0.1", zkd.port)
            stack.append(churn_zk.close)

            def churner():
                rng = random.Random(7)
                interval = 1.0 / args.churn_qps
                i = 0
                while not churn_stop.is_set():
                    h = rng.randrange(n_hosts)
                    churn_zk.put(f"/com/foo/h{h}", json.dumps(
                        {"type": "host",
                         "host": {"address":
                                  f"10.{(h >> 8) & 255}.{h & 255}."
                                  f"{1 + (i % 200)}"}}).encode())
                    i += 1
                    time.sleep(interval)

            churn_thread = threading.Thread(target=churner, daemon=True)
            churn_thread.start()
            stack.append(lambda: (churn_stop.set(), churn_thread.join()))

        barrier()
        cuda_sync()
        t0 = time.perf_counter()
        last = None
        if rank == 0:
            for s in range(args.steps):
                last = blast.step(q_step, rate=rate)
                log(f"step {s + 1}/{args.steps}: "
                    f"{last['qps']:.0f} qps, p99 {last['p99_us']}us")
        barrier()
        cuda_sync()
        elapsed = time.perf_counter() - t0

        # MAX over ranks of the measured wall time
        if dist is not None:
            import torch
            t = torch.tensor([elapsed], dtype=torch.float64)
            dist.all_reduce(t, op=dist.ReduceOp.MAX)
            elapsed = float(t.item())

        if rank == 0:
            total_queries = args.steps * q_step
            qps = total_queries / elapsed
            result = {
                "metric": "dns_queries_per_sec",
                "value": round(qps, 1),
                "unit": "queries/s",
                "n_gpus": n,
                "steps": args.steps,
                "warmup": args.warmup,
                "ms_per_step": round(elapsed * 1000 / args.steps, 3),
                "higher_is_better": True,
                "scaling": "weak",
                # measured reference-architecture Node.js baseline:
                # 138,545 qps/process on the driver box class
                # (BASELINE.md "Measured baseline"); weak scaling =>
                # baseline scales with N
                "vs_baseline": round(qps / (n * 138545.3), 2),
                "dtype": "n/a",
                "data": "synthetic",
                "config": {
                    "model": "binder-dns-zk",
                    "protocol": ("closed-loop" if closed_loop else
                                 f"fixed-rate@SLO(p99<={slo_us}us)"),
                    "offered_qps": int(rate) if rank == 0 else None,
                    "tree_records": args.tree_records,
                    "churn_qps": args.churn_qps,
                    "query_mix": "A+SRV uniform",
                    "queries_per_step": q_step,
                    "global_batch": q_step,
                    "seq_len": None,
                    "parallelism":
                        f"{n} binderd procs behind binder-balancer",
                    "store": "native zkd mirror",
                    "p50_us": last["p50_us"] if last else None,
                    "p99_us": last["p99_us"] if last else None,
                    "timeouts_last_step": last["timeouts"] if last else None,
                    # honesty guard: every counted reply must be a real
                    # NOERROR answer
                    "noerror_frac_last_step":
                        (round(last["noerror"] / max(1, last["received"]),
                               4) if last else None),
                },
            }
            print(json.dumps(result), flush=True)
    finally:
        for fn in reversed(stack):
            try:
                fn()
            except Exception:
                pass
        if tmp is not None:
            shutil.rmtree(tmp, ignore_errors=True)
        if dist is not None:
            dist.destroy_process_group()


if __name__ == "__main__":
    main()
