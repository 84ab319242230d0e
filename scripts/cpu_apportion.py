#!/usr/bin/env python3
"""Apportion the cgroup CPU quota across the bench chain: sample
utime+stime of every process (binderd / balancer / zkd / dnsblast)
while a sustained closed-loop run is in flight, and print cores
consumed per component — the denominator that bounds qps on the
quota-capped bench boxes (profiles/SCALING.md round 2).

usage: cpu_apportion.py [--procs 8] [--workers 8] [--threads 6]
"""
import argparse
import json
import socket
import subprocess
import sys
import tempfile
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

import bench  # noqa: E402

HZ = 100  # USER_HZ


def cpu_of(pid):
    try:
        parts = open(f"/proc/{pid}/stat").read().rsplit(") ", 1)[1]
        f = parts.split()
        return (int(f[11]) + int(f[12])) / HZ  # utime+stime (s)
    except (OSError, IndexError, ValueError):
        return None


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--procs", type=int, default=8)
    ap.add_argument("--workers", type=int, default=8)
    ap.add_argument("--threads", type=int, default=6)
    ap.add_argument("--seconds", type=float, default=6.0)
    ap.add_argument("--no-gso", action="store_true")
    args = ap.parse_args()

    from binder_amd.harness import free_port, NativeZkd
    from binder_amd.zkclient import ZkConn

    tmp = Path(tempfile.mkdtemp(prefix="cpu-apportion-"))
    names_file = tmp / "names.txt"
    zkd = NativeZkd().start()
    conn = ZkConn("127.0.0.1", zkd.port)
    bench.build_tree(conn, names_file, 10000)
    backends, sockdir = bench.start_backends(args.procs, tmp, zkd.port)
    port = free_port()
    bal = bench.start_balancer(tmp, sockdir, port,
                               workers=args.workers)
    bench.wait_balancer_ready(port, args.procs, tmp)

    cmd = [str(REPO / "bin" / "dnsblast"), "-s", "127.0.0.1",
           "-p", str(port), "-n", "100000000", "-c", "256",
           "-t", str(args.threads), "-P", "8", "-f", str(names_file),
           "-B", "127.0.1.1", "-T", "10000"]
    if args.no_gso:
        cmd.append("-g")
    blast = subprocess.Popen(cmd, stdout=subprocess.DEVNULL)

    def balstat():
        with socket.socket(socket.AF_UNIX) as s:
            s.settimeout(2)
            s.connect(str(tmp / "stats.sock"))
            return json.loads(s.recv(1 << 20).decode())
    try:
        time.sleep(2.0)  # ramp
        pids = {"dnsblast": [blast.pid], "balancer": [bal.pid],
                "zkd": [zkd.proc.pid],
                "binderd": [b.proc.pid for b in backends]}
        t0 = time.time()
        before = {k: [cpu_of(p) for p in v] for k, v in pids.items()}
        st0 = balstat()
        time.sleep(args.seconds)
        dt = time.time() - t0
        after = {k: [cpu_of(p) for p in v] for k, v in pids.items()}
        st1 = balstat()
        out = {"window_s": round(dt, 2), "procs": args.procs,
               "workers": args.workers, "threads": args.threads,
               "gso": not args.no_gso,
               "qps": round((st1["udp_replies"] -
                             st0["udp_replies"]) / dt)}
        total = 0.0
        for k in pids:
            cores = sum((a - b) for a, b in
                        zip(after[k], before[k])
                        if a is not None and b is not None) / dt
            out[k + "_cores"] = round(cores, 2)
            total += cores
        out["total_cores"] = round(total, 2)
        try:
            q, p = open("/sys/fs/cgroup/cpu.max").read().split()[:2]
            out["quota_cores"] = (int(q) // int(p)
                                  if q != "max" else None)
        except (OSError, ValueError):
            pass
        print(json.dumps(out), flush=True)
    finally:
        blast.kill()
        blast.wait()
        bal.terminate()
        for b in backends:
            b.stop()
        conn.close()
        zkd.stop()


if __name__ == "__main__":
    main()
