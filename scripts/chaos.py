#!/usr/bin/env python3
"""Chaos run: random failure injection under continuous load.

Full stack (stub ZK, supervisor+adjust with 3 binderd, balancer) while
a query loop measures availability. Events every ~2 s: backend SIGKILL,
ZK connection drops, ZK session expiry, churn bursts, balancer-socket
unlink+recreate (via adjust scale bounce). Asserts:
  - availability stays above 95% (brief per-event dips allowed),
  - the system converges back to fully online at the end,
  - answers remain correct after the dust settles.

usage: chaos.py [seconds] (default 60)
"""
import json
import os
import random
import signal
import subprocess
import sys
import tempfile
import threading
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

from binder_amd.digclient import dig  # noqa: E402
from binder_amd.harness import BALANCERD, free_port, SUPERVISORD, ADJUST  # noqa: E402
from binder_amd.stubzk import StubZk  # noqa: E402

BIN = REPO / "bin"
BASE = 27501


def main():
    seconds = int(sys.argv[1]) if len(sys.argv) > 1 else 60
    rng = random.Random(int(os.environ.get("CHAOS_SEED", "1234")))
    tmp = Path(tempfile.mkdtemp(prefix="chaos-"))
    statedir = tmp / "state"
    statedir.mkdir()
    zk = StubZk().start()
    zk.mkdirp("/com/foo")
    for i in range(500):
        zk.put(f"/com/foo/h{i}", json.dumps(
            {"type": "host",
             "host": {"address": f"10.9.{i // 250}.{i % 250}"}}
        ).encode())

    cfg = tmp / "binder.json"
    cfg.write_text(json.dumps({
        "dnsDomain": "foo.com", "datacenterName": "coal",
        "host": "127.0.0.1"}))
    env = dict(os.environ, LOG_LEVEL="warn", ZK_HOST="127.0.0.1",
               ZK_PORT=str(zk.port))
    sup = subprocess.Popen(
        [str(SUPERVISORD), "-d", str(statedir),
         "-x", str(BIN / "binderd")], env=env,
        stdout=open(tmp / "sup.log", "ab"), stderr=subprocess.STDOUT)
    assert subprocess.run(
        [str(ADJUST), "-i", "3", "-B", str(BASE),
         "-d", str(statedir), "-f", str(cfg), "-S", "zk", "-w", "30"],
        capture_output=True).returncode == 0
    bport = free_port()
    bal = subprocess.Popen(
        [str(BALANCERD), "-p", str(bport),
         "-H", "127.0.0.1", "-s", str(statedir / "sockets"),
         "-r", "100"], env=env,
        stdout=open(tmp / "bal.log", "ab"), stderr=subprocess.STDOUT)
    time.sleep(2)

    stats = {"ok": 0, "fail": 0}
    stop = threading.Event()

    def prober():
        while not stop.is_set():
            h = rng.randrange(500)
            try:
                r = dig(f"h{h}.foo.com", port=bport, timeout=1.0)
                if r.status == "NOERROR" and r.answers:
                    stats["ok"] += 1
                else:
                    stats["fail"] += 1
            except OSError:
                stats["fail"] += 1
            time.sleep(0.01)

    t = threading.Thread(target=prober, daemon=True)
    t.start()

    def pids():
        st = json.loads((statedir / "status.json").read_text())
        return {k: v["pid"] for k, v in st["instances"].items()
                if v["pid"] > 0}

    events = []
    t0 = time.time()
    while time.time() - t0 < seconds:
        ev = rng.choice(["kill", "zkdrop", "zkexpire", "churn",
                         "scale"])
        events.append(ev)
        try:
            if ev == "kill":
                ps = pids()
                if ps:
                    os.kill(rng.choice(list(ps.values())), signal.SIGKILL)
            elif ev == "zkdrop":
                zk.drop_connections()
            elif ev == "zkexpire":
                zk.expire_sessions()
            elif ev == "churn":
                for _ in range(200):
                    h = rng.randrange(500)
                    zk.put(f"/com/foo/h{h}", json.dumps(
                        {"type": "host",
                         "host": {"address":
                                  f"10.9.{rng.randrange(200)}."
                                  f"{h % 250}"}}).encode())
            elif ev == "scale":
                n = rng.choice([2, 3])
                subprocess.run(
                    [str(ADJUST), "-i", str(n),
                     "-B", str(BASE), "-d", str(statedir),
                     "-f", str(cfg), "-S", "zk"],
                    capture_output=True)
        except Exception as e:
            print("event error:", e)
        time.sleep(2)

    # restore to 3 and let it settle
    subprocess.run([str(ADJUST), "-i", "3", "-B",
                    str(BASE), "-d", str(statedir), "-f", str(cfg),
                    "-S", "zk", "-w", "30"], capture_output=True)
    time.sleep(3)
    stop.set()
    t.join()

    ok = True
    total = stats["ok"] + stats["fail"]
    avail = stats["ok"] / max(total, 1)
    if avail < 0.95:
        ok = False
    st = json.loads((statedir / "status.json").read_text())
    online = sum(1 for v in st["instances"].values()
                 if v["state"] == "online")
    if online != 3:
        ok = False
    # final correctness probe
    try:
        r = dig("h1.foo.com", port=bport, timeout=2)
        if r.status != "NOERROR":
            ok = False
    except OSError:
        ok = False

    print(json.dumps({"events": len(events), "probes": total,
                      "availability": round(avail, 4),
                      "final_online": online,
                      "mix": {e: events.count(e)
                              for e in set(events)}}))
    bal.terminate()
    sup.terminate()
    for p in (bal, sup):
        try:
            p.wait(timeout=10)
        except subprocess.TimeoutExpired:
            p.kill()
    zk.stop()
    print("CHAOS", "OK" if ok else "FAIL")
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
