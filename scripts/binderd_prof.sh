#!/bin/sh
# gprof capture of binderd's framed hot path: build an instrumented
# binderd into a scratch dir, drive it through the balancer at
# closed-loop saturation, SIGTERM it (clean exit => gmon.out), and
# print the flat profile. Instrumented binaries never land in bin/
# (docs/DEVNOTES.md invariant).
set -eu
cd "$(dirname "$0")/.."
OUT=${1:-gpurun_out/binderd_gprof.txt}
SCRATCH=$(mktemp -d)
trap 'rm -rf "$SCRATCH"' EXIT

make BUILD="$SCRATCH/obj" \
    CXXFLAGS="-O2 -g -pg -std=c++20 -fPIC -fno-omit-frame-pointer \
              -Wall -Wextra -Wno-unused-parameter -MMD -MP" \
    bin/binderd -j32 > /dev/null
mv bin/binderd "$SCRATCH/binderd-prof"
make bin/binderd -j32 > /dev/null   # restore the clean binary

python3 - "$SCRATCH" <<'EOF'
import json, os, subprocess, sys, time
from pathlib import Path
sys.path.insert(0, ".")
scratch = Path(sys.argv[1])
from binder_amd.harness import BALANCERD, free_port
tmp = scratch / "run"
tmp.mkdir()
tree = tmp / "t.json"
rec = {"foo.com": None}
for i in range(5000):
    rec[f"h{i}.foo.com"] = {"type": "host",
                            "host": {"address": "10.0.0.1"}}
tree.write_text(json.dumps(rec))
sockdir = tmp / "socks"
sockdir.mkdir()
cfg = tmp / "cfg.json"
port = free_port()
cfg.write_text(json.dumps({
    "dnsDomain": "foo.com", "datacenterName": "p", "port": port,
    "host": "127.0.0.1", "metricsPort": free_port()}))
# cwd = tmp so gmon.out lands there
b = subprocess.Popen(
    [str(scratch / "binderd-prof"), "-f", str(cfg),
     "-S", f"file:{tree}", "-b", str(sockdir / "b0")],
    cwd=str(tmp), env=dict(os.environ, LOG_LEVEL="warn"),
    stdout=subprocess.DEVNULL, stderr=subprocess.STDOUT)
bport = free_port()
bal = subprocess.Popen(
    [str(BALANCERD), "-p", str(bport), "-H", "127.0.0.1",
     "-s", str(sockdir), "-r", "200", "-w", "8"],
    env=dict(os.environ, LOG_LEVEL="warn"),
    stdout=subprocess.DEVNULL)
time.sleep(1.5)
names = tmp / "names.txt"
names.write_text("\n".join(f"h{i}.foo.com A" for i in range(5000)))
r = subprocess.run(
    ["bin/dnsblast", "-s", "127.0.0.1", "-p", str(bport),
     "-n", "8000000", "-c", "512", "-t", "10", "-P", "8",
     "-f", str(names), "-B", "127.0.1.1", "-T", "10000"],
    capture_output=True, text=True)
print("# load:", r.stdout.strip()[:160])
bal.terminate()
b.terminate()
b.wait(timeout=10)
print("# gmon.out:", (tmp / "gmon.out").exists())
EOF
gprof -b -p "$SCRATCH/binderd-prof" "$SCRATCH/run/gmon.out" | head -40 > "$OUT"
cat "$OUT"
