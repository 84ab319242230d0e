#!/bin/bash
# Rebuild the native tree with gcov instrumentation, run the CPU test
# suite against the instrumented binaries, and print per-file line
# coverage (numbers quoted in docs/TESTING.md).
#
# Restores clean binaries afterwards.
set -e
cd "$(dirname "$0")/.."
COV=${COV_DIR:-/tmp/binder-amd-cov}
rm -rf "$COV" && mkdir -p "$COV"
make BUILD="$COV/build" \
     CXXFLAGS="-O0 -g --coverage -std=c++20 -fPIC -Wall -Wextra -Wno-unused-parameter -MMD -MP" \
     LDFLAGS="--coverage" -j"$(nproc)"
python -m pytest tests/ -x -q -m "not gpu"
cd "$COV/build"
for src in native/dns/codec native/engine/engine native/engine/store \
           native/zk/client native/zk/mirror native/server/server \
           native/server/recursion native/server/ldap \
           native/server/metrics native/balancer/balancer_main \
           native/adjust/supervisor_main native/adjust/adjust_main \
           native/zkd/zkd_main \
           native/zklog/zklogcat_main; do
  pct=$(gcov -n -o "$(dirname "$src")" "$OLDPWD/${src}.cpp" 2>/dev/null |
        grep -A1 "File.*${src##*/}" | grep "Lines executed" | head -1)
  echo "${src##*/}: $pct"
done
cd "$OLDPWD"
# relink clean binaries
rm -f bin/binderd bin/binder-balancer bin/binder-supervisor \
      bin/binder-adjust bin/zktool bin/zklogcat bin/dnsblast bin/*.so
make -j"$(nproc)"
