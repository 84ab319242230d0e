#!/bin/sh
# Smoke-test the release tarball: unpack to a temp prefix and run the
# binaries from there (catching path assumptions / missing files).
set -eu
cd "$(dirname "$0")/.."
make release > /dev/null
TMP=$(mktemp -d)
trap 'rm -rf "$TMP"' EXIT
tar -C "$TMP" -xzf dist/binder-amd.tar.gz
P="$TMP/binder-amd"

"$P/bin/binderd" -V
"$P/bin/binder-balancer" -h > /dev/null 2>&1 || true
"$P/bin/zklogcat" -h > /dev/null 2>&1 || true
"$P/bin/zktool" -h > /dev/null 2>&1 || true

cat > "$TMP/tree.json" <<'EOT'
{"foo.com": null,
 "web.foo.com": {"type": "host", "host": {"address": "5.6.7.8"}}}
EOT
cat > "$TMP/cfg.json" <<'EOT'
{"dnsDomain": "foo.com", "datacenterName": "t", "port": 28853,
 "host": "127.0.0.1", "metricsPort": 29853}
EOT
"$P/bin/binderd" -f "$TMP/cfg.json" -S "file:$TMP/tree.json" &
BPID=$!
sleep 0.5
PYTHONPATH="$P/lib/python" python3 -c "
from binder_amd.digclient import dig
r = dig('web.foo.com', port=28853)
assert r.status == 'NOERROR' and r.answers[0]['address'] == '5.6.7.8', r
print('release smoke: resolve OK')"
kill $BPID
echo "release artifact OK"
