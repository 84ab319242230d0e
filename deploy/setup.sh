#!/bin/sh
# boot/setup.sh equivalent. Two flavors like the reference
# (boot/setup.sh: "manta" multi-process at :66-140, "sdc" single
# process at :146-178), selected with FLAVOR (default manta):
#
#   FLAVOR=manta  N binder processes behind the balancer
#                 (BINDER_PROCS, capped at 32)
#   FLAVOR=sdc    one binderd on :$PORT (default 53), metric_ports
#                 written as the single port+1000 (the reference's
#                 `mdata-put metricPorts 1053`), optional registrar
#                 self-registration (REGISTER_ADDR)
set -eu
PREFIX="${PREFIX:-/opt/binder-amd}"
STATEDIR="${STATEDIR:-/var/run/binder}"
FLAVOR="${FLAVOR:-manta}"
CONFIG="${CONFIG:-$PREFIX/etc/config.json}"

# Render config from a metadata file when provided (the reference's
# config-agent step; sapi_manifests/binder/manifest.json).
if [ -n "${METADATA:-}" ]; then
    python3 "$PREFIX/deploy/render-config.py" \
        "$PREFIX/etc/config.json.in" "$METADATA" > "$CONFIG"
fi

if [ "$FLAVOR" = "sdc" ]; then
    # ---- Triton/SDC flavor: one binder on the privileged port ----
    PORT="${PORT:-53}"
    mkdir -p "$STATEDIR/log"

    "$PREFIX/bin/binderd" -p "$PORT" -f "$CONFIG" \
        >> "$STATEDIR/log/binder.log" 2>&1 &
    echo $! > "$STATEDIR/binderd.pid"

    # cmon-agent discovery equivalent: single metric port (reference
    # writes `metricPorts 1053` for port 53; boot/setup.sh:176)
    echo "$((PORT + 1000))" > "$STATEDIR/metric_ports"

    # Registrar self-registration: binder advertises itself in the
    # registry as an rr_host + _dns._udp SRV service, the same record
    # shape it serves (sapi_manifests/registrar/template:1-30).
    if [ -n "${REGISTER_ADDR:-}" ]; then
        # Registered one level below the zone apex, like the
        # reference's SERVICE_NAME (binder.<datacenter>.<dnsDomain>);
        # the apex itself is refused by the suffix policy.
        DOMAIN=$(python3 -c "import json,sys; c=json.load(open('$CONFIG')); \
print('binder.' + c['datacenterName'] + '.' + c['dnsDomain'])")
        python3 -m binder_amd register "$DOMAIN" "$REGISTER_ADDR" \
            -p "$PORT" --zk-host "${ZK_HOST:-127.0.0.1}" \
            --zk-port "${ZK_PORT:-2181}"
    fi

    echo "binder (sdc flavor) on :$PORT, pid $(cat "$STATEDIR/binderd.pid")"
    echo "metric ports: $(cat "$STATEDIR/metric_ports")"
    exit 0
fi

# ---- Manta flavor: converge N binder processes behind the balancer ----
BINDER_PROCS="${BINDER_PROCS:-4}"
BASE_PORT="${BASE_PORT:-5301}"
[ "$BINDER_PROCS" -gt 32 ] && BINDER_PROCS=32

mkdir -p "$STATEDIR/sockets" "$STATEDIR/instances" "$STATEDIR/log"

"$PREFIX/bin/binder-adjust" \
    -b binder -B "$BASE_PORT" -i "$BINDER_PROCS" \
    -d "$STATEDIR" -f "$CONFIG" -w 60

echo "metric ports: $(cat "$STATEDIR/metric_ports")"
