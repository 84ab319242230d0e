#!/bin/sh
# boot/setup.sh equivalent (Manta flavor): converge N binder processes
# behind the balancer on this host. BINDER_PROCS defaults to 4, capped
# at 32 like the reference (boot/setup.sh:15).
set -eu
PREFIX="${PREFIX:-/opt/binder-amd}"
STATEDIR="${STATEDIR:-/var/run/binder}"
BINDER_PROCS="${BINDER_PROCS:-4}"
BASE_PORT="${BASE_PORT:-5301}"
[ "$BINDER_PROCS" -gt 32 ] && BINDER_PROCS=32

mkdir -p "$STATEDIR/sockets" "$STATEDIR/instances" "$STATEDIR/log"

"$PREFIX/bin/binder-adjust" \
    -b binder -B "$BASE_PORT" -i "$BINDER_PROCS" \
    -d "$STATEDIR" -f "$PREFIX/etc/config.json" -w 60

echo "metric ports: $(cat "$STATEDIR/metric_ports")"
