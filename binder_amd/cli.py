"""binder-amd operator CLI.

Subcommands (the reference's operational tool surface):
  balstat    dump balancer backend/remote state (bin/balstat equivalent
             — reads the stats socket instead of mdb)
  status     show supervisor instance status (svcs-style view)
  dig        wire-level query tool (test/dig.js equivalent)
  register   self-register a binder in ZooKeeper the way the Triton
             registrar does (sapi_manifests/registrar/template: rr_host
             node + _dns._udp SRV service, ttl 60)
  zk         mkdirp/rmr/get/ls against a ZK server (test fixture ops)
  zkd        run the built-in single-node registry daemon (stub ZK
             with txn-log durability) — for self-contained
             deployments without a ZooKeeper ensemble
"""
from __future__ import annotations

import argparse
import json
import socket
import sys
from pathlib import Path

from . import REPO_ROOT


def cmd_balstat(args):
    with socket.socket(socket.AF_UNIX) as s:
        s.settimeout(3)
        s.connect(args.socket)
        data = s.recv(1 << 22).decode()
    st = json.loads(data)
    if args.json:
        print(json.dumps(st, indent=2))
        return 0
    print(f"{'ID':>4} {'OK':>3} {'REMOTES':>8} {'QUERIES':>10} "
          f"{'REPLIES':>10} PATH")
    for b in st["backends"]:
        print(f"{b['id']:>4} {str(b['ok']):>3} {b['remotes']:>8} "
              f"{b['queries']:>10} {b['replies']:>10} {b['path']}")
    print(f"\n{'REMOTE':>16} -> BACKEND")
    for r in st["remotes"]:
        print(f"{r['addr']:>16} -> {r['backend']}")
    print(f"\nudp_queries={st['udp_queries']} "
          f"udp_replies={st['udp_replies']} drops={st['drops']}")
    return 0


def cmd_status(args):
    status = json.loads(
        (Path(args.dir) / "status.json").read_text())
    insts = status.get("instances", {})
    print(f"{'STATE':>9} {'PID':>8} {'RESTARTS':>9} {'PORT':>6} NAME")
    for name in sorted(insts):
        i = insts[name]
        print(f"{i['state']:>9} {i['pid']:>8} {i['restarts']:>9} "
              f"{i.get('port', '-'):>6} {name}")
    return 0


def cmd_dig(args):
    from .digclient import dig
    try:
        r = dig(args.name, args.type, server=args.server, port=args.port,
                rd=args.rd, tcp=args.tcp,
                edns=4096 if args.edns else None)
    except socket.timeout:
        print("; no response (timeout)", file=sys.stderr)
        return 1
    print(f";; status: {r.status}, id: {r['id']}, "
          f"flags: {'aa ' if r['aa'] else ''}{'tc ' if r['tc'] else ''}"
          f"{'rd ' if r['rd'] else ''}{'ra' if r['ra'] else ''}")
    for sec in ("answers", "authorities", "additionals"):
        if r[sec]:
            print(f";; {sec}:")
            for rec in r[sec]:
                print(f";  {json.dumps(rec)}")
    return 0


def cmd_register(args):
    """Write the registrar-format nodes for a binder instance.

    Creates <domain-path> as a service node with the _dns._udp SRV
    service (port 53, ttl 60) and a child rr_host record per address —
    the layout binder itself is discovered by
    (sapi_manifests/registrar/template:1-30).

    With --hold the child is EPHEMERAL and this command stays attached
    as a minimal registrar daemon: the registration disappears the
    moment the process (or its session) dies — the production
    liveness mechanism.
    """
    import time as _time

    from .zkclient import ZkConn, ZkError
    domain = args.domain
    path = "/" + "/".join(reversed(domain.split(".")))
    zk = ZkConn(args.zk_host, args.zk_port)
    try:
        zk.mkdirp(path)
        svc = {
            "type": "service",
            "service": {"srvce": "_dns", "proto": "_udp",
                        "ttl": 60, "port": args.port},
            "ttl": 60,
        }
        zk.set(path, json.dumps(svc).encode())
        host = {
            "type": "rr_host",
            "rr_host": {"address": args.address},
            "ttl": 30,
        }
        child = f"{path}/{args.instance}"
        if args.hold:
            try:
                zk.delete(child)
            except ZkError:
                pass
            zk.create(child, json.dumps(host).encode(), flags=1)
            print(f"registered {args.instance} ({args.address}) under "
                  f"{domain} (ephemeral); holding session — Ctrl-C to "
                  f"deregister", flush=True)
            try:
                while True:
                    _time.sleep(5)
                    zk.ping()
            except KeyboardInterrupt:
                pass
            return 0
        zk.mkdirp(child)
        zk.set(child, json.dumps(host).encode())
        print(f"registered {args.instance} ({args.address}) under "
              f"{domain}")
        return 0
    finally:
        zk.close()


def cmd_zk(args):
    from .zkclient import ZkConn
    with ZkConn(args.zk_host, args.zk_port) as zk:
        if args.op == "mkdirp":
            zk.mkdirp(args.path,
                      args.data.encode() if args.data else b"null")
        elif args.op == "rmr":
            zk.rmr(args.path)
        elif args.op == "get":
            sys.stdout.buffer.write(zk.get(args.path) + b"\n")
        elif args.op == "ls":
            for kid in sorted(zk.children(args.path)):
                print(kid)
        elif args.op == "set":
            zk.set(args.path, args.data.encode())
    return 0


def cmd_zkd(args):
    """Run the registry. The native bin/zkd is the supported
    single-node registry; the in-process Python stub (stubzk) remains
    a CI/test fixture and serves as fallback when the native binary
    has not been built."""
    import os as _os

    native = REPO_ROOT / "bin" / "zkd"
    if native.exists():
        _os.execv(str(native), [str(native), "-H", args.host,
                                "-p", str(args.port),
                                "-d", args.data_dir])

    import signal
    import time as _time

    from .stubzk import StubZk
    print("warning: bin/zkd not built; running the Python stub "
          "(dev fallback)", flush=True)
    zk = StubZk(host=args.host, port=args.port,
                txnlog_dir=args.data_dir)
    zk.start()
    print(f"zkd listening on {args.host}:{zk.port}, "
          f"data in {args.data_dir} ({zk.node_count()} nodes restored)",
          flush=True)
    stop = []
    signal.signal(signal.SIGTERM, lambda *a: stop.append(1))
    try:
        while not stop:
            _time.sleep(0.5)
    except KeyboardInterrupt:
        pass
    zk.stop()
    return 0


def main(argv=None):
    ap = argparse.ArgumentParser(prog="binder-amd")
    sub = ap.add_subparsers(dest="cmd", required=True)

    p = sub.add_parser("balstat")
    p.add_argument("socket", nargs="?",
                   default="/var/run/binder/balancer-stats.sock")
    p.add_argument("--json", action="store_true")
    p.set_defaults(fn=cmd_balstat)

    p = sub.add_parser("status")
    p.add_argument("-d", "--dir", default="/var/run/binder")
    p.set_defaults(fn=cmd_status)

    p = sub.add_parser("dig")
    p.add_argument("name")
    p.add_argument("type", nargs="?", default="A")
    p.add_argument("-s", "--server", default="127.0.0.1")
    p.add_argument("-p", "--port", type=int, default=53)
    p.add_argument("--rd", action="store_true")
    p.add_argument("--tcp", action="store_true")
    p.add_argument("--edns", action="store_true")
    p.set_defaults(fn=cmd_dig)

    p = sub.add_parser("register")
    p.add_argument("domain")
    p.add_argument("address")
    p.add_argument("-i", "--instance", default="binder0")
    p.add_argument("-p", "--port", type=int, default=53)
    p.add_argument("--zk-host", default="127.0.0.1")
    p.add_argument("--zk-port", type=int, default=2181)
    p.add_argument("--hold", action="store_true",
                   help="ephemeral registration; stay attached as a "
                        "minimal registrar daemon")
    p.set_defaults(fn=cmd_register)

    p = sub.add_parser("zkd")
    p.add_argument("-H", "--host", default="127.0.0.1")
    p.add_argument("-p", "--port", type=int, default=2181)
    p.add_argument("-d", "--data-dir", default="/var/lib/binder-zkd")
    p.set_defaults(fn=cmd_zkd)

    p = sub.add_parser("zk")
    p.add_argument("op", choices=["mkdirp", "rmr", "get", "ls", "set"])
    p.add_argument("path")
    p.add_argument("data", nargs="?")
    p.add_argument("--zk-host", default="127.0.0.1")
    p.add_argument("--zk-port", type=int, default=2181)
    p.set_defaults(fn=cmd_zk)

    args = ap.parse_args(argv)
    return args.fn(args)


if __name__ == "__main__":
    sys.exit(main())
