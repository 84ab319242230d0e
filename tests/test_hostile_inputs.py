"""Hostile-input hardening of the serving daemons' auxiliary surfaces.

The fuzz suite (test_fuzz.py) covers the DNS decoder; these tests hit
the surfaces an attacker or a broken peer reaches in production that
gcov showed under-covered: the metrics HTTP listener, the balancer's
front sockets and stats socket, and binderd's TCP framing edge cases.
The reference has no equivalent tests."""
import json
import os
import socket
import struct
import subprocess
import time

import pytest

from binder_amd.digclient import dig
from binder_amd.harness import BALANCERD, BinderProcess, free_port

TREE = {
    "foo.com": None,
    "web.foo.com": {"type": "host", "host": {"address": "1.2.3.4"}},
}


@pytest.fixture(scope="module")
def server(tmp_path_factory):
    tmp = tmp_path_factory.mktemp("hostile")
    store = tmp / "tree.json"
    store.write_text(json.dumps(TREE))
    srv = BinderProcess(store=f"file:{store}", workdir=tmp,
                        log_path=str(tmp / "binderd.log"))
    srv.start()
    yield srv
    srv.stop()


def test_metrics_unknown_path_404(server):
    with socket.create_connection(("127.0.0.1", server.metrics_port),
                                  timeout=3) as s:
        s.sendall(b"GET /nope HTTP/1.0\r\n\r\n")
        resp = s.recv(4096)
    assert resp.startswith(b"HTTP/1.0 404") or \
        resp.startswith(b"HTTP/1.1 404")
    # server still serves DNS and metrics afterwards
    assert server.dig("web.foo.com").status == "NOERROR"
    assert "binder_requests_completed" in server.metrics()


def test_metrics_garbage_request(server):
    for payload in (b"\x00" * 64, b"BOGUS\r\n\r\n", b"GET " + b"a" * 8192):
        with socket.create_connection(
                ("127.0.0.1", server.metrics_port), timeout=3) as s:
            s.sendall(payload)
            s.settimeout(3)
            try:
                while s.recv(4096):
                    pass
            except socket.timeout:
                pass  # closed or ignored: either is fine
    assert server.dig("web.foo.com").status == "NOERROR"


def test_metrics_slowloris_does_not_block_dns(server):
    # open a metrics connection and never complete the request
    with socket.create_connection(("127.0.0.1", server.metrics_port),
                                  timeout=3) as s:
        s.sendall(b"GET /metr")
        for _ in range(10):
            assert server.dig("web.foo.com").status == "NOERROR"
        # a parallel well-formed request still succeeds
        assert "binder_requests_completed" in server.metrics()


def test_tcp_zero_length_frame(server):
    with socket.create_connection(("127.0.0.1", server.port),
                                  timeout=3) as s:
        s.sendall(struct.pack(">H", 0))
        # follow with a real query on the same connection: the server
        # must either serve it or close cleanly, never hang
        from binder_amd import require_native
        n = require_native()
        q = n.encode_message({"id": 7, "questions":
                              [{"name": "web.foo.com", "type": "A"}]})
        s.sendall(struct.pack(">H", len(q)) + q)
        s.settimeout(3)
        try:
            hdr = s.recv(2)
            if len(hdr) == 2:
                want = struct.unpack(">H", hdr)[0]
                buf = b""
                while len(buf) < want:
                    chunk = s.recv(want - len(buf))
                    if not chunk:
                        break
                    buf += chunk
                if len(buf) == want:
                    assert n.decode_message(buf)["rcode"] == "NOERROR"
        except socket.timeout:
            pytest.fail("server hung on zero-length TCP frame")
    assert server.dig("web.foo.com").status == "NOERROR"


def test_tcp_huge_declared_length(server):
    # declare a 65535-byte query but send only garbage then stall:
    # the connection may sit until the idle sweep, but other clients
    # must be unaffected
    with socket.create_connection(("127.0.0.1", server.port),
                                  timeout=3) as s:
        s.sendall(struct.pack(">H", 65535) + b"\xde\xad")
        for _ in range(5):
            assert server.dig("web.foo.com").status == "NOERROR"


@pytest.fixture()
def balancer(tmp_path):
    sockdir = tmp_path / "socks"
    sockdir.mkdir()
    store = tmp_path / "tree.json"
    store.write_text(json.dumps(TREE))
    backend = BinderProcess(store=f"file:{store}", workdir=tmp_path,
                            balancer_socket=str(sockdir / "b0"),
                            log_path=str(tmp_path / "b0.log"))
    backend.start()
    port = free_port()
    stats = tmp_path / "stats.sock"
    bal = subprocess.Popen(
        [str(BALANCERD), "-p", str(port), "-H", "127.0.0.1",
         "-s", str(sockdir), "-S", str(stats)],
        env=dict(os.environ, LOG_LEVEL="warn"),
        stdout=open(tmp_path / "bal.log", "ab"),
        stderr=subprocess.STDOUT)
    deadline = time.time() + 20
    while time.time() < deadline:
        try:
            r = dig("web.foo.com", "A", server="127.0.0.1", port=port,
                    timeout=1)
            if r.status == "NOERROR":
                break
        except OSError:
            time.sleep(0.1)
    else:
        pytest.fail("balancer never became ready")
    yield {"port": port, "stats": stats, "sockdir": sockdir,
           "bal": bal}
    bal.terminate()
    bal.wait(timeout=5)
    backend.stop()


def test_balancer_garbage_udp(balancer):
    with socket.socket(socket.AF_INET, socket.SOCK_DGRAM) as s:
        for payload in (b"", b"\x00", b"\xff" * 11, os.urandom(512),
                        os.urandom(4096)):
            s.sendto(payload, ("127.0.0.1", balancer["port"]))
    r = dig("web.foo.com", "A", server="127.0.0.1",
            port=balancer["port"], timeout=3)
    assert r.status == "NOERROR"


def test_balancer_stats_socket_garbage(balancer):
    # the stats socket is read-only for clients; writing junk must not
    # wedge the balancer
    for payload in (b"\x00" * 1024, b"GET / HTTP/1.0\r\n\r\n"):
        with socket.socket(socket.AF_UNIX) as s:
            s.settimeout(2)
            s.connect(str(balancer["stats"]))
            try:
                s.sendall(payload)
            except OSError:
                pass
    # stats still parse and queries still flow
    with socket.socket(socket.AF_UNIX) as s:
        s.settimeout(2)
        s.connect(str(balancer["stats"]))
        st = json.loads(s.recv(1 << 20).decode())
    assert any(b["ok"] for b in st["backends"])
    r = dig("web.foo.com", "A", server="127.0.0.1",
            port=balancer["port"], timeout=3)
    assert r.status == "NOERROR"


def test_balancer_dead_socket_file_in_dir(balancer, tmp_path):
    # a socket path nobody listens on (crashed backend that left its
    # socket behind): the balancer must keep routing around it
    dead = balancer["sockdir"] / "b9"
    lsock = socket.socket(socket.AF_UNIX)
    lsock.bind(str(dead))
    lsock.close()  # file exists, connect() will be refused
    try:
        deadline = time.time() + 5
        ok = 0
        while time.time() < deadline and ok < 10:
            try:
                r = dig("web.foo.com", "A", server="127.0.0.1",
                        port=balancer["port"], timeout=2)
                if r.status == "NOERROR":
                    ok += 1
            except OSError:
                pass
        assert ok >= 10, "balancer stopped serving with a dead socket"
    finally:
        dead.unlink(missing_ok=True)


def test_balancer_tcp_garbage(balancer):
    with socket.create_connection(("127.0.0.1", balancer["port"]),
                                  timeout=3) as s:
        s.sendall(struct.pack(">H", 65535) + os.urandom(64))
    r = dig("web.foo.com", "A", server="127.0.0.1",
            port=balancer["port"], timeout=3, tcp=True)
    assert r.status == "NOERROR"


def test_balancer_sockdir_junk_and_disappearance(balancer, tmp_path):
    """Non-socket files in the socket dir are ignored; the dir
    vanishing (operator error) must not crash the balancer — it keeps
    serving already-connected backends."""
    sockdir = balancer["sockdir"]
    (sockdir / "README").write_text("not a socket")
    (sockdir / "sub").mkdir()
    time.sleep(0.6)  # a couple of rescan intervals
    r = dig("web.foo.com", "A", server="127.0.0.1",
            port=balancer["port"], timeout=3)
    assert r.status == "NOERROR"
    (sockdir / "README").unlink()
    (sockdir / "sub").rmdir()
    # dir vanishing entirely: socket presence IS registration, so all
    # backends read as drained (same as unlink-on-SIGTERM) — the
    # balancer must survive and resume when the dir comes back
    import os as _os
    _os.rename(sockdir, tmp_path / "gone")
    try:
        time.sleep(0.6)
        assert balancer["bal"].poll() is None, "balancer crashed"
    finally:
        _os.rename(tmp_path / "gone", sockdir)
    deadline = time.time() + 10
    while time.time() < deadline:
        try:
            r = dig("web.foo.com", "A", server="127.0.0.1",
                    port=balancer["port"], timeout=1)
            if r.status == "NOERROR":
                break
        except OSError:
            pass
    else:
        pytest.fail("balancer never resumed after dir returned")


def test_balancer_rogue_backend_garbage_frames(balancer, tmp_path):
    """A 'backend' that accepts the balancer's connection and then
    speaks garbage (bad magic, oversized lengths, random bytes) must
    be dropped as a bad peer while service continues via the healthy
    backend."""
    import threading

    rogue_path = balancer["sockdir"] / "b7"
    lsock = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
    lsock.bind(str(rogue_path))
    lsock.listen(4)
    stop = threading.Event()

    def rogue():
        lsock.settimeout(0.3)
        payloads = [b"\xff\xff\xff\xff\xff\xff" * 10,
                    b"\xb5\x02\xff\xff\xff\x7f" + b"x" * 64,
                    os.urandom(512)]
        i = 0
        while not stop.is_set():
            try:
                c, _ = lsock.accept()
            except socket.timeout:
                continue
            try:
                c.sendall(payloads[i % len(payloads)])
                i += 1
                time.sleep(0.1)
                c.close()
            except OSError:
                pass

    t = threading.Thread(target=rogue, daemon=True)
    t.start()
    try:
        # give the balancer a few rescan cycles to meet the rogue
        deadline = time.time() + 4
        ok = 0
        while time.time() < deadline:
            try:
                r = dig("web.foo.com", "A", server="127.0.0.1",
                        port=balancer["port"], timeout=2)
                if r.status == "NOERROR":
                    ok += 1
            except OSError:
                pass
            time.sleep(0.1)
        assert ok >= 20, f"service degraded next to rogue backend ({ok})"
        assert balancer["bal"].poll() is None
    finally:
        stop.set()
        t.join()
        lsock.close()
        rogue_path.unlink(missing_ok=True)
