"""Hostile-input tier for the native zkd (bin/zkd): malformed wire
data must drop the offending connection, never the server — the same
posture the other daemons' hostile suites pin (test_zk_hostile.py
covers the stub server and the native client)."""
import socket
import struct
import time

from zkwire import be32, connect_request, jstr, read_packet, req

from binder_amd.harness import NativeZkd
from binder_amd.zkclient import ZkConn


def _conn(port):
    s = socket.socket()
    s.settimeout(3)
    s.connect(("127.0.0.1", port))
    return s


def _alive(z):
    c = ZkConn("127.0.0.1", z.port)
    c.create("/ok", b"1")
    assert c.get("/ok") == b"1"
    c.delete("/ok")
    c.close()


def test_garbage_bytes_close_connection_only():
    z = NativeZkd().start()
    try:
        for payload in (b"\xff" * 64, b"abcd" + b"\x00" * 16,
                        b"\x7f\xff\xff\xff" + b"x" * 32):
            s = _conn(z.port)
            s.sendall(payload)
            # server must close (oversized/garbage length) or ignore;
            # either way it stays up
            try:
                s.recv(64)
            except (TimeoutError, socket.timeout, OSError):
                pass
            s.close()
        _alive(z)
    finally:
        z.stop()


def test_truncated_jute_request():
    z = NativeZkd().start()
    try:
        s = _conn(z.port)
        s.sendall(connect_request())
        read_packet(s)
        # create with a path string whose declared length exceeds the
        # packet (Reader runs out -> marshalling error, not a crash)
        bad = be32(1) + be32(1) + be32(1000) + b"/x"
        s.sendall(be32(len(bad)) + bad)
        try:
            read_packet(s)
        except (AssertionError, TimeoutError, socket.timeout, OSError):
            pass
        s.close()
        _alive(z)
    finally:
        z.stop()


def test_negative_and_huge_length_prefixes():
    z = NativeZkd().start()
    try:
        for n in (-1, -2147483648, 1 << 30):
            s = _conn(z.port)
            s.sendall(struct.pack(">i", n))
            try:
                s.recv(16)
            except (TimeoutError, socket.timeout, OSError):
                pass
            s.close()
        _alive(z)
    finally:
        z.stop()


def test_connect_disconnect_churn():
    z = NativeZkd().start()
    try:
        for i in range(100):
            s = _conn(z.port)
            if i % 3 == 0:
                s.sendall(connect_request())
                read_packet(s)
            if i % 3 == 1:
                s.sendall(b"\x00\x00")  # partial length prefix
            s.close()
        _alive(z)
    finally:
        z.stop()


def test_watch_heavy_connection_dropped_cleanly():
    """A connection holding thousands of watches must release them all
    when it drops (no lingering bookkeeping affecting later conns)."""
    z = NativeZkd().start()
    try:
        c = ZkConn("127.0.0.1", z.port)
        for i in range(200):
            c.create(f"/w{i}", b"x")
        s = _conn(z.port)
        s.sendall(connect_request())
        read_packet(s)
        for i in range(200):
            s.sendall(req(i + 1, 4, jstr(f"/w{i}") + b"\x01"))
            read_packet(s)
        s.close()  # all 200 data watches die with the conn
        time.sleep(0.2)
        # mutations must not crash into the dead conn's watches
        for i in range(200):
            c.set(f"/w{i}", b"y")
        _alive(z)
        c.close()
    finally:
        z.stop()
