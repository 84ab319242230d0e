"""Minimal synchronous ZooKeeper client (Python).

The capability counterpart of the reference test helper's raw zkstream
client (test/helper.js:52-61, zkMkdirP 98-129, zkRmr 131-166): lets
tests and operator tools write fixtures into any server speaking the ZK
protocol — our StubZk or a real ensemble. Blocking, one request at a
time; not for the serving path (binderd's native client owns that).
"""
from __future__ import annotations

import socket
import struct
from typing import List

OP_CREATE, OP_DELETE, OP_EXISTS, OP_GETDATA, OP_SETDATA = 1, 2, 3, 4, 5
OP_GETCHILDREN, OP_CLOSE = 8, -11
ZOK, ZNONODE, ZNODEEXISTS, ZNOTEMPTY = 0, -101, -110, -111


class ZkError(Exception):
    def __init__(self, code: int, op: str, path: str = ""):
        super().__init__(f"zk error {code} on {op} {path}")
        self.code = code


def _s(txt: str) -> bytes:
    b = txt.encode()
    return struct.pack(">i", len(b)) + b


def _b(data: bytes) -> bytes:
    return struct.pack(">i", len(data)) + data


class ZkConn:
    def __init__(self, host="127.0.0.1", port=2181, timeout=10.0,
                 session_timeout_ms=30000):
        self.sock = socket.create_connection((host, port), timeout)
        self.sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        self._xid = 0
        req = struct.pack(">iqiq", 0, 0, session_timeout_ms, 0) + \
            _b(b"\x00" * 16) + b"\x00"
        self._send(req)
        resp = self._recv()
        _, timeout_ms, session_id = struct.unpack_from(">iiq", resp, 0)
        if timeout_ms <= 0 or session_id == 0:
            raise ZkError(-112, "connect")
        self.session_id = session_id

    # -------- wire --------

    def _send(self, payload: bytes):
        self.sock.sendall(struct.pack(">i", len(payload)) + payload)

    def _recvn(self, count: int) -> bytes:
        buf = b""
        while len(buf) < count:
            chunk = self.sock.recv(count - len(buf))
            if not chunk:
                raise ConnectionError("zk connection closed")
            buf += chunk
        return buf

    def _recv(self) -> bytes:
        (n,) = struct.unpack(">i", self._recvn(4))
        return self._recvn(n)

    def _call(self, op: int, body: bytes) -> bytes:
        self._xid += 1
        xid = self._xid
        self._send(struct.pack(">ii", xid, op) + body)
        while True:
            resp = self._recv()
            rxid, _zxid, err = struct.unpack_from(">iqi", resp, 0)
            if rxid == -1:   # watch notification: not ours, skip
                continue
            if rxid != xid:
                continue
            if err != ZOK:
                raise ZkError(err, str(op))
            return resp[16:]

    # -------- ops --------

    def create(self, path: str, data: bytes = b"null", flags: int = 0):
        body = _s(path) + _b(data) + struct.pack(">i", 1) + \
            struct.pack(">i", 31) + _s("world") + _s("anyone") + \
            struct.pack(">i", flags)
        self._call(OP_CREATE, body)

    def put(self, path: str, data: bytes):
        """mkdirp parents + create-or-set (stubzk.put parity)."""
        parent = path.rsplit("/", 1)[0]
        if parent and parent != "/":
            self.mkdirp(parent)
        try:
            self.create(path, data)
        except ZkError as e:
            if e.code != -110:  # ZNODEEXISTS
                raise
            self.set(path, data)

    def set(self, path: str, data: bytes, version: int = -1):
        self._call(OP_SETDATA, _s(path) + _b(data) +
                   struct.pack(">i", version))

    def delete(self, path: str, version: int = -1):
        self._call(OP_DELETE, _s(path) + struct.pack(">i", version))

    def get(self, path: str) -> bytes:
        resp = self._call(OP_GETDATA, _s(path) + b"\x00")
        (n,) = struct.unpack_from(">i", resp, 0)
        return resp[4:4 + n] if n > 0 else b""

    def children(self, path: str) -> List[str]:
        resp = self._call(OP_GETCHILDREN, _s(path) + b"\x00")
        (n,) = struct.unpack_from(">i", resp, 0)
        out, off = [], 4
        for _ in range(max(n, 0)):
            (sl,) = struct.unpack_from(">i", resp, off)
            off += 4
            out.append(resp[off:off + sl].decode())
            off += sl
        return out

    def exists(self, path: str) -> bool:
        try:
            self._call(OP_EXISTS, _s(path) + b"\x00")
            return True
        except ZkError as e:
            if e.code == ZNONODE:
                return False
            raise

    # -------- helpers (zkMkdirP / zkRmr parity) --------

    def mkdirp(self, path: str, data: bytes = b"null"):
        cur = ""
        for part in [p for p in path.split("/") if p]:
            cur += "/" + part
            try:
                self.create(cur, data)
            except ZkError as e:
                if e.code != ZNODEEXISTS:
                    raise

    def rmr(self, path: str):
        for kid in self.children(path):
            self.rmr(path.rstrip("/") + "/" + kid)
        try:
            self.delete(path)
        except ZkError as e:
            if e.code != ZNONODE:
                raise

    def ping(self):
        """Keepalive (xid -2, op 11); required to hold a session open
        past the negotiated timeout (e.g. for ephemeral registrations).
        """
        self._send(struct.pack(">ii", -2, 11))
        while True:
            resp = self._recv()
            (rxid,) = struct.unpack_from(">i", resp, 0)
            if rxid == -2:
                return
            if rxid == -1:
                continue  # watch notification
            # a reply to an earlier call would be a protocol bug here
            raise ZkError(-2, "ping", "unexpected xid")

    def close(self):
        try:
            self._send(struct.pack(">ii", self._xid + 1, OP_CLOSE))
        except OSError:
            pass
        self.sock.close()

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()
