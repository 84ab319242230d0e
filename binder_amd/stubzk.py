"""In-process stub ZooKeeper server.

Speaks the ZooKeeper client wire protocol (jute framing) well enough to
back the native ZK client + mirror: session handshake, ping, create,
delete, setData, getData/getChildren/exists with one-shot watches, and
watch notifications. BASELINE config 1 and the CI suite run against this
instead of a real ensemble (SURVEY.md §4 "in-proc stub ZK for CI").

Also provides fault injection used by resilience tests:
  - drop_connections(): TCP cut; clients reconnect and resume sessions
  - expire_sessions(): refuse session resume => clients build new
    sessions (the lib/zk.js:45-47 rebuild path)

Thread model: one background IO thread (selectors); the tree API is
thread-safe and fires watches inline.
"""
from __future__ import annotations

import os
import selectors
import socket
import struct
import threading
import time
import zlib
from typing import Dict, Optional, Set

# op codes
OP_CREATE, OP_DELETE, OP_EXISTS, OP_GETDATA, OP_SETDATA = 1, 2, 3, 4, 5
OP_GETCHILDREN, OP_SYNC, OP_PING, OP_GETCHILDREN2 = 8, 9, 11, 12
OP_CLOSE = -11
XID_NOTIFICATION, XID_PING = -1, -2
ZOK, ZNONODE, ZNODEEXISTS, ZNOTEMPTY, ZBADVERSION = 0, -101, -110, -111, -103
EV_CREATED, EV_DELETED, EV_DATA, EV_CHILDREN = 1, 2, 3, 4
STATE_CONNECTED = 3


class _Node:
    __slots__ = ("data", "children", "version", "cversion",
                 "ephemeral_owner")

    def __init__(self, data: bytes = b"", ephemeral_owner: int = 0):
        self.data = data
        self.children: Set[str] = set()
        self.version = 0
        self.cversion = 0
        self.ephemeral_owner = ephemeral_owner


class _Conn:
    def __init__(self, sock: socket.socket):
        self.sock = sock
        self.inbuf = b""
        self.outbuf = b""
        self.handshaken = False
        self.session_id = 0
        self.closed = False
        # one-shot watches this connection registered
        self.data_watches: Set[str] = set()
        self.child_watches: Set[str] = set()
        self.exists_watches: Set[str] = set()


def _parent(path: str) -> str:
    if path == "/":
        return ""
    idx = path.rfind("/")
    return path[:idx] if idx > 0 else "/"


class StubZk:
    def __init__(self, host: str = "127.0.0.1", port: int = 0,
                 session_timeout_ms: int = 30000,
                 txnlog_dir: Optional[str] = None):
        self.host = host
        self._port = port
        self.session_timeout_ms = session_timeout_ms
        self._lock = threading.RLock()
        self._nodes: Dict[str, _Node] = {"/": _Node()}
        self._conns: Set[_Conn] = set()
        self._valid_sessions: Set[int] = set()
        self._next_session = 0x100000001
        self._zxid = 1
        self._sel = selectors.DefaultSelector()
        self._listener: Optional[socket.socket] = None
        self._thread: Optional[threading.Thread] = None
        self._running = False
        self._wake_r, self._wake_w = socket.socketpair()
        self.stats = {"sessions": 0, "ops": 0, "watches_fired": 0}
        # Optional on-disk transaction log in the real ZooKeeper
        # FileTxnLog v2 format (magic ZKLG) — lets zklogcat be tested
        # against logs this stub writes, and gives `binder-amd zkd`
        # restart durability via replay.
        self._txnlog = None
        self._txnlog_path = None
        if txnlog_dir:
            os.makedirs(txnlog_dir, exist_ok=True)
            self._txnlog_path = os.path.join(txnlog_dir, "log.1")
            fresh = not os.path.exists(self._txnlog_path)
            if not fresh:
                good = self._replay(self._txnlog_path)
                # a crash mid-append leaves a truncated/corrupt tail;
                # appending after it would make every later entry
                # unreachable to the next replay (which stops at the
                # first bad record, like ZooKeeper). Truncate to the
                # last fully-verified entry first.
                if good is None:
                    # header never landed: start a fresh log
                    with open(self._txnlog_path, "wb") as tf:
                        tf.write(struct.pack(">iiq", 0x5A4B4C47, 2, 0))
                elif good < os.path.getsize(self._txnlog_path):
                    with open(self._txnlog_path, "r+b") as tf:
                        tf.truncate(good)
                self._maybe_compact()
            self._txnlog = open(self._txnlog_path, "ab")
            if fresh:
                self._txnlog.write(struct.pack(">iiq", 0x5A4B4C47, 2, 0))
                self._txnlog.flush()

    def _maybe_compact(self):
        """Rewrite the log as one create per live node when history
        dominates state (ZooKeeper bounds logs with snapshots +
        rollover; a single-node dev registry can simply compact on
        restart — the log stays valid FileTxnLog v2 for zklogcat)."""
        live = [p for p in self._nodes if p != "/"]
        if self._replayed_entries <= 2 * len(live) + 64:
            return
        tmp = self._txnlog_path + ".compact"
        now = int(time.time() * 1000)
        with open(tmp, "wb") as tf:
            tf.write(struct.pack(">iiq", 0x5A4B4C47, 2, 0))
            for p in sorted(live, key=lambda x: (x.count("/"), x)):
                body = (self._jstr(p) + self._jstr(self._nodes[p].data)
                        + struct.pack(">i", 0) + b"\x00")
                hdr = struct.pack(">qiqqi", 0, 0, self._zxid, now, 1)
                txn = hdr + body
                crc = zlib.adler32(txn) & 0xFFFFFFFF
                tf.write(struct.pack(">qi", crc, len(txn)) + txn
                         + b"\x42")
        os.replace(tmp, self._txnlog_path)

    def _replay(self, path: str):
        """Rebuild the tree from a FileTxnLog (durability for zkd).
        Ephemeral creates are skipped: their sessions are gone, which
        is exactly what a real ZK restart + session expiry yields."""
        self._replayed_entries = 0
        with open(path, "rb") as f:
            data = f.read()
        if len(data) < 16 or \
                struct.unpack_from(">i", data, 0)[0] != 0x5A4B4C47:
            return None
        off = 16
        while off + 12 <= len(data):
            crc, tlen = struct.unpack_from(">qi", data, off)
            if crc == 0 or tlen <= 0 or off + 12 + tlen + 1 > len(data):
                break
            txn = data[off + 12:off + 12 + tlen]
            # adler32 + 0x42 end-of-record marker, like FileTxnLog:
            # stop at the first corrupt entry
            if (crc & 0xFFFFFFFF) != (zlib.adler32(txn) & 0xFFFFFFFF) \
                    or data[off + 12 + tlen] != 0x42:
                break
            off += 12 + tlen + 1
            self._replayed_entries += 1
            (_cid, _cxid, zxid, _t, ttype) = struct.unpack_from(
                ">qiqqi", txn, 0)
            body = txn[32:]  # TxnHeader is 8+4+8+8+4 bytes
            self._zxid = max(self._zxid, zxid)

            def jstr(b, o):
                (n,) = struct.unpack_from(">i", b, o)
                if n < 0:
                    return b"", o + 4
                return b[o + 4:o + 4 + n], o + 4 + n

            if ttype == 1:  # create
                p, o = jstr(body, 0)
                d, o = jstr(body, o)
                (nacl,) = struct.unpack_from(">i", body, o)
                o += 4
                for _ in range(max(nacl, 0)):
                    o += 4
                    _s1, o = jstr(body, o)
                    _s2, o = jstr(body, o)
                ephemeral = body[o] != 0
                path_s = p.decode()
                if ephemeral:
                    continue
                parent = _parent(path_s)
                pn = self._nodes.get(parent)
                if pn is None or path_s in self._nodes:
                    continue
                self._nodes[path_s] = _Node(d)
                pn.children.add(path_s[path_s.rfind("/") + 1:])
            elif ttype == 5:  # setData
                p, o = jstr(body, 0)
                d, o = jstr(body, o)
                n = self._nodes.get(p.decode())
                if n is not None:
                    n.data = d
                    n.version += 1
            elif ttype == 2:  # delete
                p, _ = jstr(body, 0)
                path_s = p.decode()
                n = self._nodes.get(path_s)
                if n is not None and not n.children:
                    del self._nodes[path_s]
                    pn = self._nodes.get(_parent(path_s))
                    if pn is not None:
                        pn.children.discard(
                            path_s[path_s.rfind("/") + 1:])
        return off  # first byte past the last fully-verified entry

    # ------------- lifecycle -------------

    @property
    def port(self) -> int:
        return self._port

    def start(self) -> "StubZk":
        self._listener = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._listener.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._listener.bind((self.host, self._port))
        self._listener.listen(64)
        self._listener.setblocking(False)
        self._port = self._listener.getsockname()[1]
        self._sel.register(self._listener, selectors.EVENT_READ, "accept")
        self._sel.register(self._wake_r, selectors.EVENT_READ, "wake")
        self._running = True
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        return self

    def stop(self):
        self._running = False
        self._wake()
        if self._thread:
            self._thread.join(timeout=5)
            self._thread = None
        with self._lock:
            for c in list(self._conns):
                self._close_conn(c)
            if self._listener:
                try:
                    self._sel.unregister(self._listener)
                except Exception:
                    pass
                self._listener.close()
                self._listener = None
            if self._txnlog:
                self._txnlog.close()
                self._txnlog = None
        self._sel.close()

    def _wake(self):
        try:
            self._wake_w.send(b"x")
        except OSError:
            pass

    # ------------- fault injection -------------

    def drop_connections(self):
        """Cut all TCP connections (sessions stay valid => resume)."""
        with self._lock:
            for c in list(self._conns):
                self._close_conn(c)
        self._wake()

    def expire_sessions(self):
        """Invalidate all sessions and cut connections => clients must
        build fresh sessions. Ephemeral nodes owned by the expired
        sessions are reaped (real-ZK semantics)."""
        with self._lock:
            sessions = set(self._valid_sessions)
            self._valid_sessions.clear()
            for sid in sessions:
                self._reap_ephemerals(sid)
            for c in list(self._conns):
                self._close_conn(c)
        self._wake()

    def connection_count(self) -> int:
        with self._lock:
            return len(self._conns)

    # ------------- tree API (tests/bench drive this) -------------

    def node_count(self) -> int:
        with self._lock:
            return len(self._nodes) - 1  # minus "/"

    def exists(self, path: str) -> bool:
        with self._lock:
            return path in self._nodes

    def get(self, path: str) -> Optional[bytes]:
        with self._lock:
            n = self._nodes.get(path)
            return None if n is None else n.data

    def children(self, path: str) -> Set[str]:
        with self._lock:
            n = self._nodes.get(path)
            return set() if n is None else set(n.children)

    def create(self, path: str, data: bytes = b"null",
               ephemeral_owner: int = 0) -> int:
        with self._lock:
            rc = self._do_create(path, data, ephemeral_owner)
        self._wake()
        return rc

    def mkdirp(self, path: str, data: bytes = b"null"):
        parts = [p for p in path.split("/") if p]
        cur = ""
        with self._lock:
            for p in parts:
                cur += "/" + p
                if cur not in self._nodes:
                    self._do_create(cur, data)
        self._wake()

    def put(self, path: str, data: bytes):
        """mkdirp parents + create-or-set."""
        parent = _parent(path)
        if parent and parent != "/":
            self.mkdirp(parent)
        with self._lock:
            if path in self._nodes:
                self._do_set(path, data)
            else:
                self._do_create(path, data)
        self._wake()

    def set(self, path: str, data: bytes) -> int:
        with self._lock:
            rc = self._do_set(path, data)
        self._wake()
        return rc

    def delete(self, path: str) -> int:
        with self._lock:
            rc = self._do_delete(path)
        self._wake()
        return rc

    def rmr(self, path: str):
        with self._lock:
            doomed = [p for p in self._nodes
                      if p == path or p.startswith(path + "/")]
            for p in sorted(doomed, key=len, reverse=True):
                self._do_delete(p)
        self._wake()

    # ------------- tree internals (lock held) -------------

    def _txn(self, ttype: int, body: bytes, client_id: int = 0):
        if self._txnlog is None:
            return
        hdr = struct.pack(">qiqqi", client_id, 0, self._zxid,
                          int(time.time() * 1000), ttype)
        txn = hdr + body
        crc = zlib.adler32(txn) & 0xFFFFFFFF
        self._txnlog.write(struct.pack(">qi", crc, len(txn)) + txn +
                           b"\x42")
        self._txnlog.flush()

    @staticmethod
    def _jstr(s) -> bytes:
        b = s.encode() if isinstance(s, str) else s
        return struct.pack(">i", len(b)) + b

    def _do_create(self, path: str, data: bytes,
                   ephemeral_owner: int = 0,
                   txn_session: int = 0) -> int:
        if path in self._nodes:
            return ZNODEEXISTS
        parent = _parent(path)
        pn = self._nodes.get(parent)
        if pn is None:
            return ZNONODE
        if pn.ephemeral_owner:
            return -108  # ZNOCHILDRENFOREPHEMERALS
        self._zxid += 1
        eph = b"\x01" if ephemeral_owner else b"\x00"
        self._txn(1, self._jstr(path) + self._jstr(data) +
                  struct.pack(">i", 1) + struct.pack(">i", 31) +
                  self._jstr("world") + self._jstr("anyone") +
                  eph + struct.pack(">i", pn.cversion + 1),
                  client_id=txn_session or ephemeral_owner)
        self._nodes[path] = _Node(data, ephemeral_owner)
        pn.children.add(path[path.rfind("/") + 1:])
        pn.cversion += 1
        self._fire(path, EV_CREATED, exists_only=True)
        self._fire(parent, EV_CHILDREN, child=True)
        return ZOK

    def _do_set(self, path: str, data: bytes,
                txn_session: int = 0) -> int:
        n = self._nodes.get(path)
        if n is None:
            return ZNONODE
        self._zxid += 1
        n.data = data
        n.version += 1
        self._txn(5, self._jstr(path) + self._jstr(data) +
                  struct.pack(">i", n.version), client_id=txn_session)
        self._fire(path, EV_DATA)
        return ZOK

    def _do_delete(self, path: str, txn_session: int = 0) -> int:
        n = self._nodes.get(path)
        if n is None:
            return ZNONODE
        if n.children:
            return ZNOTEMPTY
        self._zxid += 1
        self._txn(2, self._jstr(path), client_id=txn_session)
        del self._nodes[path]
        parent = _parent(path)
        pn = self._nodes.get(parent)
        if pn is not None:
            pn.children.discard(path[path.rfind("/") + 1:])
            pn.cversion += 1
            self._fire(parent, EV_CHILDREN, child=True)
        self._fire(path, EV_DELETED)
        return ZOK

    def _fire(self, path: str, ev: int, child: bool = False,
              exists_only: bool = False):
        """Fire one-shot watches for path/kind (lock held)."""
        payload = struct.pack(">iqi", XID_NOTIFICATION, -1, ZOK) + \
            struct.pack(">ii", ev, STATE_CONNECTED) + _s(path)
        frame = struct.pack(">i", len(payload)) + payload
        for c in self._conns:
            fired = False
            if child:
                if path in c.child_watches:
                    c.child_watches.discard(path)
                    fired = True
            else:
                if not exists_only and path in c.data_watches:
                    c.data_watches.discard(path)
                    fired = True
                if path in c.exists_watches:
                    c.exists_watches.discard(path)
                    fired = True
            if fired:
                c.outbuf += frame
                self.stats["watches_fired"] += 1

    # ------------- IO loop -------------

    def _loop(self):
        while self._running:
            events = self._sel.select(timeout=0.1)
            for key, mask in events:
                what = key.data
                if what == "accept":
                    self._accept()
                elif what == "wake":
                    try:
                        self._wake_r.recv(4096)
                    except OSError:
                        pass
                elif isinstance(what, _Conn):
                    self._service(what, mask)
            # flush pending output (watch notifications from API thread)
            with self._lock:
                for c in list(self._conns):
                    self._try_flush(c)

    def _accept(self):
        while True:
            try:
                sock, _ = self._listener.accept()
            except (BlockingIOError, OSError):
                return
            sock.setblocking(False)
            sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
            c = _Conn(sock)
            with self._lock:
                self._conns.add(c)
            self._sel.register(sock, selectors.EVENT_READ, c)

    def _reap_ephemerals(self, session_id: int):
        """Delete ephemeral znodes owned by a dead session (lock held),
        firing watches — the production mechanism by which dead
        registrars vanish from service discovery."""
        if not session_id:
            return
        doomed = [p for p, n in self._nodes.items()
                  if n.ephemeral_owner == session_id]
        for p in sorted(doomed, key=len, reverse=True):
            self._do_delete(p)

    def expire_session(self, session_id: int):
        """Expire one session: reap its ephemerals + drop its conns."""
        with self._lock:
            self._valid_sessions.discard(session_id)
            self._reap_ephemerals(session_id)
            for c in list(self._conns):
                if c.session_id == session_id:
                    self._close_conn(c)
        self._wake()

    def _close_conn(self, c: _Conn):
        if c.closed:
            return
        c.closed = True
        try:
            self._sel.unregister(c.sock)
        except Exception:
            pass
        try:
            c.sock.close()
        except OSError:
            pass
        self._conns.discard(c)

    def _try_flush(self, c: _Conn):
        if c.closed or not c.outbuf:
            return
        try:
            sent = c.sock.send(c.outbuf)
            c.outbuf = c.outbuf[sent:]
        except (BlockingIOError, InterruptedError):
            pass
        except OSError:
            self._close_conn(c)

    def _service(self, c: _Conn, mask):
        try:
            data = c.sock.recv(65536)
        except (BlockingIOError, InterruptedError):
            return
        except OSError:
            with self._lock:
                self._close_conn(c)
            return
        if not data:
            with self._lock:
                self._close_conn(c)
            return
        c.inbuf += data
        while len(c.inbuf) >= 4:
            (plen,) = struct.unpack(">i", c.inbuf[:4])
            if plen < 0 or plen > (64 << 20):
                with self._lock:
                    self._close_conn(c)
                return
            if len(c.inbuf) < 4 + plen:
                break
            pkt = c.inbuf[4:4 + plen]
            c.inbuf = c.inbuf[4 + plen:]
            with self._lock:
                if not c.handshaken:
                    self._handshake(c, pkt)
                else:
                    self._op(c, pkt)
                self._try_flush(c)
                if c.closed:
                    return

    # ------------- protocol (lock held) -------------

    def _handshake(self, c: _Conn, pkt: bytes):
        r = _Reader(pkt)
        r.i32()             # protocolVersion
        r.i64()             # lastZxidSeen
        timeout = r.i32()
        session_id = r.i64()
        r.buf()             # passwd
        # optional readOnly byte ignored
        if session_id != 0 and session_id not in self._valid_sessions:
            # session expired: respond with sessionId 0 / timeout 0
            body = struct.pack(">iiq", 0, 0, 0) + _b(b"\x00" * 16) + b"\x00"
            c.outbuf += struct.pack(">i", len(body)) + body
            return
        if session_id == 0:
            session_id = self._next_session
            self._next_session += 1
            self._valid_sessions.add(session_id)
            self.stats["sessions"] += 1
            self._zxid += 1
            self._txn(-10, struct.pack(">i", self.session_timeout_ms),
                      client_id=session_id)
        c.session_id = session_id
        c.handshaken = True
        neg = min(timeout or self.session_timeout_ms,
                  self.session_timeout_ms)
        body = struct.pack(">iiq", 0, neg, session_id) + \
            _b(b"\x01" * 16) + b"\x00"
        c.outbuf += struct.pack(">i", len(body)) + body

    def _reply(self, c: _Conn, xid: int, err: int, body: bytes = b""):
        payload = struct.pack(">iqi", xid, self._zxid, err) + body
        c.outbuf += struct.pack(">i", len(payload)) + payload

    def _op(self, c: _Conn, pkt: bytes):
        r = _Reader(pkt)
        xid = r.i32()
        if xid == XID_PING:
            r.i32()
            self._reply(c, XID_PING, ZOK)
            return
        op = r.i32()
        self.stats["ops"] += 1

        if op == OP_CLOSE:
            self._zxid += 1
            self._txn(-11, b"", client_id=c.session_id)
            self._reply(c, xid, ZOK)
            self._try_flush(c)
            self._close_conn(c)
            return

        if op == OP_GETDATA:
            path, watch = r.s(), r.bool()
            n = self._nodes.get(path)
            if n is None:
                self._reply(c, xid, ZNONODE)
                return
            if watch:
                c.data_watches.add(path)
            self._reply(c, xid, ZOK, _b(n.data) + self._stat(path, n))
            return

        if op in (OP_GETCHILDREN, OP_GETCHILDREN2):
            path, watch = r.s(), r.bool()
            n = self._nodes.get(path)
            if n is None:
                self._reply(c, xid, ZNONODE)
                return
            if watch:
                c.child_watches.add(path)
            body = struct.pack(">i", len(n.children))
            for kid in sorted(n.children):
                body += _s(kid)
            if op == OP_GETCHILDREN2:
                body += self._stat(path, n)
            self._reply(c, xid, ZOK, body)
            return

        if op == OP_EXISTS:
            path, watch = r.s(), r.bool()
            n = self._nodes.get(path)
            if watch:
                c.exists_watches.add(path)
                if n is not None:
                    c.data_watches.add(path)
            if n is None:
                self._reply(c, xid, ZNONODE)
            else:
                self._reply(c, xid, ZOK, self._stat(path, n))
            return

        if op == OP_CREATE:
            path = r.s()
            data = r.buf() or b""
            nacl = r.i32()
            for _ in range(max(nacl, 0)):
                r.i32()
                r.s()
                r.s()
            flags = r.i32()
            owner = c.session_id if (flags & 1) else 0  # EPHEMERAL
            rc = self._do_create(path, data, owner,
                                 txn_session=c.session_id)
            self._reply(c, xid, rc, _s(path) if rc == ZOK else b"")
            return

        if op == OP_SETDATA:
            path = r.s()
            data = r.buf() or b""
            r.i32()  # version (-1 = any; stub ignores)
            rc = self._do_set(path, data, txn_session=c.session_id)
            n = self._nodes.get(path)
            self._reply(c, xid, rc,
                        self._stat(path, n) if rc == ZOK else b"")
            return

        if op == OP_DELETE:
            path = r.s()
            r.i32()
            rc = self._do_delete(path, txn_session=c.session_id)
            self._reply(c, xid, rc)
            return

        # unknown op
        self._reply(c, xid, -6)  # ZUNIMPLEMENTED

    def _stat(self, path: str, n: _Node) -> bytes:
        return struct.pack(">qqqqiiiqiiq", 1, self._zxid, 0, 0,
                           n.version, n.cversion, 0, 0, len(n.data),
                           len(n.children), self._zxid)


class _Reader:
    def __init__(self, data: bytes):
        self.d = data
        self.o = 0

    def i32(self) -> int:
        v = struct.unpack_from(">i", self.d, self.o)[0]
        self.o += 4
        return v

    def i64(self) -> int:
        v = struct.unpack_from(">q", self.d, self.o)[0]
        self.o += 8
        return v

    def bool(self) -> bool:
        v = self.d[self.o] != 0
        self.o += 1
        return v

    def s(self) -> str:
        b = self.buf()
        return b.decode("utf-8") if b is not None else ""

    def buf(self) -> Optional[bytes]:
        n = self.i32()
        if n < 0:
            return None
        v = self.d[self.o:self.o + n]
        self.o += n
        return v


def _s(s: str) -> bytes:
    b = s.encode("utf-8")
    return struct.pack(">i", len(b)) + b


def _b(b: bytes) -> bytes:
    return struct.pack(">i", len(b)) + b
