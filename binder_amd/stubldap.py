"""In-process stub LDAP server (UFDS stand-in for tests).

Answers simple binds and subtree searches with a configured resolver
list, the way UFDS answers binder's listResolvers(region) search
(reference recursion.js:17-19, 210-219). Plain TCP, or TLS (ldaps)
when cert/key paths are given.
"""
from __future__ import annotations

import socket
import threading
from typing import Dict, List, Optional


def _tlv(tag: int, content: bytes) -> bytes:
    n = len(content)
    if n < 128:
        return bytes([tag, n]) + content
    lb = []
    while n:
        lb.append(n & 0xFF)
        n >>= 8
    return bytes([tag, 0x80 | len(lb)] + list(reversed(lb))) + content


def _int(v: int) -> bytes:
    body = v.to_bytes(max(1, (v.bit_length() + 8) // 8), "big",
                      signed=True)
    return _tlv(0x02, body)


def _enum(v: int) -> bytes:
    return _tlv(0x0A, bytes([v]))


def _octet(s: bytes, tag=0x04) -> bytes:
    return _tlv(tag, s)


def _read_tlv(data: bytes, off: int):
    if off + 2 > len(data):
        return None
    tag = data[off]
    l0 = data[off + 1]
    hdr = 2
    if l0 < 128:
        vlen = l0
    else:
        nb = l0 & 0x7F
        if off + 2 + nb > len(data):
            return None
        vlen = int.from_bytes(data[off + 2:off + 2 + nb], "big")
        hdr = 2 + nb
    if off + hdr + vlen > len(data):
        return None
    return tag, data[off + hdr:off + hdr + vlen], off + hdr + vlen


class StubLdap:
    def __init__(self, host="127.0.0.1", port=0, tls_cert=None,
                 tls_key=None):
        self.host = host
        self._port = port
        self._ssl_ctx = None
        if tls_cert:
            import ssl
            self._ssl_ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
            self._ssl_ctx.load_cert_chain(tls_cert, tls_key)
        self.resolvers: List[Dict[str, str]] = []
        self.binds: List[str] = []
        self.searches: List[str] = []
        self.require_password: Optional[str] = None
        self._sock: Optional[socket.socket] = None
        self._thread: Optional[threading.Thread] = None
        self._running = False

    @property
    def port(self):
        return self._port

    def start(self):
        self._sock = socket.socket()
        self._sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._sock.bind((self.host, self._port))
        self._sock.listen(8)
        self._port = self._sock.getsockname()[1]
        self._sock.settimeout(0.2)
        self._running = True
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        return self

    def stop(self):
        self._running = False
        if self._thread:
            self._thread.join(timeout=5)
        if self._sock:
            self._sock.close()

    def _loop(self):
        while self._running:
            try:
                conn, _ = self._sock.accept()
            except socket.timeout:
                continue
            except OSError:
                return
            threading.Thread(target=self._serve, args=(conn,),
                             daemon=True).start()

    def _serve(self, conn: socket.socket):
        conn.settimeout(10)
        buf = b""
        try:
            if self._ssl_ctx is not None:
                conn = self._ssl_ctx.wrap_socket(conn, server_side=True)
            while True:
                r = _read_tlv(buf, 0)
                if r is None:
                    data = conn.recv(65536)
                    if not data:
                        return
                    buf += data
                    continue
                tag, body, end = r
                buf = buf[end:]
                if tag != 0x30:
                    return
                # messageID
                mr = _read_tlv(body, 0)
                msg_id = int.from_bytes(mr[1], "big")
                op = _read_tlv(body, mr[2])
                op_tag, op_body = op[0], op[1]
                if op_tag == 0x60:  # bind
                    # version, name, simple-auth
                    p = _read_tlv(op_body, 0)
                    nm = _read_tlv(op_body, p[2])
                    pw = _read_tlv(op_body, nm[2])
                    self.binds.append(nm[1].decode())
                    rc = 0
                    if self.require_password is not None and \
                            pw[1].decode() != self.require_password:
                        rc = 49  # invalidCredentials
                    resp = _tlv(0x30, _int(msg_id) + _tlv(
                        0x61, _enum(rc) + _octet(b"") + _octet(b"")))
                    conn.sendall(resp)
                elif op_tag == 0x63:  # search
                    base = _read_tlv(op_body, 0)[1].decode()
                    self.searches.append(base)
                    out = b""
                    for r2 in self.resolvers:
                        attrs = b""
                        for k, v in r2.items():
                            attrs += _tlv(0x30, _octet(k.encode()) +
                                          _tlv(0x31,
                                               _octet(v.encode())))
                        dn = (f"resolver={r2.get('ip', '?')}, " +
                              base).encode()
                        entry = _tlv(0x64, _octet(dn) +
                                     _tlv(0x30, attrs))
                        out += _tlv(0x30, _int(msg_id) + entry)
                    done = _tlv(0x30, _int(msg_id) + _tlv(
                        0x65, _enum(0) + _octet(b"") + _octet(b"")))
                    conn.sendall(out + done)
                elif op_tag == 0x42:  # unbind
                    return
        except (socket.timeout, OSError):
            return
        except Exception:
            return  # TLS handshake failures etc.
        finally:
            conn.close()
