"""Wire-level DNS test client (the test suite's `dig` equivalent).

The reference drives its integration tests through the system `dig`
binary and scrapes its output (/root/reference/test/dig.js). We instead
speak the DNS wire format directly via the native codec, which asserts
on actual bytes rather than dig's presentation layer.
"""
from __future__ import annotations

import random
import socket
import struct
from typing import Optional

from . import require_native


class DigResult(dict):
    @property
    def status(self):
        return self["rcode"]

    @property
    def answers(self):
        return self["answers"]


def dig(name: str, qtype: str = "A", server: str = "127.0.0.1",
        port: int = 1053, rd: bool = False, tcp: bool = False,
        timeout: float = 2.0, edns: Optional[int] = None,
        qid: Optional[int] = None) -> DigResult:
    """Send one DNS query and decode the response.

    Raises socket.timeout when the server does not answer (reference
    tests used dig +time=1 +retry=0 similarly).
    """
    n = require_native()
    msg = {
        "id": qid if qid is not None else random.randrange(1, 65535),
        "rd": rd,
        "questions": [{"name": name, "type": qtype}],
    }
    if edns is not None:
        msg["additionals"] = [{"name": "", "type": "OPT", "udp_size": edns}]
    wire = n.encode_message(msg)

    family = socket.AF_INET6 if ":" in server else socket.AF_INET
    if tcp:
        with socket.socket(family, socket.SOCK_STREAM) as s:
            s.settimeout(timeout)
            s.connect((server, port))
            s.sendall(struct.pack(">H", len(wire)) + wire)
            hdr = _recvn(s, 2)
            (rlen,) = struct.unpack(">H", hdr)
            data = _recvn(s, rlen)
    else:
        with socket.socket(family, socket.SOCK_DGRAM) as s:
            s.settimeout(timeout)
            s.sendto(wire, (server, port))
            while True:
                data, addr = s.recvfrom(65535)
                resp = n.decode_message(data)
                if resp is not None and resp["id"] == msg["id"]:
                    break
            return DigResult(resp)

    resp = n.decode_message(data)
    if resp is None:
        raise ValueError("undecodable DNS response")
    return DigResult(resp)


def _recvn(s: socket.socket, count: int) -> bytes:
    buf = b""
    while len(buf) < count:
        chunk = s.recv(count - len(buf))
        if not chunk:
            raise ConnectionError("short read")
        buf += chunk
    return buf
