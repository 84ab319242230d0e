"""Recursion upstream anti-spoofing (cache-poisoning resistance).

The reference delegates upstream queries to mname-client, which uses
per-lookup sockets; our native forwarder must provide at least the same
off-path-forgery resistance: a reply is only accepted when its source
address+port match an upstream actually queried, the qid matches the
(randomized) outgoing qid, and the echoed question section matches the
outstanding query (native/server/recursion.cpp).
"""
import json
import socket
import threading
import time

import pytest

from binder_amd import require_native
from binder_amd.harness import BinderProcess


class FakeUpstream:
    """Scriptable upstream resolver on 127.0.0.2.

    `script(query_dict, addr, sock)` is called per received query and
    is responsible for sending whatever responses the test needs.
    """

    def __init__(self, script):
        self.native = require_native()
        self.sock = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        self.sock.bind(("127.0.0.2", 0))
        self.port = self.sock.getsockname()[1]
        self.script = script
        self.seen = []          # (qid, client_src_port) per query
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._run, daemon=True)
        self._thread.start()

    def _run(self):
        self.sock.settimeout(0.2)
        while not self._stop.is_set():
            try:
                data, addr = self.sock.recvfrom(4096)
            except socket.timeout:
                continue
            q = self.native.decode_message(data)
            if q is None:
                continue
            self.seen.append((q["id"], addr[1]))
            self.script(q, addr, self.sock)

    def answer(self, q, addr, sock, address="10.22.0.1", qid=None,
               qname=None):
        resp = {
            "id": q["id"] if qid is None else qid,
            "qr": True,
            "questions": [dict(q["questions"][0])],
            "answers": [{"name": q["questions"][0]["name"], "type": "A",
                         "ttl": 30, "address": address}],
        }
        if qname is not None:
            resp["questions"][0]["name"] = qname
            resp["answers"][0]["name"] = qname
        sock.sendto(self.native.encode_message(resp), addr)

    def close(self):
        self._stop.set()
        self._thread.join()
        self.sock.close()


@pytest.fixture()
def make_binder(tmp_path):
    procs = []

    def factory(uport):
        tree = tmp_path / "local.json"
        tree.write_text(json.dumps({"foo.com": None}))
        b = BinderProcess(
            dns_domain="foo.com", datacenter="dc1",
            store=f"file:{tree}", workdir=tmp_path,
            log_path=str(tmp_path / "binder.log"),
            config={"recursion": {
                "source": "static", "regionName": "r1",
                "dnsDomain": "foo.com", "upstreamPort": uport,
                "dcs": {"dc2": ["127.0.0.2"]},
            }})
        b.start()
        procs.append(b)
        return b

    yield factory
    for p in procs:
        p.stop()


def test_forged_source_address_ignored(make_binder):
    """A reply with the right qid but from a source address that was
    never queried must be dropped; the genuine (slower) upstream answer
    wins."""
    attacker = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
    native = require_native()

    def script(q, addr, sock):
        # Off-path attacker who somehow knows qid + client port but
        # forges from 127.0.0.3 (never queried).
        forged = {
            "id": q["id"], "qr": True,
            "questions": q["questions"],
            "answers": [{"name": q["questions"][0]["name"], "type": "A",
                         "ttl": 30, "address": "6.6.6.6"}],
        }
        attacker.sendto(native.encode_message(forged), addr)
        time.sleep(0.3)
        up.answer(q, addr, sock)

    up = FakeUpstream(script)
    # Attacker binds the same port number on a different address, so
    # even the source-port half matches — only the address check can
    # reject it.
    attacker.bind(("127.0.0.3", up.port))
    try:
        b = make_binder(up.port)
        r = b.dig("web.dc2.foo.com", rd=True, timeout=5)
        assert r.status == "NOERROR"
        assert r.answers[0]["address"] == "10.22.0.1"
    finally:
        attacker.close()
        up.close()


def test_wrong_qid_ignored(make_binder):
    """A reply from the genuine upstream address but with the wrong qid
    must be dropped."""
    def script(q, addr, sock):
        up.answer(q, addr, sock, address="6.6.6.6",
                  qid=(q["id"] + 1) & 0xFFFF)
        time.sleep(0.3)
        up.answer(q, addr, sock)

    up = FakeUpstream(script)
    try:
        b = make_binder(up.port)
        r = b.dig("web.dc2.foo.com", rd=True, timeout=5)
        assert r.status == "NOERROR"
        assert r.answers[0]["address"] == "10.22.0.1"
    finally:
        up.close()


def test_question_mismatch_ignored(make_binder):
    """A reply with matching source and qid but a different echoed
    question section must be dropped (anti-poisoning: the answer can't
    be bound to a different name than was asked)."""
    def script(q, addr, sock):
        up.answer(q, addr, sock, address="6.6.6.6",
                  qname="evil.dc2.foo.com")
        time.sleep(0.3)
        up.answer(q, addr, sock)

    up = FakeUpstream(script)
    try:
        b = make_binder(up.port)
        r = b.dig("web.dc2.foo.com", rd=True, timeout=5)
        assert r.status == "NOERROR"
        assert r.answers[0]["address"] == "10.22.0.1"
    finally:
        up.close()


def test_qids_randomized_and_ports_ephemeral(make_binder):
    """Outgoing qids must not be the old deterministic 1,2,3,... walk,
    and lookups must spread across the outgoing socket pool's
    unpredictable source ports (port+qid is what a forger must
    guess)."""
    def script(q, addr, sock):
        up.answer(q, addr, sock)

    up = FakeUpstream(script)
    try:
        b = make_binder(up.port)
        for i in range(4):
            r = b.dig(f"w{i}.dc2.foo.com", rd=True, timeout=5)
            assert r.status == "NOERROR"
        qids = [s[0] for s in up.seen]
        ports = [s[1] for s in up.seen]
        assert len(qids) == 4
        assert qids != [1, 2, 3, 4], "qids are the deterministic walk"
        # pool-drawn sockets => multiple kernel-assigned source ports
        # in play; all four landing on one port would mean a single
        # fixed socket (the original vulnerability)
        assert len(set(ports)) >= 2, f"one shared source port: {ports}"
    finally:
        up.close()
