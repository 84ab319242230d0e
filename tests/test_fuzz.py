"""Deterministic fuzzing of the wire decoders.

The DNS decoder and the engine face untrusted network input; neither
may crash, hang, or over-read on garbage. (The jute reader only ever
parses our own trusted ZK/stub traffic, but gets a pass too.)
"""
import random
import struct

from binder_amd import require_native

n = require_native()


def test_dns_decoder_random_bytes():
    rng = random.Random(0xB1D3)
    for _ in range(20000):
        size = rng.randrange(0, 128)
        data = rng.randbytes(size)
        n.decode_message(data)  # must not crash; None is fine


def test_dns_decoder_mutated_valid_packets():
    rng = random.Random(1234)
    base = n.encode_message({
        "id": 7, "qr": True,
        "questions": [{"name": "web.svc.foo.com", "type": "SRV"}],
        "answers": [
            {"name": "web.svc.foo.com", "type": "SRV", "ttl": 60,
             "target": "lb0.svc.foo.com", "port": 80},
            {"name": "x.foo.com", "type": "A", "ttl": 30,
             "address": "10.0.0.1"},
        ],
        "authorities": [
            {"name": "foo.com", "type": "SOA", "ttl": 30,
             "mname": "foo.com", "rname": "h.foo.com", "minimum": 30}],
        "additionals": [{"name": "", "type": "OPT", "udp_size": 4096}],
    })
    for _ in range(20000):
        data = bytearray(base)
        for _ in range(rng.randrange(1, 6)):
            pos = rng.randrange(len(data))
            data[pos] = rng.randrange(256)
        n.decode_message(bytes(data))
    # truncations
    for cut in range(len(base)):
        n.decode_message(base[:cut])


def test_engine_fuzzed_queries_via_wire():
    """Random query packets through the full engine path."""
    import json
    rng = random.Random(99)
    e = n.StubEngine("foo.com", "coal", False)
    e.put("foo.com", "null")
    e.put("web.foo.com", json.dumps(
        {"type": "host", "host": {"address": "1.2.3.4"}}))
    valid = n.encode_message(
        {"id": 1, "questions": [{"name": "web.foo.com", "type": "A"}]})
    for _ in range(5000):
        data = bytearray(valid)
        for _ in range(rng.randrange(1, 5)):
            data[rng.randrange(len(data))] = rng.randrange(256)
        e.query_wire(bytes(data), 512)
    for _ in range(5000):
        e.query_wire(rng.randbytes(rng.randrange(0, 80)), 512)
    # still sane afterwards
    r = e.query("web.foo.com", "A")
    assert r["rcode"] == "NOERROR"


def test_extreme_names():
    # label/name length bounds
    long_label = "a" * 63
    name = ".".join([long_label] * 4)  # 255 chars
    wire = n.encode_message(
        {"id": 1, "questions": [{"name": name, "type": "A"}]})
    m = n.decode_message(wire)
    assert m["questions"][0]["name"] == name

    # oversized label gets clamped on encode, not crash
    n.encode_message(
        {"id": 1, "questions": [{"name": "b" * 200, "type": "A"}]})

    # deep pointer chains rejected (crafted)
    hdr = struct.pack(">HHHHHH", 1, 0, 1, 0, 0, 0)
    chain = b""
    base = len(hdr)
    for i in range(30):
        chain += struct.pack(">H", 0xC000 | (base + 2 * (i + 1)))
    evil = hdr + chain + struct.pack(">HH", 1, 1)
    assert n.decode_message(evil) is None
