"""DNS wire codec unit tests.

The reference has zero codec tests (its mname dep is trusted); SURVEY.md
§7 stage 1 calls for exhaustive codec coverage as the first improvement.
"""
from binder_amd import require_native

n = require_native()


def q(name, qtype="A", rd=False, qid=0x1234):
    return {"id": qid, "rd": rd,
            "questions": [{"name": name, "type": qtype}]}


def test_query_roundtrip():
    wire = n.encode_message(q("foo.example.com", "A", rd=True))
    m = n.decode_message(wire)
    assert m["id"] == 0x1234
    assert m["rd"] is True
    assert m["qr"] is False
    assert m["questions"][0]["name"] == "foo.example.com"
    assert m["questions"][0]["type"] == "A"


def test_response_with_answers_roundtrip():
    msg = q("web.foo.com")
    msg.update({
        "qr": True, "aa": True, "rcode": "NOERROR",
        "answers": [
            {"name": "web.foo.com", "type": "A", "ttl": 30,
             "address": "10.1.2.3"},
            {"name": "web.foo.com", "type": "A", "ttl": 60,
             "address": "10.1.2.4"},
        ],
    })
    m = n.decode_message(n.encode_message(msg))
    assert m["rcode"] == "NOERROR"
    assert [a["address"] for a in m["answers"]] == ["10.1.2.3", "10.1.2.4"]
    assert [a["ttl"] for a in m["answers"]] == [30, 60]


def test_srv_with_additional():
    msg = q("_http._tcp.svc.foo.com", "SRV")
    msg.update({
        "qr": True,
        "answers": [
            {"name": "_http._tcp.svc.foo.com", "type": "SRV", "ttl": 60,
             "target": "lb0.svc.foo.com", "port": 80, "priority": 0,
             "weight": 10},
        ],
        "additionals": [
            {"name": "lb0.svc.foo.com", "type": "A", "ttl": 30,
             "address": "192.168.1.5"},
        ],
    })
    m = n.decode_message(n.encode_message(msg))
    a = m["answers"][0]
    assert (a["target"], a["port"], a["priority"], a["weight"]) == \
        ("lb0.svc.foo.com", 80, 0, 10)
    assert m["additionals"][0]["address"] == "192.168.1.5"


def test_all_record_types_roundtrip():
    msg = q("x.foo.com", "ANY")
    msg.update({
        "qr": True,
        "answers": [
            {"name": "x.foo.com", "type": "AAAA", "ttl": 5,
             "address": "fd00::1"},
            {"name": "x.foo.com", "type": "TXT", "ttl": 5,
             "target": "hello world"},
            {"name": "x.foo.com", "type": "CNAME", "ttl": 5,
             "target": "y.foo.com"},
            {"name": "3.2.1.10.in-addr.arpa", "type": "PTR", "ttl": 5,
             "target": "x.foo.com"},
        ],
        "authorities": [
            {"name": "foo.com", "type": "SOA", "ttl": 30,
             "mname": "foo.com", "rname": "hostmaster.foo.com",
             "minimum": 30},
        ],
    })
    m = n.decode_message(n.encode_message(msg))
    types = [a["type"] for a in m["answers"]]
    assert types == ["AAAA", "TXT", "CNAME", "PTR"]
    assert m["answers"][0]["address"] == "fd00::1"
    assert m["answers"][1]["target"] == "hello world"
    soa = m["authorities"][0]
    assert soa["mname"] == "foo.com" and soa["minimum"] == 30


def test_name_compression_is_applied_and_decoded():
    # 20 answers sharing a long suffix: compression must keep this small.
    name = "member.service.region.datacenter.example.com"
    msg = q(name)
    msg["qr"] = True
    msg["answers"] = [
        {"name": name, "type": "A", "ttl": 30, "address": f"10.0.0.{i}"}
        for i in range(20)
    ]
    wire = n.encode_message(msg)
    # Uncompressed each name is ~46 bytes; compressed answers use 2-byte
    # pointers. 20 answers * (2+10) + header/question << uncompressed.
    assert len(wire) < 400
    m = n.decode_message(wire)
    assert len(m["answers"]) == 20
    assert all(a["name"] == name for a in m["answers"])


def test_udp_truncation_sets_tc():
    name = "svc.foo.com"
    msg = q(name)
    msg["qr"] = True
    msg["answers"] = [
        {"name": name, "type": "A", "ttl": 30, "address": f"10.{i%250}.1.1"}
        for i in range(100)
    ]
    wire = n.encode_message(msg, max_size=512)
    assert len(wire) <= 512
    m = n.decode_message(wire)
    assert m["tc"] is True
    assert m["answers"] == []
    # question preserved so the client can retry over TCP
    assert m["questions"][0]["name"] == name


def test_edns_opt_roundtrip():
    msg = q("a.foo.com")
    msg["additionals"] = [{"name": "", "type": "OPT", "udp_size": 4096}]
    m = n.decode_message(n.encode_message(msg))
    assert m["additionals"][0]["type"] == "OPT"
    assert m["additionals"][0]["udp_size"] == 4096


def test_decode_garbage_returns_none():
    assert n.decode_message(b"") is None
    assert n.decode_message(b"\x00" * 5) is None
    assert n.decode_message(b"\xff" * 2048) is None


def test_decode_compression_loop_rejected():
    # header + a name that is a pointer to itself
    hdr = (0xBEEF).to_bytes(2, "big") + b"\x00\x00" + \
        (1).to_bytes(2, "big") + b"\x00\x00\x00\x00\x00\x00"
    evil = hdr + b"\xc0\x0c" + b"\x00\x01\x00\x01"
    assert n.decode_message(evil) is None


def test_root_and_case_names():
    wire = n.encode_message(q("", "A"))
    m = n.decode_message(wire)
    assert m["questions"][0]["name"] == ""
    wire = n.encode_message(q("WWW.Foo.COM", "A"))
    m = n.decode_message(wire)
    # codec preserves case; policy lowercasing happens in the engine
    assert m["questions"][0]["name"] == "WWW.Foo.COM"


def test_txt_longer_than_255_splits():
    text = "x" * 600
    msg = q("t.foo.com", "TXT")
    msg["qr"] = True
    msg["answers"] = [
        {"name": "t.foo.com", "type": "TXT", "ttl": 5, "target": text}]
    m = n.decode_message(n.encode_message(msg))
    assert m["answers"][0]["target"] == text


def test_overlong_label_is_encode_error_not_truncation():
    """A >63-byte label in a record name (bad store/config data) must
    not be silently truncated on the wire; the encoder answers SERVFAIL
    with the RR sections dropped instead (codec.cpp)."""
    bad = "x" * 70 + ".foo.com"
    msg = q("ok.foo.com", "A")
    msg["qr"] = True
    msg["answers"] = [
        {"name": bad, "type": "A", "ttl": 30, "address": "1.2.3.4"}]
    m = n.decode_message(n.encode_message(msg))
    assert m["rcode"] == "SERVFAIL"
    assert m["answers"] == []
    # the (valid) question survives so the client can correlate
    assert m["questions"][0]["name"] == "ok.foo.com"


def test_63_byte_label_still_encodes():
    name = "y" * 63 + ".foo.com"
    msg = q(name, "A")
    msg["qr"] = True
    msg["answers"] = [
        {"name": name, "type": "A", "ttl": 30, "address": "1.2.3.4"}]
    m = n.decode_message(n.encode_message(msg))
    assert m["rcode"] == "NOERROR"
    assert m["answers"][0]["name"] == name
