"""Cross-DC recursion: one binderd forwarding misses to another.

Mirrors lib/recursion.js behavior: RD-gated handoff, DC-label routing,
rd-cleared upstream queries, REFUSED on unroutable/empty results.
"""
import json

import pytest

from binder_amd.harness import BinderProcess


@pytest.fixture()
def dcs(tmp_path):
    # upstream binder in "dc2"
    up_tree = tmp_path / "up.json"
    up_tree.write_text(json.dumps({
        "dc2.foo.com": None,
        "svc.dc2.foo.com": {"type": "host",
                            "host": {"address": "10.22.0.1"}},
    }))
    # 127.0.0.2: the recursion module filters out its own NIC addrs
    # (recursion.js:356-376), and 127.0.0.1 IS one — same as reference.
    upstream = BinderProcess(dns_domain="dc2.foo.com", datacenter="dc2",
                             host="127.0.0.2",
                             store=f"file:{up_tree}", workdir=tmp_path,
                             log_path=str(tmp_path / "up.log"))
    upstream.start()

    # local binder in "dc1" with recursion pointing at upstream
    local_tree = tmp_path / "local.json"
    local_tree.write_text(json.dumps({
        "foo.com": None,
        "web.dc1.foo.com": {"type": "host",
                            "host": {"address": "10.11.0.1"}},
    }))
    local = BinderProcess(
        dns_domain="foo.com", datacenter="dc1",
        store=f"file:{local_tree}", workdir=tmp_path,
        log_path=str(tmp_path / "local.log"),
        config={"recursion": {
            "source": "static",
            "regionName": "region-1",
            "dnsDomain": "foo.com",
            "upstreamPort": upstream.port,
            "dcs": {"dc2": ["127.0.0.2"]},
        }})
    local.start()
    yield local, upstream
    local.stop()
    upstream.stop()


def test_local_hit_still_served(dcs):
    local, _ = dcs
    r = local.dig("web.dc1.foo.com")
    assert r.status == "NOERROR"


def test_miss_with_rd_forwards_to_dc(dcs):
    local, _ = dcs
    r = local.dig("svc.dc2.foo.com", rd=True, timeout=5)
    assert r.status == "NOERROR"
    assert r.answers[0]["address"] == "10.22.0.1"
    assert r.answers[0]["name"] == "svc.dc2.foo.com"


def test_miss_without_rd_refused(dcs):
    local, _ = dcs
    r = local.dig("svc.dc2.foo.com", rd=False)
    assert r.status == "REFUSED"


def test_unknown_dc_refused(dcs):
    local, _ = dcs
    r = local.dig("svc.dc9.foo.com", rd=True, timeout=5)
    assert r.status == "REFUSED"


def test_upstream_miss_refused(dcs):
    local, _ = dcs
    r = local.dig("missing.dc2.foo.com", rd=True, timeout=6)
    assert r.status == "REFUSED"


def test_upstream_sees_rd_cleared(dcs):
    """The forwarded query must have RD cleared (recursion.js:258-261) —
    otherwise the upstream would recurse again. Upstream logs record the
    query; a REFUSED (not forwarded) on its side proves rd was off
    because the upstream has no recursion configured anyway; instead we
    check from the upstream log that the query arrived."""
    local, upstream = dcs
    local.dig("svc.dc2.foo.com", rd=True, timeout=5)
    import time
    time.sleep(0.2)
    log = (upstream.log_path and open(upstream.log_path).read()) or ""
    assert "svc.dc2.foo.com" in log


def test_ptr_recursion_fans_out(dcs):
    """PTR misses fan out to all known DC resolvers
    (recursion.js:346-354) and accepted PTR answers come back under the
    original name."""
    local, upstream = dcs
    r = local.dig("1.0.22.10.in-addr.arpa", "PTR", rd=True, timeout=6)
    assert r.status == "NOERROR"
    assert r.answers[0]["type"] == "PTR"
    assert r.answers[0]["target"] == "svc.dc2.foo.com"
    assert r.answers[0]["name"] == "1.0.22.10.in-addr.arpa"


def test_srv_forwarding_passthrough(tmp_path):
    """SRV answers from the upstream are re-materialized with
    target/port/priority/weight (recursion.js:311-315)."""
    import json as _json
    up_tree = tmp_path / "up2.json"
    up_tree.write_text(_json.dumps({
        "dc2.foo.com": None,
        "s.dc2.foo.com": {"type": "service",
                          "service": {"srvce": "_m", "proto": "_tcp",
                                      "port": 123, "ttl": 60}},
        "m0.s.dc2.foo.com": {"type": "rr_host",
                             "rr_host": {"address": "10.22.9.9"}},
    }))
    upstream = BinderProcess(dns_domain="dc2.foo.com", datacenter="dc2",
                             host="127.0.0.3", store=f"file:{up_tree}",
                             workdir=tmp_path)
    upstream.start()
    local_tree = tmp_path / "local2.json"
    local_tree.write_text("{\"foo.com\": null}")
    local = BinderProcess(
        dns_domain="foo.com", datacenter="dc1",
        store=f"file:{local_tree}", workdir=tmp_path,
        config={"recursion": {
            "source": "static", "regionName": "r1",
            "dnsDomain": "foo.com", "upstreamPort": upstream.port,
            "dcs": {"dc2": ["127.0.0.3"]}}})
    local.start()
    try:
        r = local.dig("_m._tcp.s.dc2.foo.com", "SRV", rd=True, timeout=6)
        assert r.status == "NOERROR"
        a = r.answers[0]
        assert a["type"] == "SRV"
        assert a["port"] == 123
        assert a["target"] == "m0.s.dc2.foo.com"
        assert a["name"] == "_m._tcp.s.dc2.foo.com"
    finally:
        local.stop()
        upstream.stop()


def test_recursion_over_tcp(dcs):
    """Misses arriving over TCP also forward (async reply on the TCP
    conn)."""
    local, _ = dcs
    r = local.dig("svc.dc2.foo.com", rd=True, tcp=True, timeout=6)
    assert r.status == "NOERROR"
    assert r.answers[0]["address"] == "10.22.0.1"


def test_upstream_blackhole_times_out_refused(tmp_path):
    """Resolvers that never answer: the 3 s upstream deadline
    (recursion.js:257) must produce REFUSED, not a hang, and the
    server stays healthy."""
    import time
    local_tree = tmp_path / "local.json"
    local_tree.write_text(json.dumps({
        "foo.com": None,
        "web.dc1.foo.com": {"type": "host",
                            "host": {"address": "10.11.0.1"}}}))
    local = BinderProcess(
        dns_domain="foo.com", datacenter="dc1",
        store=f"file:{local_tree}", workdir=tmp_path,
        config={"recursion": {
            "source": "static", "regionName": "r1",
            "dnsDomain": "foo.com", "upstreamPort": 9,  # discard port
            "dcs": {"dc2": ["127.0.0.2"]},
        }})
    local.start()
    try:
        t0 = time.time()
        r = local.dig("gone.dc2.foo.com", rd=True, timeout=8)
        elapsed = time.time() - t0
        assert r.status == "REFUSED"
        assert elapsed < 6, f"deadline not enforced ({elapsed:.1f}s)"
        assert local.dig("web.dc1.foo.com").status == "NOERROR"
    finally:
        local.stop()


def test_upstream_answer_type_filter(tmp_path):
    """Upstream answers outside A/AAAA/TXT/PTR/CNAME/SRV are dropped
    (recursion.js:299-323): a response carrying A + SOA in the answer
    section forwards only the A."""
    import socket
    import threading
    from binder_amd import require_native
    n = require_native()

    usock = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
    usock.bind(("127.0.0.2", 0))
    uport = usock.getsockname()[1]
    stop = threading.Event()

    def fake_upstream():
        usock.settimeout(0.3)
        while not stop.is_set():
            try:
                data, addr = usock.recvfrom(4096)
            except socket.timeout:
                continue
            q = n.decode_message(data)
            if q is None:
                continue
            resp = {
                "id": q["id"], "qr": True,
                "questions": q["questions"],
                "answers": [
                    {"name": q["questions"][0]["name"], "type": "A",
                     "ttl": 30, "address": "10.66.0.1"},
                    {"name": q["questions"][0]["name"], "type": "SOA",
                     "ttl": 30, "mname": "x.dc2.foo.com",
                     "rname": "hostmaster.dc2.foo.com", "minimum": 30},
                ],
            }
            usock.sendto(n.encode_message(resp), addr)

    t = threading.Thread(target=fake_upstream, daemon=True)
    t.start()

    local_tree = tmp_path / "local.json"
    local_tree.write_text('{"foo.com": null}')
    local = BinderProcess(
        dns_domain="foo.com", datacenter="dc1",
        store=f"file:{local_tree}", workdir=tmp_path,
        config={"recursion": {
            "source": "static", "regionName": "r1",
            "dnsDomain": "foo.com", "upstreamPort": uport,
            "dcs": {"dc2": ["127.0.0.2"]},
        }})
    local.start()
    try:
        r = local.dig("mixed.dc2.foo.com", rd=True, timeout=5)
        assert r.status == "NOERROR"
        assert [a["type"] for a in r.answers] == ["A"]
        assert r.answers[0]["address"] == "10.66.0.1"
    finally:
        local.stop()
        stop.set()
        t.join(timeout=3)
        usock.close()


def test_zk_registry_typed_and_legacy_resolver_schema(tmp_path):
    """The zk resolver-registry source accepts both shapes documented
    in recursion.hpp: the typed {"type":"resolver","resolver":
    {"datacenter","address"}} record (explicit dc) and the legacy
    "<dc>-<n>" host-named form."""
    import time

    from binder_amd.stubzk import StubZk

    # upstream binder answering for dc2 and dc3 names
    up_tree = tmp_path / "up.json"
    up_tree.write_text(json.dumps({
        "foo.com": None,
        "svc.dc2.foo.com": {"type": "host",
                            "host": {"address": "10.22.0.2"}},
        "svc.dc3.foo.com": {"type": "host",
                            "host": {"address": "10.33.0.3"}}}))
    upstream = BinderProcess(dns_domain="foo.com", datacenter="up",
                             host="127.0.0.2", store=f"file:{up_tree}",
                             workdir=tmp_path)
    upstream.start()

    zk = StubZk().start()
    try:
        zk.mkdirp("/com/foo/resolvers")
        # typed form: dc comes from the record, name is arbitrary
        zk.put("/com/foo/resolvers/anything", json.dumps(
            {"type": "resolver",
             "resolver": {"datacenter": "dc2",
                          "address": "127.0.0.2"}}).encode())
        # legacy form: dc parsed from the "<dc>-<n>" name
        zk.put("/com/foo/resolvers/dc3-0", json.dumps(
            {"type": "host",
             "host": {"address": "127.0.0.2"}}).encode())

        local = BinderProcess(
            dns_domain="foo.com", datacenter="dc1",
            store="zk", zk_host="127.0.0.1", zk_port=zk.port,
            workdir=tmp_path, log_path=str(tmp_path / "tl.log"),
            config={"recursion": {
                "source": "zk", "regionName": "r1",
                "dnsDomain": "foo.com",
                "registryDomain": "resolvers.foo.com",
                "upstreamPort": upstream.port,
            }})
        local.start()
        try:
            deadline = time.time() + 20
            r = None
            while time.time() < deadline:
                r = local.dig("svc.dc2.foo.com", rd=True, timeout=5)
                if r.status == "NOERROR":
                    break
                time.sleep(0.5)
            assert r.status == "NOERROR", \
                open(str(tmp_path / "tl.log")).read()[-1500:]
            assert r.answers[0]["address"] == "10.22.0.2"
            r3 = local.dig("svc.dc3.foo.com", rd=True, timeout=5)
            assert r3.status == "NOERROR"
            assert r3.answers[0]["address"] == "10.33.0.3"
        finally:
            local.stop()
    finally:
        zk.stop()
        upstream.stop()
