"""Resolution-engine policy matrix tests.

Pins the reference's deliberate (and deliberately RFC-noncompliant)
behavior, from /root/reference/lib/server.js and the reference test
matrix (test/host.test.js, test/service.test.js, test/database.test.js).
"""
import json

import pytest

from binder_amd import require_native

n = require_native()


def mkengine(dns_domain="foo.com", dc="coal", recursion=False):
    return n.StubEngine(dns_domain, dc, recursion)


def put(e, domain, obj):
    e.put(domain, json.dumps(obj))


HOST = {"type": "host", "host": {"address": "192.168.0.1"}}


@pytest.fixture
def eng():
    e = mkengine()
    put(e, "foo.com", None)
    put(e, "bar.foo.com", None)
    put(e, "web.bar.foo.com", HOST)
    return e


# --- host.test.js parity ---

def test_host_a_record(eng):
    r = eng.query("web.bar.foo.com", "A")
    assert r["rcode"] == "NOERROR"
    assert len(r["answers"]) == 1
    a = r["answers"][0]
    assert a["address"] == "192.168.0.1"
    assert a["ttl"] == 30  # default TTL (server.js:270)
    assert a["name"] == "web.bar.foo.com"


def test_host_ptr_forward_reverse(eng):
    r = eng.query("1.0.168.192.in-addr.arpa", "PTR")
    assert r["rcode"] == "NOERROR"
    assert r["answers"][0]["target"] == "web.bar.foo.com"


def test_ptr_unknown_ip_refused(eng):
    r = eng.query("2.0.168.192.in-addr.arpa", "PTR")
    assert r["rcode"] == "REFUSED"


def test_ptr_not_arpa_refused(eng):
    assert eng.query("1.0.168.192.in-addr.com", "PTR")["rcode"] == "REFUSED"


def test_ptr_truncated_arpa_refused(eng):
    assert eng.query("arpa", "PTR")["rcode"] == "REFUSED"


def test_ptr_ipv6_arpa_refused(eng):
    assert eng.query("1.0.0.0.ip6.arpa", "PTR")["rcode"] == "REFUSED"


# --- service.test.js parity ---

@pytest.fixture
def svc():
    e = mkengine()
    put(e, "foo.com", None)
    put(e, "svc.foo.com", {
        "type": "service",
        "service": {"srvce": "_http", "proto": "_tcp", "port": 80,
                    "ttl": 60},
        "ttl": 60,
    })
    for i in range(3):
        put(e, f"host{i}.svc.foo.com",
            {"type": "host", "host": {"address": f"10.0.0.{i}"}})
    for i in range(2):
        put(e, f"lb{i}.svc.foo.com",
            {"type": "load_balancer",
             "load_balancer": {"address": f"10.0.1.{i}"}})
    return e


def test_service_a_only_serves_member_types(svc):
    """A-for-service returns ONLY the load_balancer members — 'host'
    children are not in the filter set (server.js:352-360)."""
    r = svc.query("svc.foo.com", "A")
    assert r["rcode"] == "NOERROR"
    addrs = sorted(a["address"] for a in r["answers"])
    assert addrs == ["10.0.1.0", "10.0.1.1"]
    assert all(a["ttl"] == 60 for a in r["answers"])


def test_service_srv(svc):
    r = svc.query("_http._tcp.svc.foo.com", "SRV")
    assert r["rcode"] == "NOERROR"
    assert len(r["answers"]) == 2
    for a in r["answers"]:
        assert a["type"] == "SRV"
        assert a["port"] == 80
        assert a["name"] == "_http._tcp.svc.foo.com"
        assert a["target"] in ("lb0.svc.foo.com", "lb1.svc.foo.com")
        assert a["ttl"] == 60
    # additional A records for targets
    adds = {x["name"]: x["address"] for x in r["additionals"]}
    assert adds == {"lb0.svc.foo.com": "10.0.1.0",
                    "lb1.svc.foo.com": "10.0.1.1"}


def test_srv_wrong_proto_nxdomain(svc):
    r = svc.query("_http._udp.svc.foo.com", "SRV")
    assert r["rcode"] == "NXDOMAIN"


def test_srv_unknown_service_refused(svc):
    r = svc.query("_http._tcp.nosvc.foo.com", "SRV")
    assert r["rcode"] == "REFUSED"


def test_direct_member_a(svc):
    r = svc.query("lb0.svc.foo.com", "A")
    assert r["rcode"] == "NOERROR"
    assert r["answers"][0]["address"] == "10.0.1.0"
    assert r["answers"][0]["ttl"] == 30


def test_unknown_name_refused_not_nxdomain(svc):
    r = svc.query("nothere.foo.com", "A")
    assert r["rcode"] == "REFUSED"


def test_srv_on_non_service_nodata_with_soa(svc):
    r = svc.query("_http._tcp.lb0.svc.foo.com", "SRV")
    assert r["rcode"] == "NOERROR"
    assert r["answers"] == []
    assert r["authorities"][0]["type"] == "SOA"
    assert r["authorities"][0]["mname"] == "foo.com"


# --- database.test.js parity ---

def test_database_record_parses_url():
    e = mkengine()
    put(e, "foo.com", None)
    put(e, "pg.foo.com", {
        "type": "database",
        "database": {
            "primary": "tcp://postgres@10.99.99.14:5432/postgres",
        },
        "ttl": 20,
    })
    r = e.query("pg.foo.com", "A")
    assert r["rcode"] == "NOERROR"
    assert r["answers"][0]["address"] == "10.99.99.14"
    assert r["answers"][0]["ttl"] == 20


# --- policy edge cases beyond the reference suite ---

def test_outside_dns_domain_refused(eng):
    assert eng.query("web.bar.other.com", "A")["rcode"] == "REFUSED"


def test_suffix_check_case_sensitive_pre_lowercase(eng):
    # matches reference ordering: suffix test happens before lowercasing
    assert eng.query("web.bar.FOO.COM", "A")["rcode"] == "REFUSED"
    # but case inside the owned part is folded
    r = eng.query("WEB.bar.foo.com", "A")
    assert r["rcode"] == "NOERROR"


def test_invalid_chars_refused(eng):
    assert eng.query("we$b.bar.foo.com", "A")["rcode"] == "REFUSED"


def test_store_not_ready_servfail(eng):
    eng.set_ready(False)
    assert eng.query("web.bar.foo.com", "A")["rcode"] == "SERVFAIL"
    assert eng.query("1.0.168.192.in-addr.arpa", "PTR")["rcode"] == \
        "SERVFAIL"


def test_unsupported_types_notimp(eng):
    for t in ("AAAA", "TXT", "MX", "NS", "CNAME", "ANY"):
        assert eng.query("web.bar.foo.com", t)["rcode"] == "NOTIMP", t


def test_invalid_record_servfail():
    e = mkengine()
    put(e, "foo.com", None)
    put(e, "bad.foo.com", {"type": "host"})  # record[type] missing
    assert e.query("bad.foo.com", "A")["rcode"] == "SERVFAIL"
    put(e, "bad2.foo.com", {"type": 42})
    assert e.query("bad2.foo.com", "A")["rcode"] == "SERVFAIL"
    # node exists but its data is JSON null (intermediate nodes);
    # note the apex "foo.com" itself fails the ".foo.com" suffix test
    # (server.js:158) and is REFUSED before lookup.
    put(e, "mid.foo.com", None)
    assert e.query("mid.foo.com", "A")["rcode"] == "SERVFAIL"
    assert e.query("foo.com", "A")["rcode"] == "REFUSED"


def test_unparseable_data_keeps_previous():
    e = mkengine()
    put(e, "foo.com", None)
    put(e, "h.foo.com", HOST)
    e.put("h.foo.com", "{not json")
    assert e.query("h.foo.com", "A")["answers"][0]["address"] == \
        "192.168.0.1"
    e.put("h.foo.com", json.dumps("a string"))
    assert e.query("h.foo.com", "A")["answers"][0]["address"] == \
        "192.168.0.1"


def test_ttl_precedence_chain():
    e = mkengine()
    put(e, "foo.com", None)
    put(e, "a.foo.com", {"type": "host",
                         "host": {"address": "1.1.1.1"}})
    assert e.query("a.foo.com", "A")["answers"][0]["ttl"] == 30
    put(e, "b.foo.com", {"type": "host", "ttl": 99,
                         "host": {"address": "1.1.1.2"}})
    assert e.query("b.foo.com", "A")["answers"][0]["ttl"] == 99
    put(e, "c.foo.com", {"type": "host", "ttl": 99,
                         "host": {"address": "1.1.1.3", "ttl": 7}})
    # deepest wins
    assert e.query("c.foo.com", "A")["answers"][0]["ttl"] == 7


def test_nested_service_service_ttl():
    e = mkengine()
    put(e, "foo.com", None)
    put(e, "s.foo.com", {
        "type": "service",
        "service": {"service": {"srvce": "_moray", "proto": "_tcp",
                                "port": 2020, "ttl": 120}},
    })
    put(e, "m0.s.foo.com", {"type": "moray_host",
                            "moray_host": {"address": "10.2.0.1"}})
    r = e.query("_moray._tcp.s.foo.com", "SRV")
    assert r["rcode"] == "NOERROR"
    assert r["answers"][0]["ttl"] == 120
    assert r["answers"][0]["port"] == 2020


def test_service_a_uses_min_of_service_and_member_ttl():
    e = mkengine()
    put(e, "foo.com", None)
    put(e, "s.foo.com", {
        "type": "service",
        "service": {"srvce": "_x", "proto": "_tcp", "port": 1, "ttl": 60},
    })
    put(e, "m0.s.foo.com", {"type": "rr_host", "ttl": 10,
                            "rr_host": {"address": "10.3.0.1"}})
    put(e, "m1.s.foo.com", {"type": "rr_host", "ttl": 90,
                            "rr_host": {"address": "10.3.0.2"}})
    r = e.query("s.foo.com", "A")
    ttls = {a["address"]: a["ttl"] for a in r["answers"]}
    assert ttls == {"10.3.0.1": 10, "10.3.0.2": 60}  # min(60, member)


def test_multi_port_member_gets_one_srv_per_port():
    e = mkengine()
    put(e, "foo.com", None)
    put(e, "s.foo.com", {
        "type": "service",
        "service": {"srvce": "_x", "proto": "_tcp", "port": 1},
    })
    put(e, "m0.s.foo.com", {
        "type": "rr_host",
        "rr_host": {"address": "10.3.0.1", "ports": [53, 8053]}})
    r = e.query("_x._tcp.s.foo.com", "SRV")
    ports = sorted(a["port"] for a in r["answers"])
    assert ports == [53, 8053]
    assert len(r["additionals"]) == 1  # one A per member, not per port


def test_member_with_null_address_skipped():
    e = mkengine()
    put(e, "foo.com", None)
    put(e, "s.foo.com", {
        "type": "service",
        "service": {"srvce": "_x", "proto": "_tcp", "port": 1},
    })
    put(e, "m0.s.foo.com", {"type": "rr_host",
                            "rr_host": {"address": None}})
    put(e, "m1.s.foo.com", {"type": "rr_host",
                            "rr_host": {"address": "10.3.0.9"}})
    r = e.query("s.foo.com", "A")
    assert [a["address"] for a in r["answers"]] == ["10.3.0.9"]


def test_invalid_member_servfail_partial():
    e = mkengine()
    put(e, "foo.com", None)
    put(e, "s.foo.com", {
        "type": "service",
        "service": {"srvce": "_x", "proto": "_tcp", "port": 1},
    })
    put(e, "m0.s.foo.com", {"type": "rr_host", "rr_host": None})
    r = e.query("s.foo.com", "A")
    assert r["rcode"] == "SERVFAIL"


def test_service_with_no_members_noerror_empty():
    e = mkengine()
    put(e, "foo.com", None)
    put(e, "s.foo.com", {
        "type": "service",
        "service": {"srvce": "_x", "proto": "_tcp", "port": 1},
    })
    r = e.query("s.foo.com", "A")
    assert r["rcode"] == "NOERROR"
    assert r["answers"] == []


def test_miss_with_rd_and_recursion_hands_off():
    e = mkengine(recursion=True)
    put(e, "foo.com", None)
    r = e.query("x.dc2.foo.com", "A", rd=True)
    assert r["action"] == "recurse"
    r = e.query("x.dc2.foo.com", "A", rd=False)
    assert r["action"] == "respond"
    assert r["rcode"] == "REFUSED"


def test_doubled_suffix_refused():
    e = mkengine()
    put(e, "foo.com", None)
    assert e.query("x.foo.com.foo.com", "A")["rcode"] == "REFUSED"
    assert e.query("x.foo.com.coal.foo.com", "A")["rcode"] == "REFUSED"


def test_empty_srv_rest_refused():
    e = mkengine(dns_domain="")
    r = e.query("_http._tcp", "SRV")
    assert r["rcode"] == "REFUSED"


def test_shuffle_varies_order():
    e = mkengine()
    put(e, "foo.com", None)
    put(e, "s.foo.com", {
        "type": "service",
        "service": {"srvce": "_x", "proto": "_tcp", "port": 1},
    })
    for i in range(8):
        put(e, f"m{i}.s.foo.com",
            {"type": "rr_host", "rr_host": {"address": f"10.9.0.{i}"}})
    orders = set()
    for _ in range(20):
        r = e.query("s.foo.com", "A")
        orders.add(tuple(a["address"] for a in r["answers"]))
    assert len(orders) > 1  # Fisher-Yates shuffle is active


def test_reverse_map_updates_on_change():
    e = mkengine()
    put(e, "foo.com", None)
    put(e, "h.foo.com", {"type": "host",
                         "host": {"address": "10.5.0.1"}})
    assert e.query("1.0.5.10.in-addr.arpa", "PTR")["rcode"] == "NOERROR"
    put(e, "h.foo.com", {"type": "host",
                         "host": {"address": "10.5.0.2"}})
    assert e.query("1.0.5.10.in-addr.arpa", "PTR")["rcode"] == "REFUSED"
    assert e.query("2.0.5.10.in-addr.arpa", "PTR")["answers"][0][
        "target"] == "h.foo.com"
    e.remove("h.foo.com")
    assert e.query("2.0.5.10.in-addr.arpa", "PTR")["rcode"] == "REFUSED"


def test_domain_path_mapping():
    assert n.domain_to_path("foo.com") == "/com/foo"
    assert n.path_to_domain("/com/foo") == "foo.com"
    assert n.domain_to_path("web.bar.foo.com") == "/com/foo/bar/web"
    assert n.path_to_domain("/com/foo/bar/web") == "web.bar.foo.com"


def test_url_hostname():
    assert n.url_hostname("tcp://a@1.2.3.4:5432/db") == "1.2.3.4"
    assert n.url_hostname("ldaps://ufds.coal.joyent.us") == \
        "ufds.coal.joyent.us"
    assert n.url_hostname("http://[::1]:8080/x") == "::1"
    assert n.url_hostname("10.0.0.1:70") == "10.0.0.1"


def test_non_query_opcode_notimp():
    e = mkengine()
    put(e, "foo.com", None)
    put(e, "web.foo.com", HOST)
    import binder_amd
    n = binder_amd.require_native()
    wire = bytearray(n.encode_message(
        {"id": 5, "questions": [{"name": "web.foo.com", "type": "A"}]}))
    wire[2] |= 0x28  # opcode 5 (UPDATE)
    resp_wire, action = e.query_wire(bytes(wire), 512)
    resp = n.decode_message(resp_wire)
    assert resp["rcode"] == "NOTIMP"


def test_database_bad_primary_urls_empty_noerror():
    """Garbage/non-IPv4 database primaries: the reference would abort
    constructing ARecord(undefined-or-hostname) (deviations ledger #3);
    here they answer NOERROR with no records — never 0.0.0.0."""
    e = mkengine()
    put(e, "foo.com", None)
    cases = {
        "g1.foo.com": {"type": "database",
                       "database": {"primary": "not a url"}},
        "g2.foo.com": {"type": "database", "database": {}},
        "g3.foo.com": {"type": "database",
                       "database": {"primary": "tcp://u@[::1]:5432/d"}},
        "g4.foo.com": {"type": "database",
                       "database": {"primary": "tcp://u@pg.local:5432/d"}},
    }
    for name, rec in cases.items():
        put(e, name, rec)
        r = e.query(name, "A")
        assert r["rcode"] == "NOERROR", name
        assert r["answers"] == [], name


def test_host_non_ipv4_address_empty_noerror():
    e = mkengine()
    put(e, "foo.com", None)
    put(e, "v6.foo.com", {"type": "host", "host": {"address": "fd00::1"}})
    put(e, "junk.foo.com", {"type": "host", "host": {"address": "zzz"}})
    for name in ("v6.foo.com", "junk.foo.com"):
        r = e.query(name, "A")
        assert r["rcode"] == "NOERROR", name
        assert r["answers"] == [], name


def test_resolver_typed_record_not_served_as_host(eng):
    """The typed resolver-registry records ({"type":"resolver",...},
    recursion.hpp schema) are registry data, not DNS answers: an A
    query for one must get the unknown-type treatment (NOERROR, no
    answers — server.js:419-424), never the resolver's address."""
    put(eng, "r1.bar.foo.com", {
        "type": "resolver",
        "resolver": {"datacenter": "dc2", "address": "10.9.9.9"}})
    r = eng.query("r1.bar.foo.com", "A")
    assert r["rcode"] == "NOERROR"
    assert r["answers"] == []
