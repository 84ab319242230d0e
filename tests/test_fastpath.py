"""Fast-path (prebuilt wire) parity: answers must be byte-equivalent in
meaning to the slow path, and caches must invalidate on updates.

The fast path engages at warn AND info level (it emits the per-query
info log from cached fragments); debug/trace still force the slow
path, so the twin-diff runs one server at warn (fast) and one at debug
(slow) and diffs responses. A separate test pins the fast path's
info-level log lines against the slow path's format.
"""
import json
import time

import pytest

from binder_amd.harness import BinderProcess
from binder_amd.stubzk import StubZk

TREE = {
    "foo.com": None,
    "web.foo.com": {"type": "host", "host": {"address": "1.2.3.4"}},
    "db.foo.com": {"type": "database",
                   "database": {"primary": "tcp://u@9.8.7.6:5432/x"},
                   "ttl": 20},
    "svc.foo.com": {"type": "service",
                    "service": {"srvce": "_x", "proto": "_tcp",
                                "port": 1, "ttl": 60}},
    "m0.svc.foo.com": {"type": "rr_host",
                       "rr_host": {"address": "10.0.0.9"}},
    # member-level TTL override exercises the min()/override chains
    "m1.svc.foo.com": {"type": "rr_host", "ttl": 7,
                       "rr_host": {"address": "10.0.0.10"}},
}


@pytest.fixture()
def twins(tmp_path):
    store = tmp_path / "tree.json"
    store.write_text(json.dumps(TREE))
    fast = BinderProcess(store=f"file:{store}", workdir=tmp_path,
                         log_level="warn")
    slow = BinderProcess(store=f"file:{store}", workdir=tmp_path,
                         log_level="debug")
    fast.start()
    slow.start()
    yield fast, slow
    fast.stop()
    slow.stop()


def normalize(r):
    return (r.status, r["aa"], r["ra"], r["tc"],
            sorted((a["type"], a.get("address"), a["ttl"], a["name"])
                   for a in r.answers))


@pytest.mark.parametrize("name,qtype,rd", [
    ("web.foo.com", "A", False),
    ("web.foo.com", "A", True),
    ("WEB.foo.com", "A", False),       # uppercase: both slow
    ("db.foo.com", "A", False),
    ("svc.foo.com", "A", False),       # service: fast path bails
    ("missing.foo.com", "A", False),
    ("web.foo.com", "SRV", False),
    ("foo.com", "A", False),           # apex refused
    ("web.foo.com.foo.com", "A", False),
])
def test_fast_equals_slow(twins, name, qtype, rd):
    fast, slow = twins
    rf = fast.dig(name, qtype, rd=rd, qid=777)
    rs = slow.dig(name, qtype, rd=rd, qid=777)
    assert normalize(rf) == normalize(rs), (name, qtype, rd)
    assert rf["id"] == 777
    assert rf["rd"] == rd


def test_edns_still_served_with_fastpath_on(twins):
    fast, _ = twins
    r = fast.dig("web.foo.com", edns=4096)
    assert r.status == "NOERROR"
    assert any(x["type"] == "OPT" for x in r["additionals"])


def test_cache_invalidation_on_update(tmp_path):
    zk = StubZk().start()
    try:
        zk.mkdirp("/com/foo")
        zk.put("/com/foo/web", json.dumps(
            {"type": "host", "host": {"address": "1.1.1.1"}}).encode())
        srv = BinderProcess(store="zk", zk_host="127.0.0.1",
                            zk_port=zk.port, workdir=tmp_path,
                            log_level="warn")
        srv.start()
        try:
            srv.wait_ready("web.foo.com")
            # prime the fast-path cache
            for _ in range(5):
                r = srv.dig("web.foo.com")
                assert r.answers[0]["address"] == "1.1.1.1"
            zk.put("/com/foo/web", json.dumps(
                {"type": "host",
                 "host": {"address": "2.2.2.2", "ttl": 77}}).encode())
            deadline = time.time() + 5
            while time.time() < deadline:
                r = srv.dig("web.foo.com")
                if r.answers and r.answers[0]["address"] == "2.2.2.2":
                    break
                time.sleep(0.05)
            else:
                pytest.fail("cached answer never invalidated")
            assert r.answers[0]["ttl"] == 77
        finally:
            srv.stop()
    finally:
        zk.stop()


def test_service_fastpath_parity(twins):
    fast, slow = twins
    for qtype, name in [("A", "svc.foo.com"),
                        ("SRV", "_x._tcp.svc.foo.com")]:
        rf = fast.dig(name, qtype, qid=555)
        rs = slow.dig(name, qtype, qid=555)
        assert rf.status == rs.status == "NOERROR"
        def norm(r):
            return sorted(
                (a["type"], a.get("address"), a.get("target"),
                 a.get("port"), a.get("priority"), a.get("weight"),
                 a["ttl"], a["name"]) for a in r.answers)
        assert norm(rf) == norm(rs), (qtype, rf, rs)
        addf = sorted((x["name"], x.get("address"), x["ttl"])
                      for x in rf["additionals"])
        adds = sorted((x["name"], x.get("address"), x["ttl"])
                      for x in rs["additionals"])
        assert addf == adds


def test_service_fastpath_shuffles(tmp_path):
    import json as _json
    tree = {"foo.com": None,
            "s.foo.com": {"type": "service",
                          "service": {"srvce": "_x", "proto": "_tcp",
                                      "port": 1}}}
    for i in range(6):
        tree[f"m{i}.s.foo.com"] = {
            "type": "rr_host", "rr_host": {"address": f"10.4.0.{i}"}}
    store = tmp_path / "t.json"
    store.write_text(_json.dumps(tree))
    srv = BinderProcess(store=f"file:{store}", workdir=tmp_path,
                        log_level="warn")
    srv.start()
    try:
        orders = set()
        for _ in range(25):
            r = srv.dig("s.foo.com")
            orders.add(tuple(a["address"] for a in r.answers))
        assert len(orders) > 1, "fast path must keep shuffling"
    finally:
        srv.stop()


def test_service_fastpath_invalidation_on_member_change(tmp_path):
    zk = StubZk().start()
    try:
        zk.mkdirp("/com/foo")
        zk.put("/com/foo/s", json.dumps(
            {"type": "service",
             "service": {"srvce": "_x", "proto": "_tcp",
                         "port": 9}}).encode())
        zk.put("/com/foo/s/m0", json.dumps(
            {"type": "rr_host",
             "rr_host": {"address": "10.5.0.1"}}).encode())
        srv = BinderProcess(store="zk", zk_host="127.0.0.1",
                            zk_port=zk.port, workdir=tmp_path,
                            log_level="warn")
        srv.start()
        try:
            srv.wait_ready("s.foo.com")
            # wait for the member itself to be mirrored (a service
            # answers NOERROR even with zero members), then prime
            deadline = time.time() + 10
            while time.time() < deadline:
                r = srv.dig("_x._tcp.s.foo.com", "SRV")
                if r["additionals"]:
                    break
                time.sleep(0.05)
            for _ in range(5):  # prime cache
                r = srv.dig("_x._tcp.s.foo.com", "SRV")
            assert r["additionals"][0]["address"] == "10.5.0.1"
            # member address change must invalidate the parent cache
            zk.put("/com/foo/s/m0", json.dumps(
                {"type": "rr_host",
                 "rr_host": {"address": "10.5.0.2"}}).encode())
            deadline = time.time() + 10
            while time.time() < deadline:
                r = srv.dig("_x._tcp.s.foo.com", "SRV")
                if r["additionals"] and \
                        r["additionals"][0]["address"] == "10.5.0.2":
                    break
                time.sleep(0.05)
            else:
                pytest.fail("stale cached service answer")
            # membership change too
            zk.put("/com/foo/s/m1", json.dumps(
                {"type": "rr_host",
                 "rr_host": {"address": "10.5.0.3"}}).encode())
            deadline = time.time() + 10
            while time.time() < deadline:
                r = srv.dig("_x._tcp.s.foo.com", "SRV")
                if len(r.answers) == 2:
                    break
                time.sleep(0.05)
            else:
                pytest.fail("new member never appeared in cached answer")
        finally:
            srv.stop()
    finally:
        zk.stop()


def test_fastpath_info_log_lines_match_slow_format(tmp_path):
    """At info level the fast path emits the per-query bunyan line
    itself (from cached fragments). Its fields must match the slow
    path's line for the same query (modulo latency/timers values and
    member shuffle order)."""
    store = tmp_path / "tree.json"
    store.write_text(json.dumps(TREE))
    fast_log = tmp_path / "fast.log"
    slow_log = tmp_path / "slow.log"
    fast = BinderProcess(store=f"file:{store}", workdir=tmp_path,
                         log_level="info", log_path=str(fast_log))
    slow = BinderProcess(store=f"file:{store}", workdir=tmp_path,
                         log_level="debug", log_path=str(slow_log))
    fast.start()
    slow.start()
    try:
        for name, qtype in [("web.foo.com", "A"),
                            ("svc.foo.com", "A"),
                            ("_x._tcp.svc.foo.com", "SRV")]:
            for _ in range(2):  # second hit is served from the cache
                fast.dig(name, qtype)
            slow.dig(name, qtype)
        time.sleep(0.3)

        def lines(path, name):
            out = []
            for ln in path.read_text().splitlines():
                try:
                    d = json.loads(ln)
                except ValueError:
                    continue
                if d.get("msg") == "DNS query" and \
                        d.get("query", {}).get("name") == name:
                    out.append(d)
            return out

        for name, qtype in [("web.foo.com", "A"),
                            ("svc.foo.com", "A"),
                            ("_x._tcp.svc.foo.com", "SRV")]:
            fl = lines(fast_log, name)
            sl = lines(slow_log, name)
            assert fl and sl, name
            f, s = fl[-1], sl[-1]
            assert f["rcode"] == s["rcode"] == "NOERROR"
            assert f["query"] == s["query"]
            assert sorted(f["answers"]) == sorted(s["answers"]), name
            assert sorted(f["additional"]) == sorted(s["additional"])
            assert f["edns"] is False
            for k in ("req_id", "client", "port", "latency", "timers"):
                assert k in f, k
            assert set(f["timers"]) == {"parse_us", "resolve_us",
                                        "encode_us"}
    finally:
        fast.stop()
        slow.stop()
