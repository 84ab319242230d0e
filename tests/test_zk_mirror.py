"""ZK client + mirror integration: binderd (zk store) against StubZk.

Covers the reference's watch-propagation behaviors (lib/zk.js) at the
wire level: initial tree materialization, data/children watch delivery,
node removal, reverse-map updates, reconnect-with-resume, and full
session-expiry rebuild — the hardest component per SURVEY.md §7.
"""
import json
import time

import pytest

from binder_amd.harness import BinderProcess
from binder_amd.stubzk import StubZk


def jput(zk, path, obj):
    zk.put(path, json.dumps(obj).encode())


@pytest.fixture()
def stack(tmp_path):
    zk = StubZk().start()
    zk.mkdirp("/com/foo")
    jput(zk, "/com/foo/bar", None)
    jput(zk, "/com/foo/bar/web",
         {"type": "host", "host": {"address": "192.168.0.1"}})
    srv = BinderProcess(store="zk", zk_host="127.0.0.1", zk_port=zk.port,
                        workdir=tmp_path, log_path=str(tmp_path / "b.log"))
    srv.start()
    yield zk, srv
    srv.stop()
    zk.stop()


def test_initial_tree_materializes(stack):
    zk, srv = stack
    r = srv.wait_ready("web.bar.foo.com")
    assert r.answers[0]["address"] == "192.168.0.1"


def test_data_change_propagates(stack):
    zk, srv = stack
    srv.wait_ready("web.bar.foo.com")
    jput(zk, "/com/foo/bar/web",
         {"type": "host", "host": {"address": "192.168.0.99"}})
    deadline = time.time() + 12
    while time.time() < deadline:
        r = srv.dig("web.bar.foo.com")
        if r.answers and r.answers[0]["address"] == "192.168.0.99":
            break
        time.sleep(0.05)
    else:
        pytest.fail("data change never propagated")
    # reverse map followed the address change
    assert srv.dig("99.0.168.192.in-addr.arpa", "PTR").status == "NOERROR"
    assert srv.dig("1.0.168.192.in-addr.arpa", "PTR").status == "REFUSED"


def test_new_and_removed_children_propagate(stack):
    zk, srv = stack
    srv.wait_ready("web.bar.foo.com")
    jput(zk, "/com/foo/bar/api",
         {"type": "host", "host": {"address": "192.168.0.2"}})
    srv.wait_ready("api.bar.foo.com")
    zk.rmr("/com/foo/bar/api")
    deadline = time.time() + 12
    while time.time() < deadline:
        if srv.dig("api.bar.foo.com").status == "REFUSED":
            break
        time.sleep(0.05)
    else:
        pytest.fail("removal never propagated")
    # sibling unaffected
    assert srv.dig("web.bar.foo.com").status == "NOERROR"


def test_service_tree_via_zk(stack):
    zk, srv = stack
    jput(zk, "/com/foo/svc", {
        "type": "service",
        "service": {"srvce": "_http", "proto": "_tcp", "port": 80,
                    "ttl": 60}})
    jput(zk, "/com/foo/svc/lb0",
         {"type": "load_balancer", "load_balancer": {"address": "10.0.1.0"}})
    jput(zk, "/com/foo/svc/lb1",
         {"type": "load_balancer", "load_balancer": {"address": "10.0.1.1"}})
    r = srv.wait_ready("_http._tcp.svc.foo.com", qtype="SRV")
    assert len(r.answers) == 2


def test_reconnect_resumes_serving(stack):
    zk, srv = stack
    srv.wait_ready("web.bar.foo.com")
    zk.drop_connections()
    # server keeps answering from the mirror during the outage
    assert srv.dig("web.bar.foo.com").status == "NOERROR"
    # after reconnect, watches work again
    deadline = time.time() + 10
    jput(zk, "/com/foo/bar/web2",
         {"type": "host", "host": {"address": "192.168.0.3"}})
    while time.time() < deadline:
        if srv.dig("web2.bar.foo.com").status == "NOERROR":
            break
        time.sleep(0.1)
    else:
        pytest.fail("watches dead after reconnect")


def test_session_expiry_rebuilds(stack):
    zk, srv = stack
    srv.wait_ready("web.bar.foo.com")
    zk.expire_sessions()
    assert srv.dig("web.bar.foo.com").status == "NOERROR"  # stale-serve
    jput(zk, "/com/foo/bar/web",
         {"type": "host", "host": {"address": "192.168.0.77"}})
    deadline = time.time() + 15
    while time.time() < deadline:
        r = srv.dig("web.bar.foo.com")
        if r.answers and r.answers[0]["address"] == "192.168.0.77":
            break
        time.sleep(0.1)
    else:
        pytest.fail("mirror never resynced after session expiry")


def test_unparseable_node_data_ignored(stack):
    zk, srv = stack
    srv.wait_ready("web.bar.foo.com")
    zk.put("/com/foo/bar/web", b"{definitely not json")
    time.sleep(0.3)
    r = srv.dig("web.bar.foo.com")
    assert r.answers[0]["address"] == "192.168.0.1"  # previous data kept


def test_larger_tree_count(stack):
    zk, srv = stack
    for i in range(200):
        jput(zk, f"/com/foo/h{i}",
             {"type": "host", "host": {"address": f"10.42.{i//250}.{i%250}"}})
    srv.wait_ready("h199.foo.com", timeout=15)
    assert srv.dig("h0.foo.com").status == "NOERROR"
    assert srv.dig("h123.foo.com").answers[0]["address"] == "10.42.0.123"


def test_ephemeral_nodes_vanish_on_session_expiry(stack):
    """Production registrars create EPHEMERAL znodes; when their session
    dies, ZK deletes the node and binder must stop serving it."""
    zk, srv = stack
    srv.wait_ready("web.bar.foo.com")

    # a "registrar" session holding an ephemeral registration
    from binder_amd.zkclient import ZkConn
    reg = ZkConn("127.0.0.1", zk.port)
    import json as _json
    reg.create("/com/foo/eph", _json.dumps(
        {"type": "host", "host": {"address": "10.66.0.1"}}).encode(),
        flags=1)
    srv.wait_ready("eph.foo.com")

    # registrar dies (session expired server-side)
    zk.expire_session(reg.session_id)
    deadline = time.time() + 10
    while time.time() < deadline:
        if srv.dig("eph.foo.com").status == "REFUSED":
            break
        time.sleep(0.1)
    else:
        pytest.fail("ephemeral registration never vanished")
    # PTR gone too (reverse-map cleanup)
    assert srv.dig("1.0.66.10.in-addr.arpa", "PTR").status == "REFUSED"


def test_rapid_delete_recreate_keeps_watches(stack):
    """delete+recreate faster than watch processing must not leave a
    retained node with a dead data watch (stale-forever hazard)."""
    zk, srv = stack
    srv.wait_ready("web.bar.foo.com")
    for cycle in range(5):
        # delete and recreate back-to-back (faster than the mirror's
        # round trips)
        zk.delete("/com/foo/bar/web")
        jput(zk, "/com/foo/bar/web",
             {"type": "host", "host": {"address": f"192.168.1.{cycle}"}})
        deadline = time.time() + 10
        while time.time() < deadline:
            r = srv.dig("web.bar.foo.com")
            if r.answers and r.answers[0]["address"] == \
                    f"192.168.1.{cycle}":
                break
            time.sleep(0.05)
        else:
            pytest.fail(f"cycle {cycle}: recreated node stale")
        # and a subsequent plain update must still propagate (the
        # watch must be alive)
        jput(zk, "/com/foo/bar/web",
             {"type": "host",
              "host": {"address": f"192.168.2.{cycle}"}})
        deadline = time.time() + 10
        while time.time() < deadline:
            r = srv.dig("web.bar.foo.com")
            if r.answers and r.answers[0]["address"] == \
                    f"192.168.2.{cycle}":
                break
            time.sleep(0.05)
        else:
            pytest.fail(f"cycle {cycle}: data watch died after "
                        "delete+recreate")
