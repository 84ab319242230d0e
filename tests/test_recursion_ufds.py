"""UFDS/LDAP resolver discovery: the reference's full recursion
bootstrap path (recursion.js:104-249) against stub ZK + stub LDAP.

Flow under test: binderd mirrors the tree (which contains the UFDS
service registration) -> recursion resolves the UFDS address from its
own mirror -> LDAP bind + search region resolvers -> misses forward to
the discovered DC resolver.
"""
import json
import time

import pytest

from binder_amd.harness import BinderProcess
from binder_amd.stubldap import StubLdap
from binder_amd.stubzk import StubZk


@pytest.fixture()
def stack(tmp_path):
    ldap = StubLdap().start()

    # upstream binder for "dc2" on a second loopback address
    up_tree = tmp_path / "up.json"
    up_tree.write_text(json.dumps({
        "dc2.foo.com": None,
        "api.dc2.foo.com": {"type": "host",
                            "host": {"address": "10.22.5.5"}},
    }))
    upstream = BinderProcess(dns_domain="dc2.foo.com", datacenter="dc2",
                             host="127.0.0.2", store=f"file:{up_tree}",
                             workdir=tmp_path,
                             log_path=str(tmp_path / "up.log"))
    upstream.start()

    ldap.resolvers = [{"datacenter": "dc2", "ip": "127.0.0.2"}]

    zk = StubZk().start()
    zk.mkdirp("/com/foo")
    # UFDS registered as a service in ZK (the resolveUfds path)
    zk.put("/com/foo/ufds", json.dumps({
        "type": "service",
        "service": {"srvce": "_ldap", "proto": "_tcp", "port": ldap.port},
    }).encode())
    zk.put("/com/foo/ufds/ufds0", json.dumps({
        "type": "host", "host": {"address": "127.0.0.1"}}).encode())

    local = BinderProcess(
        dns_domain="foo.com", datacenter="dc1", store="zk",
        zk_host="127.0.0.1", zk_port=zk.port, workdir=tmp_path,
        log_path=str(tmp_path / "local.log"),
        config={"recursion": {
            "source": "ufds",
            "regionName": "region-1",
            "dnsDomain": "foo.com",
            "upstreamPort": upstream.port,
            "ufds": {
                "url": f"ldap://ufds.foo.com:{ldap.port}",
                "bindDN": "cn=root",
                "bindPassword": "secret",
            },
        }})
    local.start()
    yield {"local": local, "upstream": upstream, "zk": zk, "ldap": ldap}
    local.stop()
    upstream.stop()
    zk.stop()
    ldap.stop()


def test_ufds_discovery_and_forwarding(stack):
    local, ldap = stack["local"], stack["ldap"]
    # wait for mirror + first LDAP refresh (retries every 15s on
    # failure; the first attempt may race mirror readiness)
    deadline = time.time() + 40
    while time.time() < deadline:
        try:
            r = local.dig("api.dc2.foo.com", rd=True, timeout=4)
            if r.status == "NOERROR":
                break
        except OSError:
            pass
        time.sleep(0.5)
    else:
        log = open(local.log_path).read()[-3000:]
        pytest.fail(f"never forwarded via ufds resolvers; log: {log}")
    assert r.answers[0]["address"] == "10.22.5.5"
    # the LDAP server saw a bind and the region search
    assert any("region=region-1" in s for s in ldap.searches)
    assert "cn=root" in stack["ldap"].binds


def test_ufds_bad_credentials_best_effort(tmp_path):
    """Bind failure leaves recursion empty (best effort) but the server
    still serves and refuses misses."""
    ldap = StubLdap()
    ldap.require_password = "correct"
    ldap.start()
    zk = StubZk().start()
    zk.mkdirp("/com/foo")
    try:
        srv = BinderProcess(
            dns_domain="foo.com", datacenter="dc1", store="zk",
            zk_host="127.0.0.1", zk_port=zk.port, workdir=tmp_path,
            config={"recursion": {
                "source": "ufds",
                "regionName": "r1",
                "dnsDomain": "foo.com",
                "ufds": {"url": f"ldap://127.0.0.1:{ldap.port}",
                         "bindDN": "cn=root",
                         "bindPassword": "wrong"},
            }})
        srv.start()
        try:
            time.sleep(1.0)
            r = srv.dig("nope.dc9.foo.com", rd=True, timeout=4)
            assert r.status == "REFUSED"
        finally:
            srv.stop()
    finally:
        zk.stop()
        ldap.stop()


def test_ufds_over_ldaps(tmp_path):
    """ldaps:// UFDS with a self-signed cert and the explicit
    tlsVerify=false opt-out (verification is ON by default now): the
    native TLS client must bind+search successfully."""
    import subprocess
    cert = tmp_path / "cert.pem"
    key = tmp_path / "key.pem"
    subprocess.run(
        ["openssl", "req", "-x509", "-newkey", "rsa:2048",
         "-keyout", str(key), "-out", str(cert), "-days", "2",
         "-nodes", "-subj", "/CN=ufds.test"],
        check=True, capture_output=True)
    ldap = StubLdap(tls_cert=str(cert), tls_key=str(key)).start()
    ldap.resolvers = [{"datacenter": "dc2", "ip": "127.0.0.2"}]

    up_tree = tmp_path / "up.json"
    up_tree.write_text(json.dumps({
        "dc2.foo.com": None,
        "tls.dc2.foo.com": {"type": "host",
                            "host": {"address": "10.31.0.1"}}}))
    upstream = BinderProcess(dns_domain="dc2.foo.com", datacenter="dc2",
                             host="127.0.0.2", store=f"file:{up_tree}",
                             workdir=tmp_path)
    upstream.start()

    local_tree = tmp_path / "local.json"
    local_tree.write_text('{"foo.com": null}')
    local = BinderProcess(
        dns_domain="foo.com", datacenter="dc1",
        store=f"file:{local_tree}", workdir=tmp_path,
        log_path=str(tmp_path / "l.log"),
        config={"recursion": {
            "source": "ufds", "regionName": "r1",
            "dnsDomain": "foo.com", "upstreamPort": upstream.port,
            "ufds": {"url": f"ldaps://127.0.0.1:{ldap.port}",
                     "bindDN": "cn=root", "bindPassword": "pw",
                     "tlsVerify": False},
        }})
    local.start()
    try:
        deadline = time.time() + 40
        while time.time() < deadline:
            try:
                r = local.dig("tls.dc2.foo.com", rd=True, timeout=4)
                if r.status == "NOERROR":
                    break
            except OSError:
                pass
            time.sleep(0.5)
        else:
            log = open(str(tmp_path / "l.log")).read()[-2000:]
            pytest.fail(f"ldaps discovery never worked; {log}")
        assert r.answers[0]["address"] == "10.31.0.1"
        assert any("region=r1" in s for s in ldap.searches)
    finally:
        local.stop()
        upstream.stop()
        ldap.stop()


def test_ufds_ldaps_tlsverify_rejects_selfsigned(tmp_path):
    """Certificate verification is the DEFAULT (no tlsVerify key in the
    config): a self-signed UFDS cert must be REJECTED, and binder must
    keep serving with recursion degraded to best-effort (misses
    refused) rather than hang or crash."""
    import subprocess
    cert = tmp_path / "cert.pem"
    key = tmp_path / "key.pem"
    subprocess.run(
        ["openssl", "req", "-x509", "-newkey", "rsa:2048",
         "-keyout", str(key), "-out", str(cert), "-days", "2",
         "-nodes", "-subj", "/CN=ufds.test"],
        check=True, capture_output=True)
    ldap = StubLdap(tls_cert=str(cert), tls_key=str(key)).start()
    ldap.resolvers = [{"datacenter": "dc2", "ip": "127.0.0.2"}]

    local_tree = tmp_path / "local.json"
    local_tree.write_text(
        '{"foo.com": null, "web.foo.com": '
        '{"type": "host", "host": {"address": "10.0.0.1"}}}')
    local = BinderProcess(
        dns_domain="foo.com", datacenter="dc1",
        store=f"file:{local_tree}", workdir=tmp_path,
        log_path=str(tmp_path / "lv.log"),
        config={"recursion": {
            "source": "ufds", "regionName": "r1",
            "dnsDomain": "foo.com",
            "ufds": {"url": f"ldaps://127.0.0.1:{ldap.port}",
                     "bindDN": "cn=root", "bindPassword": "pw"},
        }})
    local.start()
    try:
        # give init + at least one retry a chance to run
        time.sleep(2.0)
        # no resolver list can have been fetched over the rejected TLS
        # channel => cross-DC misses refuse
        r = local.dig("nope.dc2.foo.com", rd=True, timeout=4)
        assert r.status == "REFUSED"
        # local answers unaffected, process healthy
        assert local.dig("web.foo.com").status == "NOERROR"
        assert local.proc.poll() is None
        assert not any("region=r1" in s for s in ldap.searches), \
            "search succeeded despite tlsVerify against self-signed cert"
    finally:
        local.stop()
        ldap.stop()


def test_ufds_garbage_ldap_server(tmp_path):
    """A broken LDAP peer (random bytes instead of BER) must leave
    binder serving with recursion degraded — no crash, no hang."""
    import socket
    import threading

    srv = socket.socket()
    srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    srv.bind(("127.0.0.1", 0))
    srv.listen(4)
    port = srv.getsockname()[1]
    stop = threading.Event()

    def loop():
        srv.settimeout(0.3)
        import os as _os
        while not stop.is_set():
            try:
                conn, _ = srv.accept()
            except socket.timeout:
                continue
            except OSError:
                return
            try:
                conn.settimeout(1)
                try:
                    conn.recv(4096)
                except socket.timeout:
                    pass
                conn.sendall(_os.urandom(64))
            except OSError:
                pass
            finally:
                conn.close()

    t = threading.Thread(target=loop, daemon=True)
    t.start()

    local_tree = tmp_path / "t.json"
    local_tree.write_text(
        '{"foo.com": null, "web.foo.com": '
        '{"type": "host", "host": {"address": "10.0.0.1"}}}')
    local = BinderProcess(
        dns_domain="foo.com", datacenter="dc1",
        store=f"file:{local_tree}", workdir=tmp_path,
        config={"recursion": {
            "source": "ufds", "regionName": "r1",
            "dnsDomain": "foo.com",
            "ufds": {"url": f"ldap://127.0.0.1:{port}",
                     "bindDN": "cn=root", "bindPassword": "pw"},
        }})
    local.start()
    try:
        time.sleep(1.5)
        r = local.dig("nope.dc2.foo.com", rd=True, timeout=4)
        assert r.status == "REFUSED"
        assert local.dig("web.foo.com").status == "NOERROR"
        assert local.proc.poll() is None
    finally:
        local.stop()
        stop.set()
        t.join(timeout=3)
        srv.close()


def test_ufds_ldaps_with_cafile(tmp_path):
    """A private-CA deployment pins trust with recursion.ufds.caFile:
    verification stays on, the presented cert chains to the configured
    CA, and the SAN must cover the dialed address (IP SAN here)."""
    import subprocess
    cert = tmp_path / "ca-cert.pem"
    key = tmp_path / "ca-key.pem"
    subprocess.run(
        ["openssl", "req", "-x509", "-newkey", "rsa:2048",
         "-keyout", str(key), "-out", str(cert), "-days", "2",
         "-nodes", "-subj", "/CN=ufds.test",
         "-addext", "subjectAltName=IP:127.0.0.1"],
        check=True, capture_output=True)
    ldap = StubLdap(tls_cert=str(cert), tls_key=str(key)).start()
    ldap.resolvers = [{"datacenter": "dc2", "ip": "127.0.0.2"}]

    local_tree = tmp_path / "local.json"
    local_tree.write_text('{"foo.com": null}')
    local = BinderProcess(
        dns_domain="foo.com", datacenter="dc1",
        store=f"file:{local_tree}", workdir=tmp_path,
        log_path=str(tmp_path / "lc.log"),
        config={"recursion": {
            "source": "ufds", "regionName": "r1",
            "dnsDomain": "foo.com",
            "ufds": {"url": f"ldaps://127.0.0.1:{ldap.port}",
                     "bindDN": "cn=root", "bindPassword": "pw",
                     "caFile": str(cert)},
        }})
    local.start()
    try:
        deadline = time.time() + 20
        while time.time() < deadline:
            if any("region=r1" in s for s in ldap.searches):
                break
            time.sleep(0.25)
        assert any("region=r1" in s for s in ldap.searches), \
            "verified LDAPS search never happened; " + \
            open(str(tmp_path / "lc.log")).read()[-1500:]
    finally:
        local.stop()
        ldap.stop()


def test_ufds_ldaps_cafile_rejects_wrong_host(tmp_path):
    """Even with the right CA, a cert whose SAN does not cover the
    dialed address must be rejected (hostname/IP binding)."""
    import subprocess
    cert = tmp_path / "other-cert.pem"
    key = tmp_path / "other-key.pem"
    subprocess.run(
        ["openssl", "req", "-x509", "-newkey", "rsa:2048",
         "-keyout", str(key), "-out", str(cert), "-days", "2",
         "-nodes", "-subj", "/CN=ufds.test",
         "-addext", "subjectAltName=IP:10.99.99.99"],
        check=True, capture_output=True)
    ldap = StubLdap(tls_cert=str(cert), tls_key=str(key)).start()
    ldap.resolvers = [{"datacenter": "dc2", "ip": "127.0.0.2"}]

    local_tree = tmp_path / "local.json"
    local_tree.write_text('{"foo.com": null}')
    local = BinderProcess(
        dns_domain="foo.com", datacenter="dc1",
        store=f"file:{local_tree}", workdir=tmp_path,
        config={"recursion": {
            "source": "ufds", "regionName": "r1",
            "dnsDomain": "foo.com",
            "ufds": {"url": f"ldaps://127.0.0.1:{ldap.port}",
                     "bindDN": "cn=root", "bindPassword": "pw",
                     "caFile": str(cert)},
        }})
    local.start()
    try:
        time.sleep(2.0)
        assert not any("region=r1" in s for s in ldap.searches), \
            "search succeeded despite SAN/host mismatch"
        assert local.proc.poll() is None
    finally:
        local.stop()
        ldap.stop()
