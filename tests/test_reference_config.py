"""Boot binderd with a VERBATIM reference-shaped config
(sapi_manifests/binder/template, Triton flavor) — every key the SAPI
config-agent renders must be accepted, including the full UFDS block
with its cache/retry/timeout knobs, and an unreachable ldaps UFDS must
degrade to best-effort (server keeps serving, misses refused)."""
import json

from binder_amd.harness import BinderProcess
from binder_amd.stubzk import StubZk


def test_triton_flavor_config_boots_and_serves(tmp_path):
    zk = StubZk().start()
    try:
        zk.mkdirp("/us/joyent/coal")
        zk.put("/us/joyent/coal/web", json.dumps(
            {"type": "host", "host": {"address": "10.99.0.5"}}).encode())

        cfg = {
            # rendered exactly like the reference template
            "dnsDomain": "coal.joyent.us",
            "datacenterName": "coal",
            "recursion": {
                "regionName": "us-west-x",
                "datacenterName": "coal",
                "dnsDomain": "coal.joyent.us",
                "ufds": {
                    "url": "ldaps://ufds.coal.joyent.us",
                    "bindDN": "cn=root",
                    "bindPassword": "secret",
                    "cache": {"size": 5000, "expiry": 60},
                    "maxConnections": 1,
                    "retry": {"initialDelay": 1000, "retries": 3},
                    "clientTimeout": 120000,
                    "connectTimeout": 3000,
                },
            },
            "instance_uuid": "7b8bb596-0000-0000-0000-000000000000",
            "server_uuid": "44454c4c-0000-0000-0000-000000000000",
            "service_name": "binder",
        }
        srv = BinderProcess(dns_domain="coal.joyent.us",
                            datacenter="coal", store="zk",
                            zk_host="127.0.0.1", zk_port=zk.port,
                            workdir=tmp_path, config=cfg,
                            log_path=str(tmp_path / "b.log"))
        srv.start()
        try:
            r = srv.wait_ready("web.coal.joyent.us")
            assert r.answers[0]["address"] == "10.99.0.5"
            # metrics carry the rendered static labels
            text = srv.metrics()
            assert 'datacenter="coal"' in text
            assert 'service="binder"' in text
            assert 'instance="7b8bb596' in text
            # recursion is best-effort: UFDS is unreachable (no ldaps
            # listener) => misses refuse rather than hang or crash
            r = srv.dig("missing.elsewhere.coal.joyent.us", rd=True,
                        timeout=4)
            assert r.status == "REFUSED"
            # still healthy afterwards
            assert srv.dig("web.coal.joyent.us").status == "NOERROR"
        finally:
            srv.stop()
    finally:
        zk.stop()


def test_registrar_template_layout_served(tmp_path):
    """The registrar manifest's registration shape (rr_host node whose
    'service' wrapper registers _dns._udp) resolves exactly as in the
    reference deployment (sapi_manifests/registrar/template)."""
    zk = StubZk().start()
    try:
        zk.mkdirp("/us/joyent/coal/binder")
        # registrar writes: service node at the domain...
        zk.set("/us/joyent/coal/binder", json.dumps({
            "type": "service",
            "service": {"srvce": "_dns", "proto": "_udp", "ttl": 60,
                        "port": 53},
            "ttl": 60,
        }).encode())
        # ...and an rr_host child per instance
        zk.put("/us/joyent/coal/binder/zone1", json.dumps({
            "type": "rr_host",
            "rr_host": {"address": "10.99.0.8"},
            "ttl": 30,
            "ports": [53],
        }).encode())
        srv = BinderProcess(dns_domain="coal.joyent.us",
                            datacenter="coal", store="zk",
                            zk_host="127.0.0.1", zk_port=zk.port,
                            workdir=tmp_path)
        srv.start()
        try:
            r = srv.wait_ready("_dns._udp.binder.coal.joyent.us",
                               qtype="SRV")
            assert r.answers[0]["port"] == 53
            assert r.answers[0]["ttl"] == 60
            assert r["additionals"][0]["address"] == "10.99.0.8"
            r = srv.dig("binder.coal.joyent.us")
            assert r.answers[0]["address"] == "10.99.0.8"
            r = srv.dig("8.0.99.10.in-addr.arpa", "PTR")
            assert r.answers[0]["target"] == \
                "zone1.binder.coal.joyent.us"
        finally:
            srv.stop()
    finally:
        zk.stop()
