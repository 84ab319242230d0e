"""Mirror correctness under churn (BASELINE config 5 shape, CI scale).

Queries keep succeeding with fresh data while the registration tree is
mutated continuously — the watch pipeline must not lose updates or
leave stale reverse entries under load.
"""
import json
import random
import threading
import time

import pytest

from binder_amd.harness import BinderProcess
from binder_amd.stubzk import StubZk


@pytest.mark.timeout(120)
def test_queries_track_churn(tmp_path):
    zk = StubZk().start()
    try:
        zk.mkdirp("/com/foo")
        n_hosts = 400
        for i in range(n_hosts):
            zk.put(f"/com/foo/h{i}", json.dumps(
                {"type": "host",
                 "host": {"address": f"10.1.{i // 250}.{i % 250}"}}
            ).encode())
        srv = BinderProcess(store="zk", zk_host="127.0.0.1",
                            zk_port=zk.port, workdir=tmp_path,
                            log_level="warn",
                            log_path=str(tmp_path / "b.log"))
        srv.start()
        try:
            srv.wait_ready(f"h{n_hosts - 1}.foo.com", timeout=30)

            stop = threading.Event()
            mutations = {}

            def churner():
                rng = random.Random(42)
                gen = 0
                while not stop.is_set():
                    h = rng.randrange(n_hosts)
                    gen += 1
                    addr = f"10.2.{gen % 200}.{h % 250}"
                    mutations[h] = addr
                    zk.put(f"/com/foo/h{h}", json.dumps(
                        {"type": "host", "host": {"address": addr}}
                    ).encode())
                    time.sleep(0.002)  # ~500 mutations/s

            t = threading.Thread(target=churner, daemon=True)
            t.start()

            # query continuously during churn; all answers must be
            # NOERROR with plausible addresses
            end = time.time() + 4
            count = 0
            while time.time() < end:
                h = random.randrange(n_hosts)
                r = srv.dig(f"h{h}.foo.com", timeout=2)
                assert r.status == "NOERROR", (h, r)
                assert r.answers, (h, r)
                count += 1
            stop.set()
            t.join()
            assert count > 200

            # after churn settles, every mutated node must converge to
            # its final address
            deadline = time.time() + 30
            pending = dict(mutations)
            while pending and time.time() < deadline:
                for h, addr in list(pending.items()):
                    r = srv.dig(f"h{h}.foo.com", timeout=2)
                    if r.answers and r.answers[0]["address"] == addr:
                        del pending[h]
                if pending:
                    time.sleep(0.2)
            assert not pending, f"{len(pending)} nodes never converged"
        finally:
            srv.stop()
    finally:
        zk.stop()


@pytest.mark.timeout(120)
def test_node_add_remove_churn(tmp_path):
    """Add/remove cycles: no stale names survive, new names appear."""
    zk = StubZk().start()
    try:
        zk.mkdirp("/com/foo")
        srv = BinderProcess(store="zk", zk_host="127.0.0.1",
                            zk_port=zk.port, workdir=tmp_path,
                            log_level="warn")
        srv.start()
        try:
            for cycle in range(5):
                for i in range(30):
                    zk.put(f"/com/foo/c{cycle}x{i}", json.dumps(
                        {"type": "host",
                         "host": {"address": f"10.3.{cycle}.{i}"}}
                    ).encode())
                srv.wait_ready(f"c{cycle}x29.foo.com", timeout=15)
                if cycle > 0:
                    # previous cycle's nodes were removed
                    deadline = time.time() + 10
                    while time.time() < deadline:
                        r = srv.dig(f"c{cycle - 1}x0.foo.com")
                        if r.status == "REFUSED":
                            break
                        time.sleep(0.1)
                    assert r.status == "REFUSED"
                for i in range(30):
                    zk.rmr(f"/com/foo/c{cycle}x{i}")
        finally:
            srv.stop()
    finally:
        zk.stop()
