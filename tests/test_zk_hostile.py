"""Hostile/broken ZK server: binderd's ZK client must survive garbage,
truncated handshakes, and flapping connections — serving SERVFAIL
(mirror never ready) without crashing or spinning."""
import random
import socket
import struct
import threading
import time

import pytest

from binder_amd.harness import BinderProcess


class GarbageZk:
    """Accepts ZK connections and misbehaves per `mode`."""

    def __init__(self, mode):
        self.mode = mode
        self.sock = socket.socket()
        self.sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self.sock.bind(("127.0.0.1", 0))
        self.sock.listen(8)
        self.port = self.sock.getsockname()[1]
        self.sock.settimeout(0.2)
        self.accepted = 0
        self._running = True
        self._t = threading.Thread(target=self._loop, daemon=True)
        self._t.start()

    def _loop(self):
        rng = random.Random(1)
        while self._running:
            try:
                conn, _ = self.sock.accept()
            except socket.timeout:
                continue
            except OSError:
                return
            self.accepted += 1
            try:
                conn.settimeout(2)
                try:
                    conn.recv(4096)  # their handshake
                except socket.timeout:
                    pass
                if self.mode == "garbage":
                    # length-prefixed random junk
                    for _ in range(5):
                        junk = rng.randbytes(rng.randrange(1, 64))
                        conn.sendall(struct.pack(">i", len(junk)) + junk)
                elif self.mode == "huge":
                    conn.sendall(struct.pack(">i", 1 << 30))
                elif self.mode == "truncated":
                    conn.sendall(struct.pack(">i", 100) + b"\x00" * 10)
                elif self.mode == "slam":
                    pass  # close immediately
                time.sleep(0.1)
            except OSError:
                pass
            finally:
                conn.close()

    def stop(self):
        self._running = False
        self._t.join(timeout=5)
        self.sock.close()


@pytest.mark.parametrize("mode", ["garbage", "huge", "truncated",
                                  "slam"])
def test_binderd_survives_hostile_zk(tmp_path, mode):
    zk = GarbageZk(mode)
    srv = BinderProcess(store="zk", zk_host="127.0.0.1",
                        zk_port=zk.port, workdir=tmp_path,
                        log_level="warn")
    srv.start()
    try:
        end = time.time() + 5
        answered = 0
        while time.time() < end:
            try:
                r = srv.dig("web.foo.com", timeout=2)
            except OSError:
                continue  # loaded CI box: a dropped probe is fine
            assert r.status == "SERVFAIL"  # mirror never materialized
            answered += 1
            time.sleep(0.2)
        assert answered >= 3, "server stopped answering"
        assert srv.proc.poll() is None, "binderd died"
        assert zk.accepted >= 1
    finally:
        srv.stop()
        zk.stop()
