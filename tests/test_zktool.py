"""zktool: native ZK CLI against the stub server (also exercises the
C++ client's CRUD paths, which the mirror never uses)."""
import subprocess

import pytest

from binder_amd import REPO_ROOT
from binder_amd.stubzk import StubZk

ZT = str(REPO_ROOT / "bin" / "zktool")


@pytest.fixture()
def zk():
    z = StubZk().start()
    yield z
    z.stop()


def run(zk, *args, rc=0):
    out = subprocess.run([ZT, "-s", f"127.0.0.1:{zk.port}", *args],
                         capture_output=True, text=True, timeout=20)
    assert out.returncode == rc, (args, out.stdout, out.stderr)
    return out.stdout


def test_crud(zk):
    assert run(zk, "create", "/a", '{"x":1}') == "/a\n"
    assert run(zk, "get", "/a") == '{"x":1}\n'
    run(zk, "set", "/a", '{"x":2}')
    assert run(zk, "get", "/a") == '{"x":2}\n'
    run(zk, "create", "/a/b")
    run(zk, "create", "/a/c")
    assert sorted(run(zk, "ls", "/a").split()) == ["b", "c"]
    assert "numChildren=2" in run(zk, "stat", "/a")
    run(zk, "rm", "/a/b")
    assert run(zk, "ls", "/a").split() == ["c"]
    run(zk, "rmr", "/a")
    run(zk, "get", "/a", rc=2)


def test_errors(zk):
    run(zk, "get", "/missing", rc=2)
    run(zk, "create", "/x")
    run(zk, "create", "/x", rc=2)        # exists
    run(zk, "rm", "/missing", rc=2)
    run(zk, "bogus-op", "/x", rc=1)


def test_deep_rmr(zk):
    zk.mkdirp("/d/e/f/g")
    zk.mkdirp("/d/e2")
    run(zk, "rmr", "/d")
    assert not zk.exists("/d")
