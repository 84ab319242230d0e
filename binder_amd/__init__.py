"""binder-amd: native-first rebuild of TritonDataCenter/binder.

An authoritative DNS / service-discovery framework: a C++20 DNS server
(`binderd`) that mirrors a ZooKeeper registration tree into memory and
answers A/SRV/PTR queries from it, fronted by a native L4 balancer for
horizontal scale-out, with Prometheus metrics, structured JSON logging,
cross-DC recursion, and instance-convergence tooling.

Reference capability map: see SURVEY.md. The compute path in the
reference is a Node.js event loop (no GPU / tensor work exists there —
BASELINE.json "north_star"); the rebuild is native C++ end to end.
"""

__version__ = "0.2.0"

from pathlib import Path

REPO_ROOT = Path(__file__).resolve().parent.parent

try:  # built by `make` (see __graft_entry__.build)
    from . import _native  # noqa: F401
    HAVE_NATIVE = True
except ImportError:  # pragma: no cover - build() not yet run
    _native = None
    HAVE_NATIVE = False


def require_native():
    if not HAVE_NATIVE:
        raise RuntimeError(
            "binder_amd._native is not built; run `make` in the repo root "
            "(or __graft_entry__.build())")
    return _native
