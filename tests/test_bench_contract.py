"""Driver-contract tests for bench.py.

The driver runs `python bench.py --gpus N ...` under
torch.distributed.run for N>1; cover the rank/barrier path here with
the gloo backend on CPU (world_size 2), as well as the single-process
default and the JSON output schema.
"""
import json
import os
import subprocess
import sys

import pytest

from binder_amd import REPO_ROOT

REQUIRED_KEYS = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup",
    "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
    "dtype", "data", "config",
}


def last_json_line(text):
    for line in reversed(text.strip().splitlines()):
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output: {text[-500:]}")


@pytest.mark.timeout(300)
def test_bench_single_process_schema():
    out = subprocess.run(
        [sys.executable, str(REPO_ROOT / "bench.py"), "--steps", "1",
         "--warmup", "0", "--queries-per-proc", "20000"],
        capture_output=True, text=True, timeout=280, cwd=REPO_ROOT)
    assert out.returncode == 0, out.stderr[-2000:]
    d = last_json_line(out.stdout)
    assert REQUIRED_KEYS.issubset(d.keys())
    assert d["metric"] == "dns_queries_per_sec"
    assert d["n_gpus"] == 1
    assert d["value"] > 0
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert d["config"]["tree_records"] == 10000


@pytest.mark.timeout(600)
def test_bench_under_torchrun_gloo_world2():
    env = dict(os.environ)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29533", str(REPO_ROOT / "bench.py"),
         "--gpus", "2", "--steps", "1", "--warmup", "0",
         "--queries-per-proc", "15000"],
        capture_output=True, text=True, timeout=580, cwd=REPO_ROOT,
        env=env)
    assert out.returncode == 0, (out.stdout[-1000:], out.stderr[-2000:])
    d = last_json_line(out.stdout)
    assert d["n_gpus"] == 2
    assert d["config"]["queries_per_step"] == 30000
    assert d["value"] > 0


def test_calibration_converges_and_tolerates_flakes():
    """calibrate_rate must converge near the true SLO threshold and a
    single flaky probe per level must not collapse the search (the
    2-of-3 tiebreak)."""
    import random

    import bench

    true_limit = 3_000_000
    rng = random.Random(42)

    class FakeBlast:
        def __init__(self, flake_prob=0.0):
            self.flake_prob = flake_prob

        def step(self, queries, rate=0):
            ok = rate <= true_limit
            if ok and rng.random() < self.flake_prob:
                ok = False  # transient stall lands in p99
            return {"qps": rate if ok else rate * 0.9,
                    "p99_us": 500 if ok else 9000,
                    "timeouts": 0, "received": queries,
                    "noerror": queries}

    # clean probes: land within [85%, 100%] of the true limit
    r = bench.calibrate_rate(FakeBlast(), capacity=4_000_000,
                             slo_us=2000)
    assert 0.85 * true_limit <= r <= true_limit, r

    # 20% flake probability: still within [70%, 100%]
    r = bench.calibrate_rate(FakeBlast(flake_prob=0.2),
                             capacity=4_000_000, slo_us=2000)
    assert 0.70 * true_limit <= r <= true_limit, r
