"""The lint gate (`make check` -> tools/lint.py) must pass on the repo
and fail on seeded violations — parity with the reference's
eslint/jsstyle/cstyle CI gates (/root/reference/Makefile:17-20)."""
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
LINT = REPO / "tools" / "lint.py"


def run_lint(*args):
    return subprocess.run([sys.executable, str(LINT), *args],
                          capture_output=True, text=True)


def test_repo_is_lint_clean():
    r = run_lint()
    assert r.returncode == 0, r.stdout + r.stderr


def test_seeded_cxx_violation_fails(tmp_path):
    bad = tmp_path / "bad.cpp"
    bad.write_text("int main() {\n\tint x = 1;   \n"
                   "    " + "y" * 90 + ";\n}\n")
    r = run_lint(str(bad))
    assert r.returncode == 1
    assert "tab:" in r.stdout
    assert "trailing-ws:" in r.stdout
    assert "line-length:" in r.stdout


def test_seeded_using_namespace_std_fails(tmp_path):
    bad = tmp_path / "bad2.cpp"
    bad.write_text("using namespace std;\nint main() {}\n")
    r = run_lint(str(bad))
    assert r.returncode == 1
    assert "using-namespace-std" in r.stdout


def test_seeded_python_violations_fail(tmp_path):
    bad = tmp_path / "bad.py"
    bad.write_text(
        "import os\n"
        "def f(x=[]):\n"
        "    try:\n"
        "        pass\n"
        "    except:\n"
        "        pass\n"
        "    return x == None\n")
    r = run_lint(str(bad))
    assert r.returncode == 1
    for rule in ("unused-import", "mutable-default", "bare-except",
                 "eq-none"):
        assert rule in r.stdout, f"missing {rule} in:\n{r.stdout}"


def test_make_check_runs_lint():
    text = (REPO / "Makefile").read_text()
    assert "tools/lint.py" in text.split("check:")[1].split("\n\n")[0]


def test_binderd_has_stapsdt_probes():
    """op-req-start/op-req-done USDT probes must be compiled into the
    binary as .note.stapsdt ELF notes (reference: dtrace-provider
    probes, lib/server.js:25-29)."""
    r = subprocess.run(["readelf", "-n", str(REPO / "bin" / "binderd")],
                       capture_output=True, text=True)
    assert r.returncode == 0
    assert "stapsdt" in r.stdout
    assert "Provider: binder" in r.stdout
    assert "op-req-start" in r.stdout
    assert "op-req-done" in r.stdout
