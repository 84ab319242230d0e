#!/usr/bin/env python3
"""Style/lint gate for binder-amd (the reference gates merges on
eslint/jsstyle/cstyle — /root/reference/Makefile:17-20; this image has
no clang-tidy/ruff, so the gate is this self-contained checker, in the
same spirit as the reference's in-tree tools/cstyle.pl).

C++ rules (native/):
  - no tabs, no trailing whitespace, no CRLF
  - lines <= 80 columns
  - file ends with exactly one newline
  - no `using namespace std;`
  - no C-style malloc/free in C++ sources (new code uses RAII)
Python rules (binder_amd/, tests/, scripts/, repo-root *.py):
  - compiles (ast.parse)
  - no tabs in indentation, no trailing whitespace
  - lines <= 100 columns
  - no bare `except:`
  - no mutable default arguments (list/dict/set literals)
  - no `== None` / `!= None`
  - unused `import X` / `from X import Y` at module scope

Exit 0 = clean, 1 = violations (printed one per line as
path:line: rule: detail).
"""
from __future__ import annotations

import ast
import re
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent

CXX_LINE_MAX = 80
PY_LINE_MAX = 100


def check_common(path: Path, text: str, line_max: int, out: list):
    if "\r" in text:
        out.append(f"{path}:1: crlf: file contains carriage returns")
    if text and not text.endswith("\n"):
        out.append(f"{path}:{text.count(chr(10)) + 1}: eof: "
                   "missing newline at end of file")
    if text.endswith("\n\n\n"):
        out.append(f"{path}:{text.count(chr(10))}: eof: "
                   "multiple blank lines at end of file")
    for i, line in enumerate(text.splitlines(), 1):
        if line.rstrip() != line:
            out.append(f"{path}:{i}: trailing-ws: trailing whitespace")
        if "\t" in line:
            out.append(f"{path}:{i}: tab: tab character")
        if len(line) > line_max:
            out.append(f"{path}:{i}: line-length: {len(line)} > "
                       f"{line_max} columns")


def check_cxx(path: Path, out: list):
    text = path.read_text()
    check_common(path, text, CXX_LINE_MAX, out)
    for i, line in enumerate(text.splitlines(), 1):
        if re.search(r"\busing\s+namespace\s+std\b", line):
            out.append(f"{path}:{i}: using-namespace-std: "
                       "use explicit std:: qualification")
        if re.search(r"\b(malloc|calloc|realloc|free)\s*\(", line) \
                and "//" not in line.split("malloc")[0]:
            out.append(f"{path}:{i}: c-alloc: "
                       "use RAII/containers, not C allocation")


class PyVisitor(ast.NodeVisitor):
    def __init__(self, path: Path, out: list):
        self.path = path
        self.out = out
        self.imports: dict[str, int] = {}  # name -> lineno
        self.used: set[str] = set()

    def visit_Import(self, node: ast.Import):
        for a in node.names:
            name = (a.asname or a.name).split(".")[0]
            self.imports[name] = node.lineno
        self.generic_visit(node)

    def visit_ImportFrom(self, node: ast.ImportFrom):
        for a in node.names:
            if a.name == "*":
                continue
            self.imports[a.asname or a.name] = node.lineno
        self.generic_visit(node)

    def visit_Name(self, node: ast.Name):
        if isinstance(node.ctx, ast.Load):
            self.used.add(node.id)
        self.generic_visit(node)

    def visit_Attribute(self, node: ast.Attribute):
        n = node
        while isinstance(n, ast.Attribute):
            n = n.value
        if isinstance(n, ast.Name):
            self.used.add(n.id)
        self.generic_visit(node)

    def visit_ExceptHandler(self, node: ast.ExceptHandler):
        if node.type is None:
            self.out.append(f"{self.path}:{node.lineno}: bare-except: "
                            "catch specific exceptions")
        self.generic_visit(node)

    def _check_defaults(self, node):
        for d in list(node.args.defaults) + [
                d for d in node.args.kw_defaults if d is not None]:
            if isinstance(d, (ast.List, ast.Dict, ast.Set)):
                self.out.append(
                    f"{self.path}:{d.lineno}: mutable-default: "
                    "mutable default argument")

    def visit_FunctionDef(self, node):
        self._check_defaults(node)
        self.generic_visit(node)

    def visit_AsyncFunctionDef(self, node):
        self._check_defaults(node)
        self.generic_visit(node)

    def visit_Compare(self, node: ast.Compare):
        for op, cmp in zip(node.ops, node.comparators):
            if isinstance(op, (ast.Eq, ast.NotEq)) and \
                    isinstance(cmp, ast.Constant) and cmp.value is None:
                self.out.append(f"{self.path}:{node.lineno}: eq-none: "
                                "compare to None with is/is not")
        self.generic_visit(node)


def check_py(path: Path, out: list):
    text = path.read_text()
    check_common(path, text, PY_LINE_MAX, out)
    try:
        tree = ast.parse(text, filename=str(path))
    except SyntaxError as e:
        out.append(f"{path}:{e.lineno}: syntax: {e.msg}")
        return
    v = PyVisitor(path, out)
    v.visit(tree)
    # Unused imports: ignore conftest/__init__ (re-export patterns) and
    # names referenced in __all__ or string annotations.
    if path.name not in ("__init__.py", "conftest.py"):
        in_all = set()
        for node in tree.body:
            if isinstance(node, ast.Assign):
                for t in node.targets:
                    if isinstance(t, ast.Name) and t.id == "__all__" \
                            and isinstance(node.value, (ast.List,
                                                        ast.Tuple)):
                        for elt in node.value.elts:
                            if isinstance(elt, ast.Constant):
                                in_all.add(elt.value)
        for name, lineno in sorted(v.imports.items(),
                                   key=lambda kv: kv[1]):
            if name not in v.used and name not in in_all \
                    and name != "annotations":
                out.append(f"{path}:{lineno}: unused-import: {name}")


def gather():
    cxx, py = [], []
    for p in sorted((REPO / "native").rglob("*.[ch]pp")):
        cxx.append(p)
    for d in ("binder_amd", "tests", "scripts", "tools"):
        for p in sorted((REPO / d).rglob("*.py")):
            py.append(p)
    for p in sorted(REPO.glob("*.py")):
        py.append(p)
    return cxx, py


def main(argv):
    out: list[str] = []
    if len(argv) > 1:
        # explicit file list (used by tests to lint a scratch file)
        for a in argv[1:]:
            p = Path(a)
            if p.suffix in (".hpp", ".cpp", ".h", ".cc"):
                check_cxx(p, out)
            else:
                check_py(p, out)
    else:
        cxx, py = gather()
        for p in cxx:
            check_cxx(p, out)
        for p in py:
            check_py(p, out)
    for line in out:
        print(line)
    if out:
        print(f"lint: {len(out)} violation(s)", file=sys.stderr)
        return 1
    return 0


if __name__ == "__main__":
    sys.exit(main(sys.argv))
