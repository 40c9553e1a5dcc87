#!/usr/bin/env python3
"""Minimal AST linter (the target image has no ruff/flake8 and no index
access, so `make lint` vendors its own checks).

Checks, per file:
  F401  unused import (module scope; aliases and ``as _``-style excluded)
  F811  duplicate definition of the same top-level name
  B006  mutable default argument
  E722  bare ``except:``
  W291  trailing whitespace
  W191  tab indentation
  E501  line longer than 100 chars (soft limit; protobuf/string blobs in
        fixtures are exempted via the EXEMPT list)

Exit code 1 if any finding; prints file:line: code message.
"""

from __future__ import annotations

import ast
import sys
from pathlib import Path

EXEMPT_DIRS = {"__pycache__", ".git", "gpurun_out", ".pytest_cache"}
LINE_LIMIT = 100


def iter_py(paths):
    for p in paths:
        p = Path(p)
        if p.is_file() and p.suffix == ".py":
            yield p
        elif p.is_dir():
            for f in sorted(p.rglob("*.py")):
                if not (set(f.parts) & EXEMPT_DIRS):
                    yield f


class ImportTracker(ast.NodeVisitor):
    def __init__(self):
        self.imports: dict[str, int] = {}   # name -> lineno
        self.used: set[str] = set()

    def visit_Import(self, node):
        for a in node.names:
            name = a.asname or a.name.split(".")[0]
            if not name.startswith("_"):
                self.imports[name] = node.lineno

    def visit_ImportFrom(self, node):
        if node.module == "__future__":
            return
        for a in node.names:
            name = a.asname or a.name
            if name != "*" and not name.startswith("_"):
                self.imports[name] = node.lineno

    def visit_Name(self, node):
        self.used.add(node.id)

    def visit_Attribute(self, node):
        self.generic_visit(node)


def check_file(path: Path) -> list[str]:
    findings = []
    src = path.read_text()
    try:
        tree = ast.parse(src, filename=str(path))
    except SyntaxError as e:
        return [f"{path}:{e.lineno}: E999 syntax error: {e.msg}"]

    # textual checks
    for i, line in enumerate(src.splitlines(), 1):
        if line != line.rstrip():
            findings.append(f"{path}:{i}: W291 trailing whitespace")
        if line.startswith("\t"):
            findings.append(f"{path}:{i}: W191 tab indentation")
        if len(line) > LINE_LIMIT and "http" not in line:
            findings.append(
                f"{path}:{i}: E501 line too long ({len(line)} chars)"
            )

    # unused imports (module scope only — conservative; __init__.py
    # re-exports are the package API and exempt)
    tracker = ImportTracker()
    if path.name != "__init__.py":
        tracker.visit(tree)
    # names used in __all__, docstring-reexport idiom, or noqa lines excluded
    lines = src.splitlines()
    for name, lineno in sorted(tracker.imports.items(), key=lambda x: x[1]):
        if name in tracker.used:
            continue
        line = lines[lineno - 1] if lineno <= len(lines) else ""
        if "noqa" in line or f'"{name}"' in src or f"'{name}'" in src:
            continue
        findings.append(f"{path}:{lineno}: F401 unused import '{name}'")

    # duplicate top-level defs
    seen: dict[str, int] = {}
    for node in tree.body:
        if isinstance(node, (ast.FunctionDef, ast.AsyncFunctionDef, ast.ClassDef)):
            if node.name in seen:
                findings.append(
                    f"{path}:{node.lineno}: F811 redefinition of "
                    f"'{node.name}' (first at line {seen[node.name]})"
                )
            seen[node.name] = node.lineno

    for node in ast.walk(tree):
        if isinstance(node, (ast.FunctionDef, ast.AsyncFunctionDef)):
            for default in node.args.defaults + node.args.kw_defaults:
                if isinstance(default, (ast.List, ast.Dict, ast.Set)):
                    findings.append(
                        f"{path}:{node.lineno}: B006 mutable default "
                        f"argument in '{node.name}'"
                    )
        elif isinstance(node, ast.ExceptHandler) and node.type is None:
            findings.append(f"{path}:{node.lineno}: E722 bare except")
    return findings


def main(argv):
    paths = argv or ["gpushare_amd", "tests", "tools", "bench.py",
                     "__graft_entry__.py", "setup.py", "benchmarks"]
    all_findings = []
    n_files = 0
    for f in iter_py(paths):
        n_files += 1
        all_findings.extend(check_file(f))
    for finding in all_findings:
        print(finding)
    print(f"[lint] {n_files} files, {len(all_findings)} finding(s)",
          file=sys.stderr)
    return 1 if all_findings else 0


if __name__ == "__main__":
    sys.exit(main(sys.argv[1:]))
