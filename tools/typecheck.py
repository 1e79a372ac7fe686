#!/usr/bin/env python3
"""Static-check tier (behavior spec: internal/typecheck.py:41-100 — per-file
parallel type checking with py.typed opt-in; check.yml:10-36 ruff lint).

This image has no mypy/ruff wheels, so the tier degrades to what the stdlib
proves: every file must compile, and an AST pass enforces the lint rules the
repo actually relies on (no bare excepts, no tab characters).  py.typed
packages are listed so a mypy-equipped environment can run
`mypy $(python tools/typecheck.py --list-typed)` directly.

Usage: python tools/typecheck.py [--list-typed]
"""
from __future__ import annotations

import ast
import py_compile
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def typed_packages() -> list:
    return sorted(str(p.parent) for p in REPO.rglob("py.typed")
                  if "__pycache__" not in str(p))


def check_file(path: Path) -> list:
    errs = []
    try:
        py_compile.compile(str(path), doraise=True)
    except py_compile.PyCompileError as e:
        return [f"{path}: does not compile: {e.msg}"]
    src = path.read_text()
    if "\t" in src:
        errs.append(f"{path}: tab character")
    tree = ast.parse(src)
    for node in ast.walk(tree):
        if isinstance(node, ast.ExceptHandler) and node.type is None:
            errs.append(f"{path}:{node.lineno}: bare except")
    return errs


def main() -> int:
    if "--list-typed" in sys.argv:
        print("\n".join(typed_packages()))
        return 0
    errs = []
    for path in sorted((REPO / "modal_examples_amd").rglob("*.py")):
        if "__pycache__" in str(path):
            continue
        errs.extend(check_file(path))
    for e in errs:
        print(e)
    print(f"checked package: {'FAIL' if errs else 'OK'} "
          f"({len(errs)} findings); py.typed packages: "
          f"{len(typed_packages())}")
    return 1 if errs else 0


if __name__ == "__main__":
    sys.exit(main())
