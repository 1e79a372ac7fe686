#!/usr/bin/env python3
"""Run one example by stem, honoring its frontmatter `cmd` — the end-to-end
CI tier (behavior spec: internal/run_example.py:17-45: frontmatter cmd,
14-minute timeout, env overrides, exit-code = pass/fail).

Usage: python tools/run_example.py hello_world [--timeout 840]
"""
from __future__ import annotations

import argparse
import os
import subprocess
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent))
from example_utils import get_examples  # noqa: E402

REPO = Path(__file__).resolve().parent.parent


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("stem")
    ap.add_argument("--timeout", type=float, default=840)
    args = ap.parse_args()

    matches = [e for e in get_examples() if e.stem == args.stem]
    if not matches:
        print(f"no example with stem {args.stem!r}", file=sys.stderr)
        return 2
    ex = matches[0]
    cmd = ex.cmd or [sys.executable, "-m", "modal_examples_amd", "run",
                     str(ex.path.relative_to(REPO))]
    cmd = list(cmd) + ex.args
    env = dict(os.environ)
    env.update(ex.env)
    env.setdefault("MODAL_SERVE_TIMEOUT", "5.0")
    print("+", " ".join(cmd), flush=True)
    try:
        r = subprocess.run(cmd, cwd=REPO, env=env, timeout=args.timeout)
        return r.returncode
    except subprocess.TimeoutExpired:
        print(f"timed out after {args.timeout}s", file=sys.stderr)
        return 3


if __name__ == "__main__":
    sys.exit(main())
