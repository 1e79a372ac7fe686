#!/usr/bin/env python3
"""Which examples changed vs a git ref? → the CI fan-out matrix
(the diff-driven runner tier, reference internal/generate_diff_matrix.py).

Usage: python tools/generate_diff_matrix.py [BASE_REF]
Prints a JSON list of example stems whose files (or the runtime they import)
changed; CI runs `python tools/run_example.py <stem>` per entry.
"""
import json
import subprocess
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent))
from example_utils import get_examples  # noqa: E402

REPO = Path(__file__).resolve().parent.parent


def changed_files(base: str) -> list:
    out = subprocess.run(
        ["git", "diff", "--name-only", base, "--"], cwd=REPO,
        capture_output=True, text=True, check=True).stdout
    return [l for l in out.splitlines() if l.strip()]


def main():
    base = sys.argv[1] if len(sys.argv) > 1 else "HEAD~1"
    files = set(changed_files(base))
    runtime_changed = any(f.startswith(("modal_examples_amd/", "tools/"))
                          for f in files)
    stems = []
    for ex in get_examples():
        rel = str(ex.path.relative_to(REPO))
        if rel in files or runtime_changed:
            stems.append(ex.stem)
    print(json.dumps({"runtime_changed": runtime_changed,
                      "examples": sorted(set(stems))}, indent=1))
    return 0


if __name__ == "__main__":
    sys.exit(main())
