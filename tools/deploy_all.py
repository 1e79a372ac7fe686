#!/usr/bin/env python3
"""CD tier: deploy every example whose frontmatter opts in with
`deploy: true` (behavior spec: the reference's daily CD — cd.yml +
internal/deploy.py:19-53 deploys all flagged examples).

Each deploy runs in its own subprocess; apps with active schedules keep
their process alive (cli.py:cmd_deploy), so those are left running only if
`--keep-schedulers` is passed, otherwise terminated once registration is
confirmed.

Usage: python tools/deploy_all.py [--keep-schedulers] [--timeout 120]
"""
from __future__ import annotations

import argparse
import subprocess
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent))
from example_utils import get_examples  # noqa: E402

REPO = Path(__file__).resolve().parent.parent


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--timeout", type=float, default=120)
    ap.add_argument("--keep-schedulers", action="store_true")
    args = ap.parse_args()

    targets = [e for e in get_examples() if e.deploy]
    if not targets:
        print("no `deploy: true` examples found")
        return 1
    failed = []
    for e in targets:
        p = subprocess.Popen(
            [sys.executable, "-m", "modal_examples_amd", "deploy",
             str(e.path)], cwd=REPO, stdout=subprocess.PIPE,
            stderr=subprocess.STDOUT, text=True)
        deadline = time.monotonic() + args.timeout
        out_lines = []
        ok = False
        while time.monotonic() < deadline:
            line = p.stdout.readline()
            if not line:
                break
            out_lines.append(line)
            if line.startswith("deployed "):
                ok = True
                if "schedules active" not in "".join(out_lines) \
                        and p.poll() is None:
                    pass  # non-scheduled deploys exit on their own
                break
        if ok and p.poll() is None and not args.keep_schedulers:
            p.terminate()  # scheduled app keep-alive: registration is done
        elif not ok:
            p.terminate()
        p.wait(timeout=15)
        status = "ok" if ok else "FAILED"
        print(f"  {e.stem}: {status}")
        if not ok:
            failed.append((e.stem, "".join(out_lines)[-500:]))
    print(f"deployed {len(targets) - len(failed)}/{len(targets)} flagged "
          f"examples")
    for stem, tail in failed:
        print(f"--- {stem} ---\n{tail}")
    return 1 if failed else 0


if __name__ == "__main__":
    sys.exit(main())
