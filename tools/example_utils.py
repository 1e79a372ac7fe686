"""Example-corpus harness: discovery, frontmatter, literate-markdown render.

The local analog of the reference's internal CI utilities (behavior spec:
internal/utils.py:131-178 — frontmatter keys cmd/args/env/pytest/deploy,
two-level example discovery; internal/utils.py:51-99 — markdown rendering of
comment-prose examples).
"""
from __future__ import annotations

import json
import re
from dataclasses import dataclass, field
from pathlib import Path
from typing import List, Optional

EXAMPLES_ROOT = Path(__file__).resolve().parent.parent / "examples"


@dataclass
class Example:
    stem: str
    path: Path
    cmd: Optional[List[str]] = None
    args: List[str] = field(default_factory=list)
    env: dict = field(default_factory=dict)
    run_pytest: bool = True
    deploy: bool = False
    tags: List[str] = field(default_factory=list)


def parse_frontmatter(path: Path) -> dict:
    """YAML-ish frontmatter between `# ---` fences at the top of the file."""
    import yaml

    lines = path.read_text().splitlines()
    if lines and lines[0].startswith("#!"):
        lines = lines[1:]  # tolerate a shebang (misc/hello_shebang.py)
    if not lines or lines[0].strip() != "# ---":
        return {}
    block = []
    for line in lines[1:]:
        if line.strip() == "# ---":
            break
        block.append(re.sub(r"^# ?", "", line))
    try:
        return yaml.safe_load("\n".join(block)) or {}
    except Exception:
        return {}


def get_examples(root: Path = EXAMPLES_ROOT) -> List[Example]:
    """Walk numbered directories up to two levels deep (reference discovery
    shape, internal/utils.py:160-178)."""
    out = []
    for p in sorted(root.rglob("*.py")):
        rel = p.relative_to(root)
        if len(rel.parts) > 3 or rel.name.startswith("_"):
            continue
        fm = parse_frontmatter(p)
        out.append(Example(
            stem=p.stem,
            path=p,
            cmd=fm.get("cmd"),
            args=[str(a) for a in fm.get("args", [])],
            env={k: str(v) for k, v in (fm.get("env") or {}).items()},
            run_pytest=fm.get("pytest", True),
            deploy=bool(fm.get("deploy", False)),
            tags=list(fm.get("tags", [])),
        ))
    return out


def render_example_md(path: Path) -> str:
    """Literate render: leading `#` comment blocks become markdown prose,
    code becomes fenced blocks (reference renderer behavior)."""
    lines = path.read_text().splitlines()
    out, code_buf = [], []
    i = 0
    # skip frontmatter
    if lines and lines[0].strip() == "# ---":
        i = 1
        while i < len(lines) and lines[i].strip() != "# ---":
            i += 1
        i += 1

    def flush_code():
        if code_buf:
            body = "\n".join(code_buf).strip("\n")
            if body:
                out.append(f"```python\n{body}\n```")
            code_buf.clear()

    while i < len(lines):
        line = lines[i]
        if line.startswith("# ") or line == "#":
            flush_code()
            out.append(line[2:] if len(line) > 2 else "")
        else:
            code_buf.append(line)
        i += 1
    flush_code()
    return "\n".join(out) + "\n"


def example_manifest(root: Path = EXAMPLES_ROOT) -> str:
    return json.dumps(
        [{"stem": e.stem, "path": str(e.path.relative_to(root.parent)),
          "cmd": e.cmd, "deploy": e.deploy} for e in get_examples(root)],
        indent=1)
