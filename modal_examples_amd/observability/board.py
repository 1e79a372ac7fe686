"""TensorBoard-on-Volume role: scalar event logging + a wsgi dashboard.

The reference serves TensorBoard as a `@modal.wsgi_app` over a Volume with a
reload-before-request middleware (hp_sweep_gpt.py:396-414,
torch_profiling.py:294-316).  There is no network to fetch the tensorboard
package here, so this module fills the role natively: training code appends
scalar events to JSONL files on a Volume; `make_board_wsgi` serves an HTML
dashboard (inline SVG charts, zero JS dependencies) plus a JSON API; and
`VolumeReloadMiddleware` calls `volume.reload()` before every request so a
dashboard container always sees the latest committed events — the exact
middleware contract of the reference.
"""
from __future__ import annotations

import json
import time
from pathlib import Path


def log_scalar(run_dir, tag: str, step: int, value: float) -> None:
    """Append one scalar event (the SummaryWriter.add_scalar role)."""
    d = Path(run_dir)
    d.mkdir(parents=True, exist_ok=True)
    with open(d / "events.jsonl", "a") as f:
        f.write(json.dumps({"tag": tag, "step": step, "value": float(value),
                            "wall": time.time()}) + "\n")


def read_runs(logdir) -> dict:
    """{run_name: {tag: [(step, value), ...]}} from every events.jsonl."""
    out: dict = {}
    root = Path(logdir)
    if not root.exists():
        return out
    for ev in sorted(root.glob("**/events.jsonl")):
        run = str(ev.parent.relative_to(root)) or "."
        tags = out.setdefault(run, {})
        for line in ev.read_text().splitlines():
            try:
                e = json.loads(line)
            except json.JSONDecodeError:
                continue
            tags.setdefault(e["tag"], []).append((e["step"], e["value"]))
    for tags in out.values():
        for pts in tags.values():
            pts.sort()
    return out


def _svg_chart(pts, width=480, height=160) -> str:
    if len(pts) < 2:
        return "<svg></svg>"
    xs = [p[0] for p in pts]
    ys = [p[1] for p in pts]
    x0, x1 = min(xs), max(xs) or 1
    y0, y1 = min(ys), max(ys)
    if y1 == y0:
        y1 = y0 + 1
    pad = 8
    sx = lambda x: pad + (x - x0) / max(1e-9, x1 - x0) * (width - 2 * pad)  # noqa: E731
    sy = lambda y: height - pad - (y - y0) / (y1 - y0) * (height - 2 * pad)  # noqa: E731
    path = " ".join(f"{'M' if i == 0 else 'L'}{sx(x):.1f},{sy(y):.1f}"
                    for i, (x, y) in enumerate(pts))
    return (f'<svg width="{width}" height="{height}" style="background:#fafafa;'
            f'border:1px solid #ddd"><path d="{path}" fill="none" '
            f'stroke="#1f77b4" stroke-width="1.5"/>'
            f'<text x="{pad}" y="{pad + 8}" font-size="10">{y1:.4g}</text>'
            f'<text x="{pad}" y="{height - 2}" font-size="10">{y0:.4g}</text></svg>')


class VolumeReloadMiddleware:
    """wsgi middleware: volume.reload() before each request so the dashboard
    sees the latest committed training events (torch_profiling.py:294-316)."""

    def __init__(self, app, volume):
        self.app = app
        self.volume = volume

    def __call__(self, environ, start_response):
        try:
            self.volume.reload()
        except Exception:
            pass
        return self.app(environ, start_response)


def make_board_wsgi(logdir):
    """A TensorBoard-role wsgi app over ``logdir``: '/' = HTML dashboard,
    '/data' = the parsed scalars as JSON."""

    def board(environ, start_response):
        path = environ.get("PATH_INFO", "/")
        runs = read_runs(logdir)
        if path.rstrip("/").endswith("data") and path != "/":
            body = json.dumps(runs).encode()
            start_response("200 OK", [("Content-Type", "application/json")])
            return [body]
        parts = ["<html><head><title>board</title></head><body>",
                 f"<h2>scalar board — {len(runs)} run(s)</h2>"]
        for run, tags in sorted(runs.items()):
            parts.append(f"<h3>{run}</h3>")
            for tag, pts in sorted(tags.items()):
                parts.append(f"<div><b>{tag}</b> ({len(pts)} points)<br>"
                             f"{_svg_chart(pts)}</div>")
        parts.append("</body></html>")
        body = "".join(parts).encode()
        start_response("200 OK", [("Content-Type", "text/html")])
        return [body]

    return board
