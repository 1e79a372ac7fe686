"""Example-corpus CI tiers (behavior spec at SURVEY.md §4):
1. every example imports cleanly without GPUs,
2. literate-markdown rendering succeeds,
3. manifest generation works,
4. a fast subset actually RUNS end-to-end via its frontmatter cmd.
"""
import importlib.util
import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO / "tools"))

from example_utils import (  # noqa: E402
    example_manifest,
    get_examples,
    parse_frontmatter,
    render_example_md,
)

EXAMPLES = get_examples()
IDS = [str(e.path.relative_to(REPO / "examples")) for e in EXAMPLES]


def test_discovery_finds_corpus():
    assert len(EXAMPLES) >= 20
    stems = {e.stem for e in EXAMPLES}
    for must in ("hello_world", "text_to_image", "batched_whisper",
                 "hp_sweep_gpt", "simple_torch_cluster", "dicts_and_queues"):
        assert must in stems, f"missing canonical example {must}"


@pytest.mark.parametrize("ex", EXAMPLES, ids=IDS)
def test_example_imports(ex):
    spec = importlib.util.spec_from_file_location(f"ex_{ex.stem}", ex.path)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    import modal_examples_amd as modal

    assert any(isinstance(v, modal.App) for v in vars(mod).values()), \
        f"{ex.stem} defines no App"


@pytest.mark.parametrize("ex", EXAMPLES, ids=IDS)
def test_example_renders_markdown(ex):
    md = render_example_md(ex.path)
    assert "```python" in md
    assert "# ---" not in md.splitlines()[0:1]


@pytest.mark.parametrize("ex", EXAMPLES, ids=IDS)
def test_example_frontmatter_cmd(ex):
    fm = parse_frontmatter(ex.path)
    assert "cmd" in fm, f"{ex.stem} missing frontmatter cmd"
    # either the CLI runner or a plain `python file.py` (programmatic
    # app.run() examples, e.g. import_sklearn.py — the reference has both)
    assert fm["cmd"][0] == "python", fm["cmd"]
    if fm["cmd"][1] == "-m":
        assert fm["cmd"][1:3] == ["-m", "modal_examples_amd"]
    else:
        assert fm["cmd"][1].endswith(".py")


def test_manifest_json():
    import json

    m = json.loads(example_manifest())
    assert len(m) == len(EXAMPLES)


FAST_RUN = ["hello_world", "generators", "basic_grid_search",
            "dynamic_batching", "parallel_execution", "secret_env",
            "dicts_and_queues", "volume_ingest", "safe_code_execution",
            "pushgateway_metrics", "restricted_volume", "code_interpreter",
            "flask_app", "mcp_server", "tensor_parallel",
            "pipeline_orchestration", "doc_ocr_webapp", "feed_alerts"]


@pytest.mark.parametrize("stem", FAST_RUN)
def test_example_runs_end_to_end(stem):
    r = subprocess.run(
        [sys.executable, str(REPO / "tools" / "run_example.py"), stem,
         "--timeout", "240"],
        capture_output=True, text=True, cwd=REPO, timeout=300,
    )
    assert r.returncode == 0, f"{stem} failed:\n{r.stdout[-2000:]}\n{r.stderr[-2000:]}"


def test_diff_matrix_tool():
    """tools/generate_diff_matrix.py: diff-driven CI fan-out (reference
    internal/generate_diff_matrix.py role) emits valid JSON and flags
    runtime-wide changes."""
    import json

    r = subprocess.run(
        [sys.executable, str(REPO / "tools" / "generate_diff_matrix.py"),
         "HEAD~3"],
        capture_output=True, text=True, cwd=REPO, timeout=60)
    assert r.returncode == 0, r.stderr[-1000:]
    out = json.loads(r.stdout)
    assert set(out) == {"runtime_changed", "examples"}
    assert isinstance(out["examples"], list)
    if out["runtime_changed"]:  # runtime edits fan out to every example
        assert len(out["examples"]) == len(EXAMPLES)
