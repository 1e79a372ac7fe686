"""Driver-facing contracts: __graft_entry__ exposes build()/smoke(), bench.py
parses its contract flags and defaults to N=1."""
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def test_graft_entry_surface():
    sys.path.insert(0, str(REPO))
    import __graft_entry__ as g

    assert callable(g.build) and callable(g.smoke)
    import inspect

    assert not any(p.default is inspect.Parameter.empty
                   for p in inspect.signature(g.build).parameters.values())
    assert not any(p.default is inspect.Parameter.empty
                   for p in inspect.signature(g.smoke).parameters.values())


def test_bench_help_and_defaults():
    r = subprocess.run([sys.executable, "bench.py", "--help"],
                       capture_output=True, text=True, cwd=REPO, timeout=120)
    assert r.returncode == 0
    for flag in ("--gpus", "--steps", "--warmup"):
        assert flag in r.stdout


def test_typecheck_tier_passes():
    """tools/typecheck.py (internal/typecheck.py role) is green."""
    import subprocess
    import sys

    r = subprocess.run([sys.executable, "tools/typecheck.py"],
                       capture_output=True, text=True, timeout=120)
    assert r.returncode == 0, r.stdout + r.stderr


def test_deploy_all_cd_tier():
    """tools/deploy_all.py (cd.yml / internal/deploy.py role) deploys every
    `deploy: true` example."""
    import subprocess
    import sys

    r = subprocess.run([sys.executable, "tools/deploy_all.py"],
                       capture_output=True, text=True, timeout=280)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "deployed 3/3" in r.stdout, r.stdout


def test_migration_doc_api_claims_hold():
    """Every `modal.X` symbol MIGRATION.md claims is 'same' actually
    exists, and every LlamaEngine kwarg in its flag table is real."""
    import inspect

    import modal_examples_amd as m
    from modal_examples_amd.models.llama.engine import LlamaEngine

    for name in ["App", "Volume", "Dict", "Queue", "Secret",
                 "NetworkFileSystem", "CloudBucketMount", "Image", "Sandbox",
                 "FunctionCall", "method", "enter", "exit", "batched",
                 "concurrent", "parameter", "fastapi_endpoint", "asgi_app",
                 "wsgi_app", "web_server", "forward"]:
        assert hasattr(m, name), name
    assert hasattr(m.experimental, "clustered")
    sig = inspect.signature(LlamaEngine.__init__)
    for kw in ["tp", "spec_tokens", "chunked_prefill", "prefix_cache",
               "gpu_mem_util", "kv_dtype", "use_graph", "init_weights"]:
        assert kw in sig.parameters, kw
