"""observability/profiling + gpu/kernel_cache + parallel/collectives: the
CPU-verifiable halves of the profiling and compile-cache subsystems."""
import torch


def test_profile_call_writes_chrome_trace(tmp_path):
    from modal_examples_amd.observability.profiling import profile_call

    def work():
        a = torch.randn(64, 64)
        return (a @ a).sum()

    out = profile_call(work, trace_dir=str(tmp_path), steps=2, warmup=1)
    import json

    trace = json.load(open(out))
    assert "traceEvents" in trace and len(trace["traceEvents"]) > 10


def test_rocprof_command_builders_follow_safety_rule():
    """--pmc and trace domains must never be combined (node-stability rule);
    the builders keep them in separate invocations by construction."""
    from modal_examples_amd.observability.profiling import (
        rocprof_counters_command, rocprof_stats_command)

    stats = rocprof_stats_command(["python", "bench.py"])
    pmc = rocprof_counters_command(["python", "bench.py"])
    assert stats[0] == "rocprofv3" and "--stats" in stats and "--pmc" not in stats
    assert "--pmc" in pmc
    for banned in ("--stats", "-s", "--sys-trace", "-r", "--runtime-trace"):
        assert banned not in pmc
    assert stats[-2:] == ["python", "bench.py"]


def test_kernel_cache_roundtrip(tmp_path, monkeypatch):
    """capture() then restore() moves MIOpen find-db files atomically."""
    from modal_examples_amd.gpu import kernel_cache

    user_db = tmp_path / "userdb"
    user_db.mkdir()
    (user_db / "gfx950.ufdb.txt").write_text("shape1=algo7\n")
    monkeypatch.setenv("MIOPEN_USER_DB_PATH", str(user_db))

    cache = tmp_path / "cache"
    assert kernel_cache.capture(cache_dir=cache) == 1

    # wipe the user db (fresh box) and restore from the shipped cache
    (user_db / "gfx950.ufdb.txt").unlink()
    assert kernel_cache.restore(cache_dir=cache) == 1
    assert (user_db / "gfx950.ufdb.txt").read_text() == "shape1=algo7\n"
    # idempotent (same size → no rewrite, still counted)
    assert kernel_cache.restore(cache_dir=cache) == 1


def test_shipped_find_db_is_present():
    """The in-tree MIOpen find-db that kills the cold-start conv search must
    ship with the repo (gpu/miopen_udb/)."""
    from modal_examples_amd.gpu.kernel_cache import DEFAULT_CACHE

    assert DEFAULT_CACHE.is_dir()
    assert any(DEFAULT_CACHE.iterdir()), "find-db cache is empty"


def test_collectives_single_process_fallbacks():
    """Without torch.distributed initialized the helpers degrade to no-ops
    (rank 0, world 1) so single-GPU paths share the multi-GPU code."""
    from modal_examples_amd.parallel.collectives import (all_reduce_mean,
                                                         broadcast_module,
                                                         init_distributed)

    rank, world, device = init_distributed(backend="gloo")
    try:
        lin = torch.nn.Linear(4, 4)
        before = lin.weight.clone()
        broadcast_module(lin)
        assert torch.equal(lin.weight, before)
        t = torch.ones(3)
        assert torch.equal(all_reduce_mean(t.clone()), t)
    finally:
        import torch.distributed as dist

        if dist.is_initialized():
            dist.destroy_process_group()
