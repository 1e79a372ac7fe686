"""Tensor-parallel layers (parallel/tp.py): sharded column→row pairs must
reproduce the full single-process result bit-for-bit-ish over gloo world 2
(the CPU stand-in for RCCL over xGMI)."""
import os
import socket

import pytest

import torch


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _tp_worker(rank, world, port, results):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world), "LOCAL_RANK": str(rank),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    import torch.distributed as dist

    from modal_examples_amd.parallel.tp import (ColumnParallelLinear,
                                                RowParallelLinear, TPGroup,
                                                TPMLP, shard_linear)

    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        torch.manual_seed(0)  # same full weights on every rank
        d, ff, B = 16, 32, 3
        up = torch.nn.Linear(d, ff)
        down = torch.nn.Linear(ff, d)
        x = torch.randn(B, d)
        ref = down(torch.nn.functional.gelu(up(x), approximate="tanh"))

        tp = TPGroup()
        up_s = shard_linear(up, "column", tp)
        down_s = shard_linear(down, "row", tp)
        y = down_s(torch.nn.functional.gelu(up_s(x), approximate="tanh"))
        results[f"pair-{rank}"] = float((y - ref).abs().max())

        # gather_output round-trips the column shard to the full tensor
        full_col = shard_linear(up, "column", tp)
        full_col.gather_output = True
        results[f"gather-{rank}"] = float((full_col(x) - up(x)).abs().max())

        # row layer can split a replicated input itself
        row2 = shard_linear(down, "row", tp)
        row2.input_is_parallel = False
        h = torch.nn.functional.gelu(up(x), approximate="tanh")
        results[f"split-{rank}"] = float((row2(h) - down(h)).abs().max())

        # TPMLP trains: grads flow to both shards
        mlp = TPMLP(d, ff)
        mlp(x).sum().backward()
        results[f"grad-{rank}"] = (mlp.up.linear.weight.grad is not None
                                   and mlp.down.linear.weight.grad is not None)
    finally:
        dist.destroy_process_group()


def test_tp_pair_matches_full_model_world2():
    import multiprocessing as mp

    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = _free_port()
        procs = [ctx.Process(target=_tp_worker, args=(r, 2, port, results))
                 for r in range(2)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=120)
            assert p.exitcode == 0, f"tp worker exit {p.exitcode}"
        for r in range(2):
            assert results[f"pair-{r}"] < 1e-5
            assert results[f"gather-{r}"] < 1e-6
            assert results[f"split-{r}"] < 1e-5
            assert results[f"grad-{r}"] is True


def test_tp_single_process_degenerates_to_local():
    """world=1 (no dist init): layers behave as plain Linears."""
    from modal_examples_amd.parallel.tp import TPMLP, shard_linear

    torch.manual_seed(1)
    lin = torch.nn.Linear(8, 8)
    col = shard_linear(lin, "column")
    row = shard_linear(lin, "row")
    x = torch.randn(2, 8)
    assert torch.allclose(col(x), lin(x), atol=1e-6)
    assert torch.allclose(row(x), lin(x), atol=1e-6)
    mlp = TPMLP(8, 16)
    assert mlp(x).shape == (2, 8)


def test_clustered_launches_all_ranks_gloo():
    """@clustered(size=4): fn.remote() launches 4 rank workers with injected
    rendezvous env; a gloo all_reduce proves they form one process group;
    caller gets rank 0's result (14_clusters contract, now runtime-wired)."""
    import modal_examples_amd as modal

    app = modal.App("test-clustered-gloo")

    @app.function(timeout=120)
    @modal.experimental.clustered(size=4)
    def rank_sum() -> float:
        import os

        import torch
        import torch.distributed as dist

        from modal_examples_amd.parallel.cluster import get_cluster_info

        info = get_cluster_info()
        assert len(info.container_ips) == 4
        assert os.environ["MASTER_ADDR"] == "127.0.0.1"
        dist.init_process_group("gloo", init_method="env://")
        t = torch.tensor([float(info.rank)])
        dist.all_reduce(t)
        dist.barrier()
        dist.destroy_process_group()
        return float(t.item()) + 100.0 * info.rank  # rank-dependent result

    out = rank_sum.remote()
    assert out == 6.0  # 0+1+2+3, rank 0's value (no +100 offset)


def test_clustered_rank_failure_propagates():
    import modal_examples_amd as modal

    app = modal.App("test-clustered-fail")

    @app.function(timeout=60, retries=0)
    @modal.experimental.clustered(size=2)
    def fail_on_rank1() -> int:
        from modal_examples_amd.parallel.cluster import get_cluster_info

        if get_cluster_info().rank == 1:
            raise RuntimeError("rank 1 exploded")
        return 7

    with pytest.raises(Exception, match="rank 1 exploded"):
        fail_on_rank1.remote()


def test_clustered_spawn_gather():
    import modal_examples_amd as modal

    app = modal.App("test-clustered-spawn")

    @app.function(timeout=60)
    @modal.experimental.clustered(size=2)
    def whoami() -> int:
        from modal_examples_amd.parallel.cluster import get_cluster_info

        return get_cluster_info().rank

    fc = whoami.spawn()
    assert fc.get(timeout=60) == 0


def test_all_reduce_smart_matches_ring_both_regimes():
    """Size-aware all-reduce: the one-shot small-message path and the ring
    large-message path both produce the exact all_reduce sum (gloo world 4,
    SURVEY §5.8 algorithm-selection row)."""
    import modal_examples_amd as modal

    app = modal.App("test-smart-allreduce")

    @app.function(timeout=120)
    @modal.experimental.clustered(size=4)
    def reduce_both() -> list:
        import torch
        import torch.distributed as dist

        from modal_examples_amd.parallel.collectives import all_reduce_smart

        dist.init_process_group("gloo", init_method="env://")
        r = dist.get_rank()
        small = torch.full((64,), float(r + 1))          # one-shot path
        big = torch.full((64 * 1024,), float(r + 1))     # ring path (256 KiB)
        all_reduce_smart(small)
        all_reduce_smart(big)
        want = float(sum(range(1, 5)))
        ok = bool(small.eq(want).all()) and bool(big.eq(want).all())
        dist.barrier()
        dist.destroy_process_group()
        return [ok, float(small[0]), float(big[0])]

    out = reduce_both.remote()
    assert out[0], out


def _llama_tp_worker(rank, world, port, results):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world), "LOCAL_RANK": str(rank),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    import torch.distributed as dist

    from modal_examples_amd.models.llama.model import (LlamaConfig,
                                                       LlamaModel,
                                                       shard_llama_state)
    from modal_examples_amd.parallel.tp import TPGroup

    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        cfg = LlamaConfig.small()
        torch.manual_seed(0)  # same FULL model on every rank
        full = LlamaModel(cfg).to(torch.bfloat16)
        tp_model = LlamaModel(cfg, tp=TPGroup()).to(torch.bfloat16)
        tp_model.load_state_dict(
            shard_llama_state(dict(full.state_dict()), cfg, rank, world))
        toks = torch.randint(0, cfg.vocab_size, (2, 10),
                             generator=torch.Generator().manual_seed(7))
        ref = full.prefill(toks)
        got = tp_model.prefill(toks)
        results[f"err-{rank}"] = float((ref - got).abs().max())
        # per-rank KV writer sees only the local head shard
        shards = {}
        tp_model.prefill(toks, kv_writer=lambda li, k, v:
                         shards.setdefault(li, k.shape))
        results[f"kvshape-{rank}"] = shards[0]
    finally:
        dist.destroy_process_group()


def test_llama_tp_world2_matches_full_model():
    """Head-sharded TP LlamaModel == full model logits (gloo world 2)."""
    import multiprocessing as mp

    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = _free_port()
        procs = [ctx.Process(target=_llama_tp_worker, args=(r, 2, port, results))
                 for r in range(2)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=180)
            assert p.exitcode == 0, f"llama tp worker exit {p.exitcode}"
        for r in range(2):
            # bf16 partial-sum rounding across the reduce: loose but tight
            # enough to catch any mis-sharding (wrong slices are O(1) off)
            assert results[f"err-{r}"] < 0.05, results[f"err-{r}"]
            assert results[f"kvshape-{r}"] == (2, 10, 1, 64)  # nkv 2 -> 1/rank


def _llama_tp_engine_worker(rank, world, port, results):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world), "LOCAL_RANK": str(rank),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    import torch.distributed as dist

    from modal_examples_amd.models.llama.engine import LlamaEngine
    from modal_examples_amd.models.llama.model import (LlamaConfig,
                                                       LlamaModel,
                                                       shard_llama_state)
    from modal_examples_amd.parallel.tp import TPGroup

    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        cfg = LlamaConfig.small()
        torch.manual_seed(0)
        full_state = dict(LlamaModel(cfg).to(torch.bfloat16).state_dict())

        def drive(eng):
            g = torch.Generator().manual_seed(11)
            for i in range(3):
                prompt = torch.randint(0, cfg.vocab_size, (6 + i,),
                                       generator=g).tolist()
                eng.add_request(prompt, max_new_tokens=5, temperature=0.0)
            eng.run_until_done(max_steps=200)
            return [eng.finished[i].out_tokens for i in sorted(eng.finished)]

        full = LlamaEngine(cfg, device="cpu", dtype=torch.bfloat16,
                           use_graph=False, eos_id=-1, seed=3)
        full.model.load_state_dict(full_state)
        want = drive(full)

        tp_eng = LlamaEngine(cfg, device="cpu", dtype=torch.bfloat16,
                             use_graph=False, eos_id=-1, seed=3, tp=TPGroup())
        tp_eng.model.load_state_dict(
            shard_llama_state(full_state, cfg, rank, world))
        got = drive(tp_eng)
        results[f"match-{rank}"] = (got == want)
        results[f"toks-{rank}"] = got
    finally:
        dist.destroy_process_group()


def test_llama_tp_engine_world2_continuous_batching():
    """TP=2 engines (each rank: sharded KV cache + in-model all-reduces)
    produce token-for-token the single-engine output under continuous
    batching, and identically on both ranks (no sampling divergence)."""
    import multiprocessing as mp

    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = _free_port()
        procs = [ctx.Process(target=_llama_tp_engine_worker,
                             args=(r, 2, port, results)) for r in range(2)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=240)
            assert p.exitcode == 0, f"tp engine worker exit {p.exitcode}"
        assert results["match-0"] is True and results["match-1"] is True
        assert results["toks-0"] == results["toks-1"]


def _tp_coldboot_worker(rank, world, port, results):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world), "LOCAL_RANK": str(rank),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    import tempfile

    import torch.distributed as dist

    from modal_examples_amd.gpu import fastload
    from modal_examples_amd.models.llama.engine import LlamaEngine
    from modal_examples_amd.models.llama.model import LlamaConfig, LlamaModel
    from modal_examples_amd.parallel.tp import TPGroup

    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        cfg = LlamaConfig.small()
        torch.manual_seed(0)
        full = LlamaModel(cfg).to(torch.bfloat16)
        path = os.path.join(tempfile.gettempdir(), f"tpboot-{port}.safetensors")
        if rank == 0:
            fastload.save_file(dict(full.state_dict()), path)
        dist.barrier()
        eng = LlamaEngine.from_safetensors(
            path, cfg=cfg, device="cpu", dtype=torch.bfloat16,
            use_graph=False, tp=TPGroup())
        toks = torch.randint(0, cfg.vocab_size, (1, 8),
                             generator=torch.Generator().manual_seed(2))
        results[f"err-{rank}"] = float(
            (full.prefill(toks) - eng.model.prefill(toks)).abs().max())
    finally:
        dist.destroy_process_group()


def test_tp_cold_boot_shards_baked_checkpoint():
    """from_safetensors(tp=...) carves each rank's shard from the FULL
    baked file; restored TP engine matches the full model."""
    import multiprocessing as mp

    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = _free_port()
        procs = [ctx.Process(target=_tp_coldboot_worker,
                             args=(r, 2, port, results)) for r in range(2)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=240)
            assert p.exitcode == 0, f"tp coldboot worker exit {p.exitcode}"
        for r in range(2):
            assert results[f"err-{r}"] < 0.05, results[f"err-{r}"]
