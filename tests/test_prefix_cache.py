"""Block-level KV prefix caching (RadixAttention/automatic-prefix-caching
role, inkling_small.py:99): shared prompt prefixes reuse cached blocks,
prefill touches only the suffix, outputs stay exactly equal."""
import torch

from modal_examples_amd.models.llama.engine import BLOCK, LlamaEngine
from modal_examples_amd.models.llama.model import LlamaConfig


def _mk(pc=False, blocks=256):
    return LlamaEngine(LlamaConfig.small(), device="cpu",
                       dtype=torch.bfloat16, use_graph=False, eos_id=-1,
                       seed=0, prefix_cache=pc, kv_blocks=blocks)


def _run_one(eng, prompt, n=5):
    rid = eng.add_request(prompt, max_new_tokens=n, temperature=0.0)
    while eng.has_work:
        eng.step()
    return eng.finished[rid].out_tokens


def test_prefix_hit_reuses_blocks_and_matches():
    g = torch.Generator().manual_seed(6)
    system = torch.randint(0, 1024, (3 * BLOCK,), generator=g).tolist()
    q1 = system + [7, 8, 9]
    q2 = system + [11, 12, 13, 14]

    plain = _mk(False)
    want1, want2 = _run_one(plain, q1), _run_one(plain, q2)

    eng = _mk(True)
    assert _run_one(eng, q1) == want1
    assert eng.prefix_hit_tokens == 0  # first request is the cold fill
    assert _run_one(eng, q2) == want2
    assert eng.prefix_hit_tokens == 3 * BLOCK  # whole system prompt reused


def test_prefix_cache_exact_multiple_of_block():
    """L % BLOCK == 0: the last block is recomputed (logits need >=1 token),
    earlier blocks still shared; output identical."""
    g = torch.Generator().manual_seed(8)
    p = torch.randint(0, 1024, (2 * BLOCK,), generator=g).tolist()
    want = _run_one(_mk(False), p)
    eng = _mk(True)
    assert _run_one(eng, p) == want
    assert _run_one(eng, p) == want
    assert eng.prefix_hit_tokens == BLOCK  # blocks-1 shared on the rerun


def test_shared_blocks_are_actually_shared():
    g = torch.Generator().manual_seed(9)
    system = torch.randint(0, 1024, (2 * BLOCK,), generator=g).tolist()
    eng = _mk(True)
    r1 = eng.add_request(system + [1], max_new_tokens=30, temperature=0.0)
    eng.step()  # r1 prefills and registers its blocks
    r2 = eng.add_request(system + [2], max_new_tokens=30, temperature=0.0)
    eng.step()
    a = eng.finished.get(r1) or next(r for r in eng.running if r.req_id == r1)
    b = eng.finished.get(r2) or next(r for r in eng.running if r.req_id == r2)
    assert a.blocks[:2] == b.blocks[:2]  # same physical blocks
    assert a.blocks[2:] != b.blocks[2:]
    while eng.has_work:
        eng.step()


def test_eviction_recycles_cached_blocks_under_pressure():
    """With a tiny pool, unreferenced cached blocks are evicted so new
    requests still run (and produce the right tokens)."""
    g = torch.Generator().manual_seed(10)
    eng = _mk(True, blocks=16)
    plain = _mk(False, blocks=16)
    for i in range(6):
        p = torch.randint(0, 1024, (2 * BLOCK + i,), generator=g).tolist()
        g2 = torch.Generator().manual_seed(10)
        torch.randint(0, 1024, (2 * BLOCK + i,), generator=g2)
        want = _run_one(plain, p, n=3)
        assert _run_one(eng, p, n=3) == want, i


def test_all_engine_features_compose():
    """prefix_cache + chunked_prefill + spec_tokens together still produce
    exactly the plain engine's greedy output."""
    g = torch.Generator().manual_seed(12)
    system = torch.randint(0, 1024, (2 * BLOCK + 5,), generator=g).tolist()
    prompts = [system + [3, 4, 5, 3, 4, 5], system + [9], system]
    plain = _mk(False)
    want = []
    for p in prompts:
        want.append(_run_one(plain, p, n=6))
    eng = LlamaEngine(LlamaConfig.small(), device="cpu",
                      dtype=torch.bfloat16, use_graph=False, eos_id=-1,
                      seed=0, prefix_cache=True, chunked_prefill=8,
                      spec_tokens=3, kv_blocks=256)
    got = []
    for p in prompts:
        got.append(_run_one(eng, p, n=6))
    assert got == want
    assert eng.prefix_hit_tokens > 0


def test_stress_pressure_mixed_features_all_requests_finish():
    """Allocator soak: tiny pool, all features on, preemption + eviction +
    sharing interleave — every request must finish with the right length
    and the pool must be fully recovered at the end."""
    g = torch.Generator().manual_seed(13)
    eng = LlamaEngine(LlamaConfig.small(), device="cpu",
                      dtype=torch.bfloat16, use_graph=False, eos_id=-1,
                      seed=0, prefix_cache=True, chunked_prefill=8,
                      spec_tokens=2, kv_blocks=24, max_batch=4)
    base = torch.randint(0, 1024, (BLOCK,), generator=g).tolist()
    want_len = {}
    for i in range(10):
        tail = torch.randint(0, 1024, (1 + i % 5,), generator=g).tolist()
        n = 3 + i % 4
        rid = eng.add_request(base + tail, max_new_tokens=n,
                              temperature=0.0 if i % 3 else 0.7)
        want_len[rid] = n
    steps = 0
    while eng.has_work:
        eng.step()
        steps += 1
        assert steps < 2000, "scheduler livelock"
    assert len(eng.finished) == 10
    for rid, n in want_len.items():
        r = eng.finished[rid]
        assert r.error is None and len(r.out_tokens) == n, (rid, r.error)
    # every block is either free or held by the prefix cache — none leaked
    cached = len(eng._pc_hash)
    assert len(eng.free_blocks) + cached == eng.num_blocks - 1  # -1 pad
    assert set(eng.free_blocks).isdisjoint(eng._pc_hash.keys())


def test_multi_turn_conversation_reuse():
    """Turn 2's prompt replays turn 1 (prompt + generated answer): the
    conversation's blocks are cached at retire, so turn 2 prefills only the
    new user text — and produces exactly the cold engine's tokens."""
    g = torch.Generator().manual_seed(14)
    turn1 = torch.randint(0, 1024, (2 * BLOCK,), generator=g).tolist()

    def conversation(eng):
        a1 = _run_one(eng, turn1, n=BLOCK)  # answer spans a full block
        turn2 = turn1 + a1 + [101, 102, 103]
        a2 = _run_one(eng, turn2, n=4)
        return a1, a2

    want = conversation(_mk(False))
    eng = _mk(True)
    got = conversation(eng)
    assert got == want
    # turn 1's cache holds prompt(32) + 15 generated tokens (the final
    # emitted token's KV is never appended), so exactly the 2 full prompt
    # blocks are reusable — and both hit
    assert eng.prefix_hit_tokens == 2 * BLOCK, eng.prefix_hit_tokens


def test_differential_fuzz_exact_features_vs_plain():
    """Randomized differential soak: prefix caching reuses bit-identical
    KV and speculation verifies with the decode kernel itself, so ANY
    greedy request mix through spec+prefix must match plain exactly.
    (Chunked prefill is excluded here by design: its chunks sum attention
    in a different bf16 order than the monolithic pass — same property as
    vLLM's chunked prefill — so argmax ties may resolve differently; its
    own tests pin exactness on fixed seeds.)"""
    for seed in range(5):
        g = torch.Generator().manual_seed(100 + seed)
        prompts = []
        base = torch.randint(0, 1024, (int(torch.randint(4, 40, (1,),
                             generator=g)),), generator=g).tolist()
        for i in range(4):
            if int(torch.randint(0, 2, (1,), generator=g)) and prompts:
                p = list(prompts[-1][: len(prompts[-1]) // 2]) + \
                    torch.randint(0, 1024, (3,), generator=g).tolist()
            else:
                p = base + torch.randint(
                    0, 1024, (int(torch.randint(1, 20, (1,), generator=g)),),
                    generator=g).tolist()
            prompts.append(p)
        plain = _mk(False)
        rich = LlamaEngine(LlamaConfig.small(), device="cpu",
                           dtype=torch.bfloat16, use_graph=False, eos_id=-1,
                           seed=0, prefix_cache=True,
                           spec_tokens=3, kv_blocks=64, max_batch=3)
        for p in prompts:
            assert _run_one(rich, p, n=5) == _run_one(plain, p, n=5), seed


def test_mixed_admission_ragged_group_plus_prefix_hits():
    """One step admits BOTH a prefix-hit request (chunk lane, suffix only)
    and fresh ragged requests (group lane) — outputs match plain engine."""
    g = torch.Generator().manual_seed(21)
    system = torch.randint(0, 1024, (2 * BLOCK,), generator=g).tolist()
    fresh_a = torch.randint(0, 1024, (7,), generator=g).tolist()
    fresh_b = torch.randint(0, 1024, (12,), generator=g).tolist()

    def run(pc):
        eng = _mk(pc)
        # warm the cache with the system prompt
        r0 = eng.add_request(system + [1], max_new_tokens=3, temperature=0.0)
        while eng.has_work:
            eng.step()
        # same step: one hit + two fresh ragged
        r1 = eng.add_request(system + [2, 3], max_new_tokens=4,
                             temperature=0.0)
        r2 = eng.add_request(fresh_a, max_new_tokens=4, temperature=0.0)
        r3 = eng.add_request(fresh_b, max_new_tokens=4, temperature=0.0)
        while eng.has_work:
            eng.step()
        return [eng.finished[i].out_tokens for i in (r0, r1, r2, r3)]

    assert run(True) == run(False)
