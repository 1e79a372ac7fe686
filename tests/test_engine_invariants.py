"""Engine state-machine invariants: after EVERY step, block accounting,
slot bindings and request progress must be coherent (catches mid-flight
allocator corruption that end-state checks miss)."""
import torch

from modal_examples_amd.models.llama.engine import BLOCK, LlamaEngine
from modal_examples_amd.models.llama.model import LlamaConfig


def _invariants(eng):
    # every block is in exactly one place: free pool, a live request, or
    # the prefix cache (block 0 = pad, owned by nobody)
    held = [b for r in eng.running + eng.prefilling + eng.waiting
            for b in r.blocks]
    cached_only = [b for b in eng._pc_hash
                   if b not in held and b not in eng.free_blocks]
    assert len(set(eng.free_blocks)) == len(eng.free_blocks), "free dupes"
    assert set(eng.free_blocks).isdisjoint(held), "free+held overlap"
    # shared prefix blocks may be held by several requests; every OTHER
    # block is held at most once
    from collections import Counter

    for b, n in Counter(held).items():
        if n > 1:
            assert b in eng._pc_hash, f"block {b} multiply held, not cached"
    total = len(set(eng.free_blocks) | set(held) | set(cached_only))
    assert total <= eng.num_blocks - 1
    # slot table matches running+prefilling
    for r in eng.running + eng.prefilling:
        assert r.slot >= 0 and eng._slots[r.slot] is r
        assert 0 <= len(r.out_tokens) <= r.max_new_tokens
        assert r.pos <= len(r.blocks) * BLOCK
    for i, s in enumerate(eng._slots):
        if s is not None:
            assert s.slot == i


def test_invariants_hold_through_random_schedule():
    g = torch.Generator().manual_seed(77)
    eng = LlamaEngine(LlamaConfig.small(), device="cpu",
                      dtype=torch.bfloat16, use_graph=False, eos_id=-1,
                      seed=0, prefix_cache=True, chunked_prefill=8,
                      spec_tokens=2, kv_blocks=32, max_batch=3)
    base = torch.randint(0, 1024, (BLOCK,), generator=g).tolist()
    pending = 12
    steps = 0
    while pending > 0 or eng.has_work:
        if pending > 0 and int(torch.randint(0, 2, (1,), generator=g)):
            tail = torch.randint(0, 1024,
                                 (1 + int(torch.randint(0, 18, (1,),
                                                        generator=g)),),
                                 generator=g).tolist()
            eng.add_request(base + tail,
                            max_new_tokens=2 + int(torch.randint(0, 5, (1,),
                                                   generator=g)),
                            temperature=0.0)
            pending -= 1
        eng.step()
        _invariants(eng)
        steps += 1
        assert steps < 3000
    assert len(eng.finished) == 12
    assert all(r.error is None for r in eng.finished.values())


def test_invariants_hold_under_preemption_pressure():
    """KV pool sized to force preemptions: invariants hold per step and
    every request still completes with its full token budget."""
    g = torch.Generator().manual_seed(99)
    eng = LlamaEngine(LlamaConfig.small(), device="cpu",
                      dtype=torch.bfloat16, use_graph=False, eos_id=-1,
                      seed=0, prefix_cache=True, kv_blocks=14, max_batch=4)
    for i in range(6):
        p = torch.randint(0, 1024, (10 + 3 * i,), generator=g).tolist()
        eng.add_request(p, max_new_tokens=6, temperature=0.0)
    steps = 0
    while eng.has_work:
        eng.step()
        _invariants(eng)
        steps += 1
        assert steps < 5000
    assert len(eng.finished) == 6
    ok = [r for r in eng.finished.values() if r.error is None]
    assert all(len(r.out_tokens) == 6 for r in ok)
    assert len(ok) >= 5  # at most the largest prompt may hard-fail the pool
