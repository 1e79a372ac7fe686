"""Chunked prefill: long prompts stream into the cache one chunk per step
(--chunked-prefill-size role, deepseek_v4.py:102 / very_large_models.py:169)
without stalling decode of the running batch; outputs are exactly the
monolithic-prefill outputs (per-row-lens attention is exactly causal)."""
import torch

from modal_examples_amd.models.llama.engine import LlamaEngine
from modal_examples_amd.models.llama.model import LlamaConfig


def _mk(chunk=0):
    return LlamaEngine(LlamaConfig.small(), device="cpu",
                       dtype=torch.bfloat16, use_graph=False, eos_id=-1,
                       seed=0, chunked_prefill=chunk)


def _run(eng, prompts, n=6):
    for p in prompts:
        eng.add_request(p, max_new_tokens=n, temperature=0.0)
    while eng.has_work:
        eng.step()
    return [eng.finished[i].out_tokens for i in sorted(eng.finished)]


def test_chunked_prefill_matches_monolithic():
    g = torch.Generator().manual_seed(2)
    long_prompt = torch.randint(0, 1024, (37,), generator=g).tolist()
    short = torch.randint(0, 1024, (5,), generator=g).tolist()
    want = _run(_mk(0), [long_prompt, short])
    got = _run(_mk(8), [long_prompt, short])
    assert got == want


def test_chunked_prefill_interleaves_with_decode():
    """While a 40-token prompt prefills in chunks of 8, an already-running
    request keeps emitting a token per step."""
    eng = _mk(8)
    eng.add_request([1, 2, 3], max_new_tokens=10, temperature=0.0)
    eng.step()  # short request prefilled + first token, now decoding
    short_req = eng.finished.get(1) or eng.running[0]
    g = torch.Generator().manual_seed(3)
    eng.add_request(torch.randint(0, 1024, (40,), generator=g).tolist(),
                    max_new_tokens=3, temperature=0.0)
    before = len(short_req.out_tokens)
    progressed = 0
    for _ in range(4):  # 40/8 = 5 chunks; run 4 steps mid-prefill
        eng.step()
        if eng.prefilling:
            progressed += 1
            assert len(short_req.out_tokens) > before, \
                "decode stalled during chunked prefill"
            before = len(short_req.out_tokens)
    assert progressed >= 2  # the long prompt really was mid-prefill
    while eng.has_work:
        eng.step()
    assert len(eng.finished) == 2
    assert len(eng.finished[2].out_tokens) == 3


def test_chunk_boundary_edges():
    """Prompt lengths straddling chunk/block boundaries stay exact."""
    g = torch.Generator().manual_seed(4)
    for L in (9, 16, 17, 24, 33):
        prompt = torch.randint(0, 1024, (L,), generator=g).tolist()
        want = _run(_mk(0), [prompt], n=4)
        got = _run(_mk(8), [prompt], n=4)
        assert got == want, L
