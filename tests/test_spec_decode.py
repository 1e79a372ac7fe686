"""Ngram speculative decoding: proposer lookup + engine verify path.

The invariant that makes speculation safe: greedy outputs are IDENTICAL to
plain decode whatever the drafts are (verification accepts only tokens the
model itself would have produced).  The oracle test feeds perfect drafts and
checks the engine actually skips steps.
"""
import torch

from modal_examples_amd.models.llama.engine import LlamaEngine
from modal_examples_amd.models.llama.model import LlamaConfig
from modal_examples_amd.models.llama.spec import ngram_propose


def test_ngram_propose_lookup():
    #            0  1  2  3  4  5  6  7
    ctx = [5, 6, 7, 9, 5, 6, 7]
    # suffix [5,6,7] matched at position 0 -> continuation [9, 5, ...]
    assert ngram_propose(ctx, 2) == [9, 5]
    assert ngram_propose(ctx, 4) == [9, 5, 6, 7]
    # no repeated suffix anywhere -> no proposal
    assert ngram_propose([1, 2, 3, 4], 3) == []
    # most RECENT earlier occurrence wins
    ctx = [1, 2, 8, 1, 2, 9, 1, 2]
    assert ngram_propose(ctx, 1) == [9]
    assert ngram_propose([], 3) == []
    assert ngram_propose([3], 3) == []


def _mk(spec_tokens=0, seed=0):
    eng = LlamaEngine(LlamaConfig.small(), device="cpu", dtype=torch.bfloat16,
                      use_graph=False, eos_id=-1, seed=seed,
                      spec_tokens=spec_tokens)
    return eng


def _greedy(eng, prompts, n=8):
    for p in prompts:
        eng.add_request(p, max_new_tokens=n, temperature=0.0)
    steps = 0
    while eng.has_work:
        eng.step()
        steps += 1
    return [eng.finished[i].out_tokens for i in sorted(eng.finished)], steps


def test_spec_greedy_identical_to_plain():
    """Whatever the ngram drafts are, greedy output is token-identical."""
    g = torch.Generator().manual_seed(5)
    prompts = [torch.randint(0, 1024, (7,), generator=g).tolist(),
               [3, 4, 5, 3, 4, 5, 3, 4],  # repetitive: drafts will fire
               torch.randint(0, 1024, (11,), generator=g).tolist()]
    want, _ = _greedy(_mk(spec_tokens=0), prompts)
    got, _ = _greedy(_mk(spec_tokens=4), prompts)
    assert got == want


def test_spec_oracle_drafts_skip_steps():
    """Perfect drafts -> ~N/(k+1) decode steps and full acceptance."""
    prompt = [9, 8, 7, 6, 5]
    n = 12
    want, plain_steps = _greedy(_mk(0), [prompt], n=n)
    answer = want[0]

    eng = _mk(spec_tokens=3)
    eng._propose = lambda r: answer[len(r.out_tokens):
                                    len(r.out_tokens) + eng.spec_tokens]
    got, spec_steps = _greedy(eng, [prompt], n=n)
    assert got == want
    assert eng.spec_accepted >= n - n // 4 - 2, eng.spec_accepted
    # 12 tokens at k=3: 1 prefill-token + ceil(11/4) verify steps + scheduler
    assert spec_steps < plain_steps / 2, (spec_steps, plain_steps)


def test_spec_mixed_temperature_batch():
    """Sampled requests ride along (0 drafts) while greedy requests spec."""
    eng = _mk(spec_tokens=4)
    eng.add_request([3, 4, 5, 3, 4, 5, 3, 4], max_new_tokens=6,
                    temperature=0.0)
    eng.add_request([1, 2, 3, 4, 5, 6], max_new_tokens=6, temperature=0.8)
    while eng.has_work:
        eng.step()
    outs = [eng.finished[i].out_tokens for i in sorted(eng.finished)]
    assert len(outs) == 2 and all(len(o) == 6 for o in outs)
    # greedy request must match a plain engine run of the same prompt
    want, _ = _greedy(_mk(0), [[3, 4, 5, 3, 4, 5, 3, 4]], n=6)
    assert outs[0] == want[0]


def test_frequency_penalty_prevents_repeats():
    """A huge frequency penalty makes every greedy output token distinct;
    penalty=0 reproduces the plain run exactly."""
    prompt = [5, 6, 7, 8]
    base, _ = _greedy(_mk(0), [prompt], n=10)

    eng = _mk(0)
    eng.add_request(prompt, max_new_tokens=10, temperature=0.0,
                    frequency_penalty=1e9)
    while eng.has_work:
        eng.step()
    toks = eng.finished[1].out_tokens
    assert len(set(toks)) == len(toks) == 10
    assert toks[0] == base[0][0]  # first token unaffected (no output yet)

    eng0 = _mk(0)
    eng0.add_request(prompt, max_new_tokens=10, temperature=0.0,
                     presence_penalty=0.0, frequency_penalty=0.0)
    while eng0.has_work:
        eng0.step()
    assert eng0.finished[1].out_tokens == base[0]


def test_penalized_requests_skip_speculation_but_match():
    """Penalties disable drafts (they evolve within a run) — output equals
    an unspeculated penalized engine."""
    def run(spec):
        eng = _mk(spec)
        eng.add_request([3, 4, 5, 3, 4, 5], max_new_tokens=8,
                        temperature=0.0, presence_penalty=2.0)
        while eng.has_work:
            eng.step()
        return eng.finished[1].out_tokens, eng.spec_proposed

    plain, _ = run(0)
    spec, proposed = run(4)
    assert spec == plain
    assert proposed == 0  # no drafts were even offered


def test_ngram_propose_properties():
    """Property sweep: any proposal is a verbatim continuation of an
    earlier occurrence of the current suffix, length-capped at k."""
    from hypothesis import given, settings
    from hypothesis import strategies as st

    @settings(max_examples=200, deadline=None)
    @given(st.lists(st.integers(0, 7), max_size=40), st.integers(1, 6))
    def prop(ctx, k):
        out = ngram_propose(ctx, k)
        assert len(out) <= k
        if out:
            # some suffix n-gram of ctx occurs earlier, followed by `out`
            found = False
            for n in range(1, 4):
                if len(ctx) < n + 1:
                    continue
                tail = ctx[-n:]
                for i in range(len(ctx) - n):
                    if ctx[i:i + n] == tail and \
                            ctx[i + n:i + n + len(out)] == out:
                        found = True
            assert found, (ctx, k, out)

    prop()
