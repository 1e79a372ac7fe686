"""N-gram (prompt-lookup) speculative decoding — proposer side.

Reference role: the canonical vLLM serving example turns on ngram
speculation (06_gpu_and_ml/llm-serving/vllm_inference.py:195-202,
`{"method": "ngram", "num_speculative_tokens": 4}`): draft tokens come from
matching the current context suffix against its own history and copying what
followed — no draft model, pure lookup, strong on code/extraction/agentic
loops where output repeats context.

The verify side lives in the engine (`LlamaEngine._decode_batch_spec`): the
existing paged-decode kernel checks all k+1 positions in ONE forward by
expanding the batch — one row per draft position, sharing the request's
block table with per-row lens (row j attends positions <= pos+j, exactly
causal), so no kernel changes are needed.
"""
from __future__ import annotations

from typing import List


def ngram_propose(ctx: List[int], k: int, max_ngram: int = 3,
                  min_ngram: int = 1) -> List[int]:
    """Propose up to k draft tokens by suffix lookup.

    Finds the most recent earlier occurrence of the longest matching
    suffix n-gram (n from max_ngram down to min_ngram) and returns the
    tokens that followed it.  Empty list = no match (caller decodes
    normally).
    """
    L = len(ctx)
    for n in range(min(max_ngram, L - 1), min_ngram - 1, -1):
        tail = ctx[L - n:]
        # scan backwards; stop before the suffix occurrence itself
        for i in range(L - n - 1, -1, -1):
            if ctx[i:i + n] == tail:
                cont = ctx[i + n:i + n + k]
                if cont:
                    return list(cont)
    return []
