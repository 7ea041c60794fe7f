"""Prompt-lookup (n-gram) speculative decoding.

Drafts come from the sequence's own token history: if the trailing n-gram
appeared earlier, the tokens that followed it are proposed and verified in
ONE extend-attention forward — the model scores every draft position, the
longest agreeing prefix is accepted, and the first disagreeing position
contributes the model's own token ("bonus"), so a step emits 1..k+1 tokens
with exactly the greedy output the non-speculative engine produces.

The reference's runtime images (vLLM/SGLang) ship the same speculator as
`--speculative-model [ngram]` / prompt-lookup; here it is first-party.
Opt-in via EngineConfig.speculative="ngram"; only greedy requests without
penalties/logprobs are speculated (others decode normally in-batch), which
keeps acceptance exact rather than distribution-approximate.

Exactness caveat (same property as vLLM's greedy verify): "exact" means
exact w.r.t. the verify forward's own logits. On GPU the verify runs the
extend kernel while plain decode runs the graphed decode kernel — two bf16
reduction orders — so a near-tie argmax can resolve differently between
the speculative and plain engines; both are valid greedy outputs of the
model. The CPU fp32 path is bitwise single-path and pins exact equality in
tests/test_spec.py.

Scheduling note: speculation applies to decode-only steps. When mixed
batching folds prefill chunks into a step, that step decodes normally —
under sustained prefill arrival the engine alternates between the two,
which is the right trade (prefill throughput dominates those phases).
"""

from __future__ import annotations

from .sequence import Sequence

# trailing-gram lengths tried in order (longer = higher precision)
NGRAM_SIZES = (3, 2)


def propose_ngram(tokens: list[int], k: int) -> list[int]:
    """Up to k draft tokens following the most recent earlier occurrence of
    the trailing n-gram. Empty list when no history match."""
    T = len(tokens)
    for n in NGRAM_SIZES:
        if T <= n:
            continue
        gram = tokens[T - n:]
        # scan backwards over earlier positions (most recent match wins)
        for i in range(T - n - 1, -1, -1):
            if tokens[i:i + n] == gram:
                nxt = tokens[i + n:i + n + k]
                if nxt:
                    return list(nxt)
                break
    return []


def eligible(seq: Sequence) -> bool:
    """Speculate only where greedy acceptance is exact."""
    sp = seq.sampling
    return (
        sp.temperature == 0.0
        and not sp.has_penalties
        and sp.logprobs is None
        and not sp.logit_bias
        and sp.min_tokens == 0
    )
