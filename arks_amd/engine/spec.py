"""Prompt-lookup (n-gram) speculative decoding.

Drafts come from the sequence's own token history: if the trailing n-gram
appeared earlier, the tokens that followed it are proposed and verified in
ONE extend-attention forward — the model scores every draft position, the
longest agreeing prefix is accepted, and the first disagreeing position
contributes the model's own token ("bonus"), so a step emits 1..k+1 tokens
with exactly the greedy output the non-speculative engine produces.

The reference's runtime images (vLLM/SGLang) ship the same speculator as
`--speculative-model [ngram]` / prompt-lookup; here it is first-party.
Opt-in via EngineConfig.speculative="ngram". Greedy requests accept by
argmax agreement; plain temperature requests accept by exact rejection
sampling (reject_sample_token below); everything else (penalties, bias,
logprobs, seeds) decodes normally in the same batch — acceptance is always
exact, never distribution-approximate.

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


def propose_ngram_cached(seq: Sequence, k: int) -> list[int]:
    """Same result as propose_ngram, O(1) amortized per decode step: each
    sequence carries {n-gram -> latest end position} maps, extended only
    for tokens appended since the last call (a full backwards scan per step
    costs O(T) Python time per sequence — milliseconds at serving batch
    sizes, which would eat the speculative win)."""
    st = getattr(seq, "_spec_ngrams", None)
    if st is None:
        st = seq._spec_ngrams = {"maps": {n: {} for n in NGRAM_SIZES}, "done": 0}
    toks = seq.all_token_ids
    T = len(toks)
    for n in NGRAM_SIZES:
        m = st["maps"][n]
        # register grams ending at p (follower = toks[p]); the trailing
        # gram (ending at T) has no follower yet and must not self-match
        for p in range(max(st["done"], n), T):
            m[tuple(toks[p - n:p])] = p
    st["done"] = T
    for n in NGRAM_SIZES:
        if T <= n:
            continue
        p = st["maps"][n].get(tuple(toks[T - n:]))
        if p is not None:
            nxt = toks[p:p + k]
            if nxt:
                return list(nxt)
    return []


def reject_sample_token(probs, draft: int, u_accept: float,
                        u_pick: float) -> tuple[bool, int]:
    """One speculative-sampling step with a DETERMINISTIC proposal (the
    n-gram draft): accept `draft` with probability probs[draft]; otherwise
    sample from the residual (probs with the draft zeroed, renormalized)
    by inverse CDF. Marginal over (u_accept, u_pick) is exactly `probs` —
    the standard spec-sampling identity with q = delta(draft).
    Returns (accepted, token)."""
    import torch

    pd = float(probs[draft])
    if u_accept < pd:
        return True, draft
    residual = probs.clone()
    residual[draft] = 0
    z = float(residual.sum())
    if z <= 0.0:  # degenerate: draft carries all mass
        return True, draft
    cdf = torch.cumsum(residual / z, 0)
    tok = int(torch.searchsorted(
        cdf, torch.tensor(u_pick, dtype=cdf.dtype, device=cdf.device)
    ).clamp(max=cdf.numel() - 1))
    return False, tok


def eligible(seq: Sequence) -> bool:
    """Greedy speculation: acceptance is plain argmax agreement."""
    sp = seq.sampling
    return (
        sp.temperature == 0.0
        and not sp.has_penalties
        and sp.logprobs is None
        and not sp.logit_bias
        and sp.min_tokens == 0
    )


def eligible_sampled(seq: Sequence) -> bool:
    """Sampled speculation via rejection sampling — distribution-exact for
    temperature sampling with top-p/top-k/min-p filters (the filters are
    applied to the verify logits before acceptance). Excluded: penalties
    (their logits depend on the evolving token set inside the draft),
    logprobs (reported values would mix accept/resample paths), logit_bias
    / min_tokens (kept on the plain sampler), per-request seeds (their
    generator's draw sequence is part of the reproducibility contract)."""
    sp = seq.sampling
    return (
        sp.temperature > 0.0
        and not sp.has_penalties
        and sp.logprobs is None
        and not sp.logit_bias
        and sp.min_tokens == 0
        and sp.seed is None
    )
