"""Draft-model speculative decoding (engine/draft.py): CPU end-to-end.

Exactness property (same as the ngram speculator): whatever the draft
model proposes, greedy outputs equal the plain engine's. With the draft
sharing the target's weights every greedy draft is accepted, so each step
emits k+1 tokens and acceptance is 100%.
"""

import torch

from arks_amd.config import EngineConfig
from arks_amd.engine import LLMEngine, SamplingParams


def mk(tmp_path=None, speculative=None, draft_model=None, **kw):
    cfg = EngineConfig(
        preset=kw.pop("preset", "tiny"),
        model_path=kw.pop("model_path", None),
        device="cpu",
        kv_cache_blocks=256,
        max_model_len=512,
        speculative=speculative,
        draft_model=draft_model,
        num_speculative_tokens=kw.pop("k", 4),
        **kw,
    )
    return LLMEngine(cfg)


PROMPTS = [[1, 5, 9, 20, 31, 7], [3, 3, 7, 90, 4], [17] * 12]


def test_draft_same_weights_full_acceptance(tmp_path):
    """Draft == target weights: every greedy draft agrees, acceptance is
    100%, outputs identical to plain decode."""
    from arks_amd.config import PRESET_CONFIGS
    from arks_amd.loader.safetensors_loader import save_random_checkpoint

    ckpt = str(tmp_path / "m")
    save_random_checkpoint(PRESET_CONFIGS["tiny"], ckpt, seed=5)

    sp = SamplingParams(max_tokens=16, ignore_eos=True)
    torch.manual_seed(0)
    plain = mk(model_path=ckpt, preset=None).generate(PROMPTS, sp)
    torch.manual_seed(0)
    e = mk(model_path=ckpt, preset=None, speculative="draft",
           draft_model=ckpt)
    spec = e.generate(PROMPTS, sp)
    assert spec == plain
    assert e.spec_drafted_tokens > 0
    # same weights -> disagreements only at bf16 near-ties (the draft runs
    # decode-shaped forwards, the verify extend-shaped ones — different
    # reduction orders; same caveat as spec.py's GPU note), so acceptance
    # is high but not bitwise 100%
    assert e.spec_accepted_tokens >= 0.4 * e.spec_drafted_tokens


def test_draft_different_weights_still_exact():
    """A mismatched (random) draft model changes NOTHING about greedy
    outputs — only the acceptance rate."""
    sp = SamplingParams(max_tokens=14, ignore_eos=True)
    torch.manual_seed(0)
    plain = mk().generate(PROMPTS, sp)
    torch.manual_seed(0)
    e = mk(speculative="draft", draft_model="preset:tiny")
    spec = e.generate(PROMPTS, sp)
    assert spec == plain
    assert e.spec_drafted_tokens > 0
    assert 0 <= e.spec_accepted_tokens <= e.spec_drafted_tokens


def test_draft_sampled_requests_unbiased_path():
    """Temperature requests go through exact rejection sampling with the
    draft's proposals; mixed greedy+sampled batches run in one verify."""
    sp_greedy = SamplingParams(max_tokens=10, ignore_eos=True)
    sp_temp = SamplingParams(max_tokens=10, temperature=0.8, seed=None,
                             ignore_eos=True)
    torch.manual_seed(0)
    e = mk(speculative="draft", draft_model="preset:tiny")
    s1 = e.add_request(PROMPTS[0], sp_greedy)
    s2 = e.add_request(PROMPTS[1], sp_temp)
    while e.has_work():
        e.step()
    assert len(s1.output_token_ids) == 10
    assert len(s2.output_token_ids) == 10
    # greedy row must equal the plain engine's greedy output
    torch.manual_seed(0)
    plain = mk().generate([PROMPTS[0]], sp_greedy)
    assert s1.output_token_ids == plain[0]


def test_draft_vocab_mismatch_rejected():
    import pytest

    with pytest.raises(Exception, match="vocab"):
        mk(speculative="draft", draft_model="preset:tiny-gpu")  # vocab 2048


def test_draft_survives_abort_and_requeue():
    """Aborting mid-generation and adding new work keeps the draft KV
    mirror consistent (release resets _draft_len)."""
    torch.manual_seed(0)
    e = mk(speculative="draft", draft_model="preset:tiny")
    sp = SamplingParams(max_tokens=64, ignore_eos=True)
    s1 = e.add_request(PROMPTS[0], sp)
    for _ in range(3):
        e.step()
    assert e.abort_request(s1.request_id)
    torch.manual_seed(0)
    out = e.generate([PROMPTS[2]], SamplingParams(max_tokens=12,
                                                  ignore_eos=True))
    assert len(out[0]) == 12
