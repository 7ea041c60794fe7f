"""Prompt-lookup (n-gram) speculative decoding: the speculative engine must
emit exactly the non-speculative engine's tokens (greedy acceptance is
exact), accept drafts on repetitive text, and leave non-greedy requests on
the normal sampler path."""

import torch

from arks_amd.config import EngineConfig
from arks_amd.engine import LLMEngine, SamplingParams
from arks_amd.engine.spec import eligible, propose_ngram


def _cfg(spec=None, **kw):
    return EngineConfig(
        preset="tiny", device="cpu", kv_cache_blocks=256, max_model_len=512,
        seed=7, speculative=spec, **kw,
    )


def test_propose_ngram():
    # trailing [7, 8] seen earlier -> propose what followed it
    assert propose_ngram([1, 7, 8, 9, 4, 5, 7, 8], k=3) == [9, 4, 5]
    # trigram match preferred over bigram
    toks = [1, 2, 3, 99, 5, 6, 2, 3, 42, 1, 2, 3]
    assert propose_ngram(toks, k=2) == [99, 5]
    # no history match
    assert propose_ngram([1, 2, 3, 4, 5], k=4) == []
    # match at the very start
    assert propose_ngram([5, 6, 7, 1, 5, 6, 7], k=8) == [1, 5, 6, 7]
    # k caps the draft
    assert propose_ngram([1, 7, 8, 9, 4, 5, 7, 8], k=1) == [9]


def test_eligibility():
    from arks_amd.engine.sequence import Sequence

    assert eligible(Sequence([1], SamplingParams()))
    assert not eligible(Sequence([1], SamplingParams(temperature=0.5)))
    assert not eligible(Sequence([1], SamplingParams(presence_penalty=1.0)))
    assert not eligible(Sequence([1], SamplingParams(logprobs=0)))


def test_spec_matches_plain_greedy():
    prompts = [
        [1, 2, 3, 4] * 8,          # repetitive -> drafts accept
        [9, 31, 7, 2, 55, 14, 3],  # no repetition -> mostly plain decode
        [5] * 3,
    ]
    sp = SamplingParams(max_tokens=24, ignore_eos=True)
    torch.manual_seed(0)
    ref = LLMEngine(_cfg()).generate(prompts, sp)
    torch.manual_seed(0)
    eng = LLMEngine(_cfg(spec="ngram"))
    out = eng.generate(prompts, sp)
    assert out == ref
    # drafts were proposed and some accepted (the model output itself
    # repeats on the periodic prompt)
    assert eng.spec_drafted_tokens > 0
    assert 0 <= eng.spec_accepted_tokens <= eng.spec_drafted_tokens


def test_spec_mixed_batch_with_sampled_request():
    """A seeded sampled request rides the normal sampler inside a spec
    step; a greedy request speculates — both match the plain engine."""
    prompts = [[1, 2, 3, 4] * 6, [4, 4, 2, 9] * 4]
    sps = [
        SamplingParams(max_tokens=16, ignore_eos=True),
        SamplingParams(max_tokens=16, ignore_eos=True, temperature=0.8,
                       seed=123),
    ]

    def run(spec):
        torch.manual_seed(0)
        e = LLMEngine(_cfg(spec=spec))
        seqs = [e.add_request(p, s) for p, s in zip(prompts, sps)]
        while e.has_work():
            e.step()
        return [s.output_token_ids for s in seqs]

    assert run("ngram") == run(None)


def test_spec_max_tokens_truncation_mid_step():
    """Finishing inside a multi-token emission must stop exactly at
    max_tokens, like the plain engine."""
    prompts = [[3, 1, 3, 1, 3, 1, 3, 1]]
    sp = SamplingParams(max_tokens=5, ignore_eos=True)
    torch.manual_seed(0)
    ref = LLMEngine(_cfg()).generate(prompts, sp)
    torch.manual_seed(0)
    out = LLMEngine(_cfg(spec="ngram")).generate(prompts, sp)
    assert out == ref and len(out[0]) == 5


def test_spec_with_prefix_cache_and_mixed_batching_off():
    prompts = [[1, 2] * 10]
    sp = SamplingParams(max_tokens=12, ignore_eos=True)
    torch.manual_seed(0)
    ref = LLMEngine(_cfg(enable_prefix_caching=False,
                         enable_mixed_batching=False)).generate(prompts, sp)
    torch.manual_seed(0)
    out = LLMEngine(_cfg(spec="ngram", enable_prefix_caching=False,
                         enable_mixed_batching=False)).generate(prompts, sp)
    assert out == ref


def test_spec_fuzz_equivalence():
    """Random prompt soups from a tiny alphabet (heavy repetition): spec
    engine output must equal the plain engine's on the CPU fp32 path."""
    import random

    rng = random.Random(11)
    for trial in range(4):
        prompts = [
            [rng.randrange(1, 6) for _ in range(rng.randrange(3, 40))]
            for _ in range(rng.randrange(1, 6))
        ]
        sp = SamplingParams(max_tokens=rng.randrange(3, 20), ignore_eos=True)
        torch.manual_seed(0)
        ref = LLMEngine(_cfg()).generate(prompts, sp)
        torch.manual_seed(0)
        eng = LLMEngine(_cfg(spec="ngram"))
        assert eng.generate(prompts, sp) == ref, f"trial {trial}"


def test_spec_server_stream_e2e():
    """HTTP SSE stream with speculation enabled: chunk tokens must add up
    and usage must be exact even when a step emits several tokens."""
    import asyncio
    import json

    import httpx

    from arks_amd.server.api import create_app
    from arks_amd.server.async_engine import AsyncEngine
    from arks_amd.server.tokenizer import ByteTokenizer

    cfg = _cfg(spec="ngram")
    engine = AsyncEngine(cfg, model_name="tiny")
    mc = cfg.model_config()
    tok = ByteTokenizer(mc.vocab_size, mc.eos_token_id)
    app = create_app(engine, "tiny", tok)

    async def go():
        async with app.router.lifespan_context(app):
            rt = httpx.ASGITransport(app=app)
            async with httpx.AsyncClient(transport=rt, base_url="http://t",
                                         timeout=60) as client:
                r = await client.post("/v1/completions", json={
                    "model": "tiny",
                    "prompt": "abab abab abab",
                    "max_tokens": 12,
                    "temperature": 0,
                    "ignore_eos": True,
                    "stream": True,
                    "stream_options": {"include_usage": True},
                })
                assert r.status_code == 200, r.text
                chunks = [json.loads(line[6:]) for line in r.text.splitlines()
                          if line.startswith("data: ") and line != "data: [DONE]"]
                usage = chunks[-1]["usage"]
                assert usage["completion_tokens"] == 12

    asyncio.new_event_loop().run_until_complete(go())


def test_propose_ngram_cached_equals_pure():
    """The incremental n-gram map must reproduce the pure backwards-scan
    proposer as a sequence grows token by token."""
    import random

    from arks_amd.engine.sequence import Sequence
    from arks_amd.engine.spec import propose_ngram, propose_ngram_cached

    rng = random.Random(5)
    for trial in range(20):
        start = [rng.randrange(1, 5) for _ in range(rng.randrange(1, 10))]
        seq = Sequence(list(start))
        for _ in range(40):
            toks = seq.all_token_ids
            for k in (1, 4):
                assert propose_ngram_cached(seq, k) == propose_ngram(toks, k), (
                    trial, toks, k)
            seq.append_token(rng.randrange(1, 5))


def test_reject_sample_token_math():
    """Deterministic-proposal speculative sampling: acceptance boundary at
    p(draft); residual removes the draft and renormalizes."""
    from arks_amd.engine.spec import reject_sample_token

    p = torch.tensor([0.5, 0.3, 0.2])
    assert reject_sample_token(p, 0, 0.49, 0.0) == (True, 0)
    # rejected: residual over [1, 2] is [0.6, 0.4]
    assert reject_sample_token(p, 0, 0.51, 0.59) == (False, 1)
    assert reject_sample_token(p, 0, 0.51, 0.61) == (False, 2)
    # degenerate: draft carries all mass -> accept regardless
    q = torch.tensor([1.0, 0.0, 0.0])
    assert reject_sample_token(q, 0, 0.999, 0.5) == (True, 0)


def test_reject_sample_marginal_matches_target():
    """Empirical marginal of accept-or-resample equals the target
    distribution (the spec-sampling identity), generous tolerance."""
    from arks_amd.engine.spec import reject_sample_token

    torch.manual_seed(0)
    p = torch.tensor([0.6, 0.3, 0.1])
    n = 20000
    u = torch.rand(2, n).tolist()
    counts = [0, 0, 0]
    for i in range(n):
        _, tok = reject_sample_token(p, 1, u[0][i], u[1][i])
        counts[tok] += 1
    freq = [c / n for c in counts]
    assert all(abs(f - t) < 0.02 for f, t in zip(freq, p.tolist())), freq


def test_spec_sampled_requests_deterministic_and_drafted():
    """Unseeded temperature sampling under speculation: same global seed
    -> same outputs, and drafts actually flow on repetitive text."""
    prompts = [[1, 2, 3, 4] * 8, [2, 9] * 10]
    sp = SamplingParams(max_tokens=16, ignore_eos=True, temperature=0.8,
                        top_p=0.9)

    def run():
        torch.manual_seed(3)
        e = LLMEngine(_cfg(spec="ngram"))
        out = e.generate([list(p) for p in prompts], sp)
        return out, e.spec_drafted_tokens, e.spec_accepted_tokens

    out1, drafted1, accepted1 = run()
    out2, drafted2, _ = run()
    assert out1 == out2 and drafted1 == drafted2
    assert 0 <= accepted1 <= drafted1
    assert all(len(o) == 16 for o in out1)
    # sampled outputs rarely repeat, so drive the rejection verify path
    # directly: a decode batch with an injected draft emits 1..len(d)+1
    # tokens deterministically under a fixed global seed
    torch.manual_seed(4)
    e = LLMEngine(_cfg(spec="ngram"))
    seq = e.add_request([4, 9, 2, 7, 7, 1],
                        SamplingParams(max_tokens=64, ignore_eos=True,
                                       temperature=0.8, top_p=0.9))
    while e.scheduler.num_waiting:
        e.step()
    sb = e.scheduler.schedule()
    assert not sb.is_prefill and sb.seqs == [seq]
    from arks_amd.engine.kv_cache import BlockAllocator

    draft = [5, 9, 3]
    need = (BlockAllocator.blocks_needed(seq.num_tokens + len(draft),
                                         e.cfg.block_size)
            - len(seq.block_table))
    if need > 0:
        seq.block_table.extend(e.scheduler.allocator.allocate(need))
    torch.manual_seed(5)
    got1 = e.runner.execute_spec(sb, [list(draft)])[0]
    assert 1 <= len(got1) <= len(draft) + 1
    torch.manual_seed(5)
    got2 = e.runner.execute_spec(sb, [list(draft)])[0]
    assert got1 == got2


def test_spec_after_disagg_inject_matches_plain():
    """Speculation on a decode instance over injected (remotely prefilled)
    KV must emit the plain engine's greedy continuation."""
    prompt = [1, 2, 3, 4] * 8
    sp1 = SamplingParams(max_tokens=1, ignore_eos=True)
    spn = SamplingParams(max_tokens=10, ignore_eos=True)

    def run(spec):
        torch.manual_seed(0)
        a = LLMEngine(_cfg())
        torch.manual_seed(0)
        b = LLMEngine(_cfg(spec=spec))
        seq = a.add_request(list(prompt), sp1, request_id="p", hold_pages=True)
        while not seq.is_finished:
            a.step()
        _, kv = a.extract_prefilled("p")
        b.add_prefilled(list(prompt), seq.output_token_ids[0], kv, spn, "d")
        toks = [seq.output_token_ids[0]]
        while b.has_work():
            for o in b.step():
                toks.append(o.new_token_id)
        return toks

    assert run("ngram") == run(None)
