"""Prefill/decode disaggregation e2e on CPU: prefill app + decode app +
router app wired in-process via httpx ASGI transports. Covers the KV page
transfer (extract -> wire -> inject), first-token handoff, router policies
and SSE relay with the final usage chunk.

Mirrors the reference's ArksDisaggregatedApplication data path
(arksdisaggregatedapplication_controller.go:1630-1724) with our first-party
engine + router instead of SGLang + sglang-router.
"""

import asyncio
import json

import httpx

from arks_amd.config import EngineConfig
from arks_amd.engine import LLMEngine, SamplingParams
from arks_amd.router.app import RouterState, create_router_app
from arks_amd.server.api import create_app
from arks_amd.server.async_engine import AsyncEngine
from arks_amd.server.disagg import decode_kv, encode_kv
from arks_amd.server.tokenizer import ByteTokenizer


def _cfg(seed=5):
    return EngineConfig(
        preset="tiny", device="cpu", kv_cache_blocks=256, max_model_len=512,
        seed=seed,
    )


def _mk_apps():
    cfg = _cfg()
    mc = cfg.model_config()
    tok = ByteTokenizer(mc.vocab_size, mc.eos_token_id)
    prefill_engine = AsyncEngine(cfg, model_name="tiny", disagg_mode="prefill")
    decode_engine = AsyncEngine(_cfg(), model_name="tiny", disagg_mode="decode")
    prefill_app = create_app(prefill_engine, "tiny", tok, disagg_mode="prefill")
    decode_app = create_app(decode_engine, "tiny", tok, disagg_mode="decode")
    # decode pulls KV from the prefill app in-process
    decode_engine.http_transport = httpx.ASGITransport(app=prefill_app)
    return prefill_app, decode_app, tok


def test_kv_wire_roundtrip():
    import torch

    kv = torch.randn(2, 2, 3, 2, 16, 8, dtype=torch.bfloat16)
    shape, body = encode_kv(kv)
    back = decode_kv(shape, body)
    assert torch.equal(kv, back)
    # bare-shape headers from older peers imply bf16
    legacy = decode_kv(shape.split(":", 1)[1], body)
    assert torch.equal(kv, legacy)


def test_kv_wire_roundtrip_fp8():
    import torch

    kv = (torch.randn(2, 2, 3, 2, 16, 8) * 4).to(torch.float8_e4m3fn)
    shape, body = encode_kv(kv)
    assert shape.startswith("float8_e4m3fn:")
    back = decode_kv(shape, body)
    assert back.dtype == torch.float8_e4m3fn
    assert torch.equal(kv.view(torch.uint8), back.view(torch.uint8))


def test_disagg_engine_fp8_kv_pages():
    """fp8 KV pages end to end through extract -> wire -> inject (both
    instances on kv_cache_dtype=fp8)."""
    def cfg():
        return EngineConfig(
            preset="tiny", device="cpu", kv_cache_blocks=256,
            max_model_len=512, seed=5, kv_cache_dtype="fp8",
        )

    prompt = [7, 3, 9, 1] * 9
    sp = SamplingParams(max_tokens=6, ignore_eos=True)
    mono = LLMEngine(cfg()).generate([prompt], sp)[0]
    a, b = LLMEngine(cfg()), LLMEngine(cfg())
    seq = a.add_request(prompt, SamplingParams(max_tokens=1, ignore_eos=True),
                        request_id="r0", hold_pages=True)
    while not seq.is_finished:
        a.step()
    _, kv = a.extract_prefilled("r0")
    import torch

    assert kv.dtype == torch.float8_e4m3fn
    shape, body = encode_kv(kv)
    kv2 = decode_kv(shape, body)
    b.add_prefilled(prompt, seq.output_token_ids[0], kv2, sp, request_id="r0")
    out = [seq.output_token_ids[0]]
    while b.has_work():
        for o in b.step():
            out.append(o.new_token_id)
    assert out == mono


def test_disagg_engine_matches_monolithic():
    """Sync-engine level: extract on A, inject into B, greedy outputs equal
    a monolithic engine's."""
    prompts = [[7, 3, 9, 1] * 9, [2, 8] * 5]
    sp = SamplingParams(max_tokens=6, ignore_eos=True)
    mono = LLMEngine(_cfg()).generate(prompts, sp)

    a = LLMEngine(_cfg())
    b = LLMEngine(_cfg())
    outs = []
    for i, prompt in enumerate(prompts):
        rid = f"r{i}"
        seq = a.add_request(prompt, SamplingParams(max_tokens=1, ignore_eos=True),
                            request_id=rid, hold_pages=True)
        while not seq.is_finished:
            a.step()
        first = seq.output_token_ids[0]
        _, kv = a.extract_prefilled(rid)
        b.add_prefilled(prompt, first, kv, sp, request_id=rid)
        outs.append([first])
    while b.has_work():
        for o in b.step():
            outs[int(o.request_id[1:])].append(o.new_token_id)
    assert outs == mono


def test_disagg_http_e2e_stream():
    prefill_app, decode_app, tok = _mk_apps()
    state = RouterState([], ["http://decode"], policy="cache_aware")
    state.transport = httpx.ASGITransport(app=decode_app)
    state.set_workers(prefill_urls=["http://prefill"])
    router_app = create_router_app(state)

    # expected output from a monolithic engine with identical seed
    sp = SamplingParams(max_tokens=5, temperature=0.0, ignore_eos=True)
    prompt_text = "hello pd"
    token_ids = tok.encode(prompt_text)
    expected = LLMEngine(_cfg()).generate([token_ids], sp)[0]

    async def go():
        async with prefill_app.router.lifespan_context(prefill_app):
            async with decode_app.router.lifespan_context(decode_app):
                rt = httpx.ASGITransport(app=router_app)
                async with httpx.AsyncClient(
                    transport=rt, base_url="http://router", timeout=120
                ) as client:
                    r = await client.post("/v1/completions", json={
                        "model": "tiny",
                        "prompt": prompt_text,
                        "max_tokens": 5,
                        "temperature": 0,
                        "ignore_eos": True,
                        "stream": True,
                        "stream_options": {"include_usage": True},
                    })
                    assert r.status_code == 200, r.text
                    chunks = []
                    for line in r.text.splitlines():
                        if line.startswith("data: ") and line != "data: [DONE]":
                            chunks.append(json.loads(line[6:]))
                    # usage arrives in the final empty-choices chunk
                    # (gateway contract, reference handle_response.go:113-133)
                    usage = chunks[-1]["usage"]
                    assert usage["completion_tokens"] == 5
                    assert usage["prompt_tokens"] == len(token_ids)
                    text = "".join(
                        c["choices"][0]["text"] for c in chunks if c["choices"]
                    )
                    assert text == tok.decode(expected)
                    # router metrics counted the request
                    m = await client.get("/metrics")
                    assert "arks_router_requests_total" in m.text

    asyncio.new_event_loop().run_until_complete(go())


def test_router_no_decode_workers_503():
    state = RouterState(["http://p"], [], policy="round_robin")
    app = create_router_app(state)

    async def go():
        rt = httpx.ASGITransport(app=app)
        async with httpx.AsyncClient(transport=rt, base_url="http://r") as c:
            r = await c.post("/v1/chat/completions", json={"model": "x"})
            assert r.status_code == 503
            h = await c.get("/health")
            assert h.status_code == 503

    asyncio.new_event_loop().run_until_complete(go())


def test_router_cache_aware_affinity():
    state = RouterState(["http://p1", "http://p2", "http://p3"], ["http://d"],
                        policy="cache_aware")
    body = {"messages": [{"role": "user", "content": "same prompt"}]}
    from arks_amd.router.app import _prompt_key

    picks = {state.pick_prefill(_prompt_key(body)) for _ in range(10)}
    assert len(picks) == 1  # same prompt -> same prefill worker every time


def test_router_fails_over_dead_decode_worker():
    """A decode worker that refuses connections (service discovery hasn't
    caught up) is skipped: the request lands on a live worker; with every
    worker dead the router answers 502, not a raw 500."""
    import asyncio

    import httpx
    from fastapi import FastAPI
    from fastapi.responses import JSONResponse

    from arks_amd.router.app import RouterState, create_router_app

    live = FastAPI()

    @live.post("/v1/chat/completions")
    async def chat():
        return JSONResponse({"ok": True, "served_by": "live"})

    live_transport = httpx.ASGITransport(app=live)

    class FlakyTransport(httpx.AsyncBaseTransport):
        async def handle_async_request(self, request):
            if request.url.host == "dead":
                raise httpx.ConnectError("refused", request=request)
            return await live_transport.handle_async_request(request)

    state = RouterState([], ["http://dead", "http://live"],
                        policy="round_robin")
    state.transport = FlakyTransport()
    app = create_router_app(state)
    rt = httpx.ASGITransport(app=app)

    async def go():
        async with httpx.AsyncClient(transport=rt, base_url="http://r") as c:
            # several requests: round_robin starts on either worker, all
            # must succeed via failover
            for _ in range(4):
                r = await c.post("/v1/chat/completions", json={"messages": []})
                assert r.status_code == 200 and r.json()["served_by"] == "live"
        state.set_workers(decode_urls=["http://dead"])
        async with httpx.AsyncClient(transport=rt, base_url="http://r") as c:
            r = await c.post("/v1/chat/completions", json={"messages": []})
            assert r.status_code == 502

    asyncio.run(go())
    # inflight counters fully released after failovers
    assert all(v == 0 for v in state.inflight.values())
