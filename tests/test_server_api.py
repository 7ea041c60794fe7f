"""HTTP API tests (CPU, tiny model): OpenAI contract incl. the SSE
final-usage-chunk shape the gateway depends on."""

import asyncio
import json

import httpx
import pytest

from arks_amd.config import EngineConfig
from arks_amd.server.api import create_app
from arks_amd.server.async_engine import AsyncEngine
from arks_amd.server.tokenizer import ByteTokenizer


@pytest.fixture()
def app():
    cfg = EngineConfig(
        preset="tiny", device="cpu", kv_cache_blocks=256, max_model_len=512
    )
    engine = AsyncEngine(cfg, model_name="tiny")
    tok = ByteTokenizer(cfg.model_config().vocab_size, cfg.model_config().eos_token_id)
    return create_app(engine, "tiny", tok)


def run_with_client(app, fn):
    async def go():
        async with app.router.lifespan_context(app):
            transport = httpx.ASGITransport(app=app)
            async with httpx.AsyncClient(
                transport=transport, base_url="http://t", timeout=60
            ) as client:
                await fn(client)

    asyncio.new_event_loop().run_until_complete(go())


def test_models_and_health(app):
    async def fn(client):
        r = await client.get("/health")
        assert r.status_code == 200
        r = await client.get("/v1/models")
        assert r.json()["data"][0]["id"] == "tiny"

    run_with_client(app, fn)


def test_chat_completion_non_stream(app):
    async def fn(client):
        r = await client.post(
            "/v1/chat/completions",
            json={
                "model": "tiny",
                "messages": [{"role": "user", "content": "hi"}],
                "max_tokens": 4,
                "temperature": 0,
                "ignore_eos": True,
            },
        )
        assert r.status_code == 200, r.text
        body = r.json()
        assert body["object"] == "chat.completion"
        assert body["usage"]["completion_tokens"] == 4
        assert body["usage"]["total_tokens"] == body["usage"]["prompt_tokens"] + 4
        assert body["choices"][0]["message"]["role"] == "assistant"

    run_with_client(app, fn)


def test_chat_completion_stream_final_usage_chunk(app):
    """Streaming with include_usage: final chunk has EMPTY choices + usage,
    then [DONE] — exactly what reference handle_response.go:113-133 parses."""

    async def fn(client):
        chunks = []
        async with client.stream(
            "POST",
            "/v1/chat/completions",
            json={
                "model": "tiny",
                "messages": [{"role": "user", "content": "hello"}],
                "max_tokens": 3,
                "temperature": 0,
                "stream": True,
                "stream_options": {"include_usage": True},
                "ignore_eos": True,
            },
        ) as r:
            assert r.status_code == 200
            async for line in r.aiter_lines():
                if line.startswith("data: "):
                    chunks.append(line[len("data: "):])
        assert chunks[-1] == "[DONE]"
        final = json.loads(chunks[-2])
        assert final["choices"] == []
        assert final["usage"]["completion_tokens"] == 3
        content_chunks = [json.loads(c) for c in chunks[:-2]]
        assert content_chunks[0]["choices"][0]["delta"]["role"] == "assistant"
        deltas = [
            c["choices"][0]["delta"].get("content", "") for c in content_chunks[1:]
        ]
        # incremental detokenization may merge held (incomplete-rune) pieces
        # into the next chunk, so chunk count <= tokens; text is complete
        assert 1 <= len(deltas) <= 3

    run_with_client(app, fn)


def test_wrong_model_404(app):
    async def fn(client):
        r = await client.post(
            "/v1/chat/completions",
            json={"model": "nope", "messages": [{"role": "user", "content": "x"}]},
        )
        assert r.status_code == 404
        assert "error" in r.json()

    run_with_client(app, fn)


def test_completions_and_metrics(app):
    async def fn(client):
        r = await client.post(
            "/v1/completions",
            json={
                "model": "tiny",
                "prompt": "abc",
                "max_tokens": 2,
                "temperature": 0,
                "ignore_eos": True,
            },
        )
        assert r.status_code == 200, r.text
        assert r.json()["usage"]["completion_tokens"] == 2
        m = await client.get("/metrics")
        text = m.text
        assert "vllm:generation_tokens" in text
        assert "vllm:num_requests_running" in text
        assert "vllm:time_to_first_token_seconds" in text

    run_with_client(app, fn)


def test_concurrent_requests_batched(app):
    """Multiple concurrent HTTP requests share engine steps (continuous
    batching) and each gets its own complete stream."""

    async def fn(client):
        async def one(i):
            r = await client.post(
                "/v1/completions",
                json={
                    "model": "tiny",
                    "prompt": f"request {i}",
                    "max_tokens": 5,
                    "temperature": 0,
                    "ignore_eos": True,
                },
            )
            assert r.status_code == 200
            return r.json()["usage"]["completion_tokens"]

        results = await asyncio.gather(*[one(i) for i in range(6)])
        assert results == [5] * 6

    run_with_client(app, fn)


def test_completion_stop_string(app):
    async def fn(client):
        # ByteTokenizer round-trips ASCII; with temperature 0 on random
        # weights the output is arbitrary bytes, so use a stop that is
        # guaranteed to appear: a single char from the decoded output of a
        # no-stop run.
        r0 = await client.post("/v1/completions", json={
            "model": "tiny", "prompt": "abc", "max_tokens": 8,
            "temperature": 0, "ignore_eos": True,
        })
        full = r0.json()["choices"][0]["text"]
        assert full
        stop_ch = full[len(full) // 2]
        r = await client.post("/v1/completions", json={
            "model": "tiny", "prompt": "abc", "max_tokens": 8,
            "temperature": 0, "ignore_eos": True, "stop": [stop_ch],
        })
        body = r.json()
        text = body["choices"][0]["text"]
        assert stop_ch not in text
        assert text == full.split(stop_ch)[0]
        assert body["choices"][0]["finish_reason"] == "stop"

    run_with_client(app, fn)


def test_stop_string_tracker_split_across_pieces():
    from arks_amd.server.api import StopStringTracker

    t = StopStringTracker(["END"])
    out = ""
    stopped = False
    for piece in ["hello E", "N", "D tail"]:
        emit, stopped = t.feed(piece)
        out += emit
        if stopped:
            break
    assert stopped and out == "hello "


def test_completion_n_choices(app):
    async def fn(client):
        r = await client.post("/v1/completions", json={
            "model": "tiny", "prompt": "abc", "max_tokens": 4,
            "temperature": 0.8, "n": 3, "ignore_eos": True,
        })
        body = r.json()
        assert len(body["choices"]) == 3
        assert [c["index"] for c in body["choices"]] == [0, 1, 2]
        assert body["usage"]["completion_tokens"] == 12
        # n>1 + stream is rejected
        r2 = await client.post("/v1/completions", json={
            "model": "tiny", "prompt": "abc", "max_tokens": 2, "n": 2,
            "stream": True,
        })
        assert r2.status_code == 400

    run_with_client(app, fn)


def test_prompt_too_long_rejected(app):
    async def fn(client):
        r = await client.post("/v1/completions", json={
            "model": "tiny", "prompt": "x" * 2000, "max_tokens": 2,
        })
        assert r.status_code == 400
        assert "max_model_len" in r.json()["error"]["message"]
        # server still healthy afterwards (engine loop not killed)
        r2 = await client.post("/v1/completions", json={
            "model": "tiny", "prompt": "ok", "max_tokens": 2,
            "ignore_eos": True,
        })
        assert r2.status_code == 200

    run_with_client(app, fn)


def test_incremental_detokenizer_multibyte():
    from arks_amd.server.tokenizer import ByteTokenizer, IncrementalDetokenizer

    tok = ByteTokenizer(512, 2)
    text = "héllo wörld ✓ ok"
    ids = tok.encode(text)
    detok = IncrementalDetokenizer(tok)
    out = "".join(detok.feed(i) for i in ids)
    # multibyte runes must never be emitted as U+FFFD fragments
    assert "�" not in out
    assert text.startswith(out) and len(text) - len(out) <= 3


def test_hf_tokenizer_incremental_stream(tmp_path):
    """A real BPE tokenizer (built in-process — no network) through
    load_tokenizer + IncrementalDetokenizer: streamed pieces must
    concatenate to the full decode."""
    import json as _json

    from tokenizers import Tokenizer, decoders, models, pre_tokenizers, trainers

    from arks_amd.server.tokenizer import (
        HFTokenizer,
        IncrementalDetokenizer,
        load_tokenizer,
    )

    tok = Tokenizer(models.BPE(unk_token="[UNK]"))
    tok.pre_tokenizer = pre_tokenizers.ByteLevel()
    tok.decoder = decoders.ByteLevel()
    trainer = trainers.BpeTrainer(
        vocab_size=300, special_tokens=["[UNK]", "<eos>"]
    )
    corpus = [
        "the quick brown fox jumps over the lazy dog",
        "hello world, hello tokenizer streams",
        "paged attention over xgmi links",
    ]
    tok.train_from_iterator(corpus, trainer)
    tok.save(str(tmp_path / "tokenizer.json"))
    (tmp_path / "tokenizer_config.json").write_text(_json.dumps({
        "tokenizer_class": "PreTrainedTokenizerFast", "eos_token": "<eos>",
    }))

    t = load_tokenizer(str(tmp_path), 512, 1)
    assert isinstance(t, HFTokenizer)
    text = "hello world, the quick brown fox streams"
    ids = t.encode(text)
    assert len(ids) > 3
    full = t.decode(ids)
    detok = IncrementalDetokenizer(t)
    streamed = "".join(detok.feed(i) for i in ids) + detok.flush()
    assert streamed == full


def test_non_stream_cancellation_aborts_engine(app):
    """A client disconnect (task cancellation) mid non-stream request must
    abort the engine sequence instead of decoding to completion."""

    async def fn(client):
        engine = app.state.engine
        task = asyncio.get_running_loop().create_task(client.post(
            "/v1/chat/completions",
            json={
                "model": "tiny",
                "messages": [{"role": "user", "content": "hello there"}],
                "max_tokens": 4000,
                "temperature": 0,
                "ignore_eos": True,
            },
        ))
        # let the request get admitted and produce a few tokens
        for _ in range(200):
            await asyncio.sleep(0.01)
            if engine.engine.scheduler.num_running > 0:
                break
        assert engine.engine.scheduler.num_running > 0
        task.cancel()
        try:
            await task
        except (asyncio.CancelledError, Exception):
            pass
        # the abort drains on the next engine iterations
        for _ in range(300):
            await asyncio.sleep(0.01)
            if not engine.engine.has_work():
                break
        assert not engine.engine.has_work(), "sequence kept decoding after disconnect"

    run_with_client(app, fn)


def test_completions_echo_with_logprobs(app):
    """OpenAI echo contract: prompt text precedes the completion; with
    logprobs the prompt tokens are scored (first one null)."""

    async def fn(client):
        r = await client.post("/v1/completions", json={
            "model": "tiny", "prompt": "hello", "max_tokens": 3,
            "temperature": 0, "ignore_eos": True,
            "echo": True, "logprobs": 0,
        })
        assert r.status_code == 200, r.text
        body = r.json()
        choice = body["choices"][0]
        assert choice["text"].startswith("hello")
        lp = choice["logprobs"]
        n_prompt = body["usage"]["prompt_tokens"]
        assert len(lp["tokens"]) == n_prompt + 3
        assert len(lp["token_logprobs"]) == n_prompt + 3
        assert lp["token_logprobs"][0] is None
        assert all(isinstance(v, float) and v <= 0.0
                   for v in lp["token_logprobs"][1:])
        # echo without logprobs: text only
        r2 = await client.post("/v1/completions", json={
            "model": "tiny", "prompt": "hello", "max_tokens": 3,
            "temperature": 0, "ignore_eos": True, "echo": True,
        })
        assert r2.json()["choices"][0]["text"].startswith("hello")
        assert r2.json()["choices"][0]["logprobs"] is None
        # echo + stream rejected
        r3 = await client.post("/v1/completions", json={
            "model": "tiny", "prompt": "x", "max_tokens": 1,
            "echo": True, "stream": True,
        })
        assert r3.status_code == 400

    run_with_client(app, fn)


def test_engine_loop_failure_fails_streams_and_health(app):
    """A fatal exception inside engine.step must end open streams (not hang
    them) and flip /health to 503 for readiness-based ejection."""

    async def fn(client):
        eng = app.state.engine

        def boom():
            raise RuntimeError("injected engine fault")

        eng.engine.step = boom
        r = await client.post("/v1/completions", json={
            "model": "tiny", "prompt": "x", "max_tokens": 5,
            "temperature": 0, "ignore_eos": True,
        })
        # request completes (empty output) instead of hanging
        assert r.status_code == 200
        assert r.json()["usage"]["completion_tokens"] == 0
        h = await client.get("/health")
        assert h.status_code == 503
        assert isinstance(eng.failed, RuntimeError)
        # submissions after the failure end immediately instead of hanging
        r2 = await client.post("/v1/completions", json={
            "model": "tiny", "prompt": "y", "max_tokens": 3,
            "temperature": 0, "ignore_eos": True,
        })
        assert r2.status_code == 200
        assert r2.json()["usage"]["completion_tokens"] == 0

    run_with_client(app, fn)


def test_chat_and_completion_token_logprobs(app):
    """Per-generated-token logprobs on both endpoints (OpenAI shapes:
    chat logprobs.content[], completions token_logprobs/top_logprobs)."""

    async def fn(client):
        r = await client.post("/v1/chat/completions", json={
            "model": "tiny",
            "messages": [{"role": "user", "content": "hi"}],
            "max_tokens": 3, "temperature": 0, "ignore_eos": True,
            "logprobs": True, "top_logprobs": 2,
        })
        assert r.status_code == 200, r.text
        content = r.json()["choices"][0]["logprobs"]["content"]
        assert len(content) == 3
        for entry in content:
            assert entry["logprob"] <= 0.0
            assert len(entry["top_logprobs"]) == 2
            # chosen-token logprob can't beat the best alternative
            assert entry["logprob"] <= max(
                t["logprob"] for t in entry["top_logprobs"]) + 1e-6

        r2 = await client.post("/v1/completions", json={
            "model": "tiny", "prompt": "abc", "max_tokens": 4,
            "temperature": 0, "ignore_eos": True, "logprobs": 1,
        })
        lp = r2.json()["choices"][0]["logprobs"]
        assert len(lp["token_logprobs"]) == 4
        assert all(v <= 0.0 for v in lp["token_logprobs"])
        assert len(lp["top_logprobs"]) == 4 and all(
            len(d) == 1 for d in lp["top_logprobs"])

    run_with_client(app, fn)


def test_chat_stream_logprobs(app):
    """Streamed chat with logprobs: chunks carry per-token logprob entries
    (held-token merges allowed), totalling one entry per generated token."""

    async def fn(client):
        async with client.stream("POST", "/v1/chat/completions", json={
            "model": "tiny",
            "messages": [{"role": "user", "content": "hello"}],
            "max_tokens": 4, "temperature": 0, "ignore_eos": True,
            "stream": True, "logprobs": True, "top_logprobs": 1,
        }) as r:
            assert r.status_code == 200
            entries = []
            async for line in r.aiter_lines():
                if not line.startswith("data: ") or line == "data: [DONE]":
                    continue
                c = json.loads(line[6:])
                for ch in c.get("choices", []):
                    lp = ch.get("logprobs")
                    if lp:
                        entries.extend(lp["content"])
            assert len(entries) == 4
            assert all(e["logprob"] <= 0.0 and len(e["top_logprobs"]) == 1
                       for e in entries)

    run_with_client(app, fn)


def test_cli_flags_map_to_engine_config():
    """The operator-composed command line (controlplane/commands.py contract)
    parses into the right EngineConfig fields — including the speculative
    and prefix-cache flags the docs reference."""
    from arks_amd.server.__main__ import build_engine_config, parse_args

    args = parse_args([
        "--model", "/models/models/default/m", "--served-model-name", "m",
        "--tensor-parallel-size", "2", "--port", "8080",
        "--max-model-len", "4096", "--quantization", "fp8",
        "--kv-cache-dtype", "fp8", "--speculative", "draft",
        "--draft-model", "/models/models/default/m-draft",
        "--num-speculative-tokens", "3", "--no-prefix-cache",
    ])
    cfg = build_engine_config(args)
    assert cfg.model_path == "/models/models/default/m"
    assert cfg.served_model_name == "m"
    assert cfg.tensor_parallel_size == 2
    assert cfg.max_model_len == 4096
    assert cfg.quantization == "fp8"
    assert cfg.kv_cache_dtype == "fp8"
    assert cfg.speculative == "draft"
    assert cfg.draft_model == "/models/models/default/m-draft"
    assert cfg.num_speculative_tokens == 3
    assert cfg.enable_prefix_caching is False
    # preset form
    args2 = parse_args(["--model", "preset:tiny"])
    cfg2 = build_engine_config(args2)
    assert cfg2.preset == "tiny" and cfg2.model_path is None
    assert cfg2.enable_prefix_caching is True
