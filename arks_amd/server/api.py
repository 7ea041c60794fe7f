"""OpenAI-compatible HTTP server (FastAPI/ASGI).

Implements the runtime-slot HTTP contract the reference delegates to
vLLM/SGLang images: /v1/chat/completions and
/v1/completions with SSE streaming whose FINAL chunk carries `usage`
(the gateway's accounting depends on it — reference
handle_response.go:113-133), /v1/models, a readiness probe on the serving
port, and vLLM-compatible Prometheus /metrics for the runtime
ServiceMonitor.
"""

from __future__ import annotations

import asyncio
import json
import time
import uuid

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse, PlainTextResponse, Response, StreamingResponse

from ..engine import SamplingParams
from .async_engine import AsyncEngine
from .openai_types import (
    ChatChoice,
    ChatCompletionChunk,
    ChatCompletionRequest,
    ChatCompletionResponse,
    ChatDeltaChoice,
    ChatMessage,
    CompletionChoice,
    CompletionRequest,
    CompletionResponse,
    ErrorResponse,
    ModelCard,
    ModelList,
    Usage,
)


def _error(status: int, message: str) -> JSONResponse:
    return JSONResponse(
        status_code=status, content=ErrorResponse.make(message, status).model_dump()
    )


def create_app(engine: AsyncEngine, served_model_name: str, tokenizer,
               disagg_mode: str | None = None) -> FastAPI:
    from contextlib import asynccontextmanager

    @asynccontextmanager
    async def lifespan(_app):
        await engine.start()
        try:
            yield
        finally:
            await engine.stop()

    app = FastAPI(title="arks_amd", lifespan=lifespan)
    app.state.engine = engine
    app.state.tokenizer = tokenizer
    app.state.model_name = served_model_name

    if disagg_mode == "prefill":
        from .disagg import add_prefill_routes

        add_prefill_routes(app, engine)

    @app.get("/health")
    async def health():
        return Response(status_code=200)

    @app.get("/v1/models")
    async def models():
        return ModelList(data=[ModelCard(id=served_model_name)]).model_dump()

    @app.get("/metrics")
    async def metrics():
        return PlainTextResponse(engine.metrics.render().decode())

    def _sampling(max_tokens, temperature, top_p, ignore_eos) -> SamplingParams:
        return SamplingParams(
            max_tokens=max_tokens if max_tokens is not None else 1024,
            temperature=max(0.0, temperature),
            top_p=top_p,
            ignore_eos=ignore_eos,
        )

    @app.post("/v1/chat/completions")
    async def chat_completions(req: ChatCompletionRequest, raw: Request):
        if req.model != served_model_name:
            return _error(404, f"model {req.model!r} not found")
        prompt = tokenizer.apply_chat_template([m.model_dump() for m in req.messages])
        token_ids = tokenizer.encode(prompt)
        sp = _sampling(
            req.max_completion_tokens or req.max_tokens,
            req.temperature, req.top_p, req.ignore_eos,
        )
        rid = f"chatcmpl-{uuid.uuid4().hex}"
        if req.stream:
            return StreamingResponse(
                _chat_stream(engine, tokenizer, req, rid, token_ids, sp, raw),
                media_type="text/event-stream",
            )
        text_ids: list[int] = []
        finish = None
        async for out in engine.generate_stream(
            rid, token_ids, sp, prefill_addr=raw.headers.get("x-arks-prefill-addr")
        ):
            text_ids.append(out.new_token_id)
            if out.finished:
                finish = out.finish_reason
        usage = Usage(
            prompt_tokens=len(token_ids),
            completion_tokens=len(text_ids),
            total_tokens=len(token_ids) + len(text_ids),
        )
        return ChatCompletionResponse(
            id=rid,
            model=req.model,
            choices=[
                ChatChoice(
                    message=ChatMessage(role="assistant", content=tokenizer.decode(text_ids)),
                    finish_reason=finish or "stop",
                )
            ],
            usage=usage,
        ).model_dump()

    async def _chat_stream(engine, tokenizer, req, rid, token_ids, sp, raw):
        created = int(time.time())
        include_usage = bool(req.stream_options and req.stream_options.include_usage)
        first = ChatCompletionChunk(
            id=rid, model=req.model, created=created,
            choices=[ChatDeltaChoice(delta={"role": "assistant", "content": ""})],
        )
        yield f"data: {first.model_dump_json(exclude_none=True)}\n\n"
        n_out = 0
        finish = None
        try:
            async for out in engine.generate_stream(
            rid, token_ids, sp, prefill_addr=raw.headers.get("x-arks-prefill-addr")
        ):
                if await raw.is_disconnected():
                    engine.abort(rid)
                    return
                n_out += 1
                piece = tokenizer.decode([out.new_token_id])
                chunk = ChatCompletionChunk(
                    id=rid, model=req.model, created=created,
                    choices=[ChatDeltaChoice(delta={"content": piece},
                                             finish_reason=out.finish_reason if out.finished else None)],
                )
                if out.finished:
                    finish = out.finish_reason
                yield f"data: {chunk.model_dump_json(exclude_none=True)}\n\n"
        except asyncio.CancelledError:
            engine.abort(rid)
            raise
        if include_usage:
            # final chunk: EMPTY choices + usage (the gateway parses exactly
            # this shape — reference handle_response.go:113-133)
            final = ChatCompletionChunk(
                id=rid, model=req.model, created=created, choices=[],
                usage=Usage(
                    prompt_tokens=len(token_ids),
                    completion_tokens=n_out,
                    total_tokens=len(token_ids) + n_out,
                ),
            )
            yield f"data: {final.model_dump_json(exclude_none=True)}\n\n"
        yield "data: [DONE]\n\n"

    @app.post("/v1/completions")
    async def completions(req: CompletionRequest, raw: Request):
        if req.model != served_model_name:
            return _error(404, f"model {req.model!r} not found")
        # normalize prompt to a single token list (n=1, single prompt v0)
        p = req.prompt
        if isinstance(p, list) and p and isinstance(p[0], int):
            token_ids = list(p)
        elif isinstance(p, str):
            token_ids = tokenizer.encode(p)
        elif isinstance(p, list) and p and isinstance(p[0], str):
            token_ids = tokenizer.encode(p[0])
        elif isinstance(p, list) and p and isinstance(p[0], list):
            token_ids = list(p[0])
        else:
            return _error(400, "invalid prompt")
        sp = _sampling(req.max_tokens, req.temperature, req.top_p, req.ignore_eos)
        rid = f"cmpl-{uuid.uuid4().hex}"
        if req.stream:
            return StreamingResponse(
                _completion_stream(engine, tokenizer, req, rid, token_ids, sp, raw),
                media_type="text/event-stream",
            )
        out_ids: list[int] = []
        finish = None
        async for out in engine.generate_stream(
            rid, token_ids, sp, prefill_addr=raw.headers.get("x-arks-prefill-addr")
        ):
            out_ids.append(out.new_token_id)
            if out.finished:
                finish = out.finish_reason
        return CompletionResponse(
            id=rid, model=req.model,
            choices=[CompletionChoice(text=tokenizer.decode(out_ids),
                                      finish_reason=finish or "stop")],
            usage=Usage(
                prompt_tokens=len(token_ids),
                completion_tokens=len(out_ids),
                total_tokens=len(token_ids) + len(out_ids),
            ),
        ).model_dump()

    async def _completion_stream(engine, tokenizer, req, rid, token_ids, sp, raw):
        created = int(time.time())
        include_usage = bool(req.stream_options and req.stream_options.include_usage)
        n_out = 0
        try:
            async for out in engine.generate_stream(
            rid, token_ids, sp, prefill_addr=raw.headers.get("x-arks-prefill-addr")
        ):
                if await raw.is_disconnected():
                    engine.abort(rid)
                    return
                n_out += 1
                chunk = {
                    "id": rid, "object": "text_completion", "created": created,
                    "model": req.model,
                    "choices": [{
                        "index": 0,
                        "text": tokenizer.decode([out.new_token_id]),
                        "finish_reason": out.finish_reason if out.finished else None,
                    }],
                }
                yield f"data: {json.dumps(chunk)}\n\n"
        except asyncio.CancelledError:
            engine.abort(rid)
            raise
        if include_usage:
            final = {
                "id": rid, "object": "text_completion", "created": created,
                "model": req.model, "choices": [],
                "usage": {
                    "prompt_tokens": len(token_ids),
                    "completion_tokens": n_out,
                    "total_tokens": len(token_ids) + n_out,
                },
            }
            yield f"data: {json.dumps(final)}\n\n"
        yield "data: [DONE]\n\n"

    return app
