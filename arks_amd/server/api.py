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
import dataclasses
import json
import time
import uuid

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse, PlainTextResponse, Response, StreamingResponse

from ..engine import SamplingParams
from .async_engine import AsyncEngine
from .tokenizer import IncrementalDetokenizer
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




def _logprobs_of(req) -> int | None:
    lp = getattr(req, "logprobs", None)
    if isinstance(lp, bool):  # chat API shape: logprobs + top_logprobs
        return getattr(req, "top_logprobs", 0) if lp else None
    return lp  # completions API shape: int | None


def _stop_list(stop) -> list[str]:
    if stop is None:
        return []
    return [stop] if isinstance(stop, str) else [x for x in stop if x]


def _choice_sp(sp: SamplingParams, i: int) -> SamplingParams:
    """Per-choice SamplingParams for n>1 fan-out: a fixed seed must still
    produce n DISTINCT choices (identical seeds would sample n copies), so
    each choice derives seed+i; choice 0 keeps the request's exact seed."""
    if sp.seed is None or i == 0:
        return sp
    return dataclasses.replace(sp, seed=sp.seed + i)


class StopStringTracker:
    """Incremental stop-string matching over streamed text: holds back the
    longest possible partial match so a stop split across two pieces is
    still caught before it is emitted."""

    def __init__(self, stops: list[str]):
        self.stops = stops
        self.buf = ""
        self.hold = max((len(x) for x in stops), default=1) - 1

    def feed(self, piece: str) -> tuple[str, bool]:
        if not self.stops:
            return piece, False
        self.buf += piece
        cut = min((i for i in (self.buf.find(x) for x in self.stops) if i != -1),
                  default=-1)
        if cut != -1:
            emit, self.buf = self.buf[:cut], ""
            return emit, True
        emit = self.buf[: len(self.buf) - self.hold] if self.hold else self.buf
        self.buf = self.buf[len(emit):]
        return emit, False

    def flush(self) -> str:
        out, self.buf = self.buf, ""
        return out


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
        # go unready after a fatal engine-loop error so the endpoint
        # controller pulls this pod out of the HTTPRoute
        return Response(status_code=503 if engine.failed else 200)

    @app.get("/v1/models")
    async def models():
        return ModelList(data=[ModelCard(id=served_model_name)]).model_dump()

    @app.get("/metrics")
    async def metrics():
        return PlainTextResponse(engine.metrics.render().decode())

    def _sampling(req, max_tokens) -> SamplingParams:
        return SamplingParams(
            max_tokens=max_tokens if max_tokens is not None else 1024,
            temperature=max(0.0, req.temperature),
            top_p=req.top_p,
            top_k=getattr(req, "top_k", 0),
            min_p=getattr(req, "min_p", 0.0),
            presence_penalty=getattr(req, "presence_penalty", 0.0),
            frequency_penalty=getattr(req, "frequency_penalty", 0.0),
            repetition_penalty=getattr(req, "repetition_penalty", 1.0),
            logprobs=_logprobs_of(req),
            seed=req.seed,
            ignore_eos=req.ignore_eos,
            logit_bias=(
                {int(k): float(v) for k, v in req.logit_bias.items()}
                if getattr(req, "logit_bias", None) else None
            ),
            min_tokens=getattr(req, "min_tokens", 0),
        )


    async def _collect_choice(rid, token_ids, sp, stops, prefill_addr):
        """One non-streamed choice: (token_ids, text, finish, n_generated)."""
        tracker = StopStringTracker(stops)
        # incremental detok even in the non-stream path: per-token decode
        # garbles multi-byte runes split across BPE tokens (and stop
        # matching would then run over the garbled text)
        detok = IncrementalDetokenizer(tokenizer)
        out_ids: list[int] = []
        text_acc = ""
        finish = None
        try:
            async for out in engine.generate_stream(rid, token_ids, sp,
                                                    prefill_addr=prefill_addr):
                out_ids.append(out.new_token_id)
                if stops:
                    emit, stopped = tracker.feed(detok.feed(out.new_token_id))
                    text_acc += emit
                    if stopped:
                        finish = "stop"
                        engine.abort(rid)
                        break
                if out.finished:
                    finish = out.finish_reason
        except asyncio.CancelledError:
            engine.abort(rid)  # client went away mid-request
            raise
        if stops and finish != "stop":
            emit, _ = tracker.feed(detok.flush())
            text_acc += emit + tracker.flush()
        text = text_acc if stops else tokenizer.decode(out_ids)
        return out_ids, text, finish or "stop", len(out_ids)

    @app.post("/v1/chat/completions")
    async def chat_completions(req: ChatCompletionRequest, raw: Request):
        if req.model != served_model_name:
            return _error(404, f"model {req.model!r} not found")
        prompt = tokenizer.apply_chat_template([m.model_dump() for m in req.messages])
        token_ids = tokenizer.encode(prompt)
        if len(token_ids) > engine.cfg.max_model_len:
            return _error(400, f"prompt is {len(token_ids)} tokens; "
                               f"max_model_len is {engine.cfg.max_model_len}")
        sp = _sampling(req, req.max_completion_tokens or req.max_tokens)
        rid = f"chatcmpl-{uuid.uuid4().hex}"
        if req.stream:
            if req.n > 1:
                return _error(400, "n > 1 is not supported with stream=true")
            return StreamingResponse(
                _chat_stream(engine, tokenizer, req, rid, token_ids, sp, raw),
                media_type="text/event-stream",
            )
        if req.n > 1:
            paddr = raw.headers.get("x-arks-prefill-addr")
            stops_n = _stop_list(req.stop)
            results = await asyncio.gather(*[
                _collect_choice(f"{rid}-{i}", token_ids, _choice_sp(sp, i),
                                stops_n, paddr)
                for i in range(req.n)
            ])
            total_out = sum(r[3] for r in results)
            return ChatCompletionResponse(
                id=rid, model=req.model,
                choices=[
                    ChatChoice(index=i,
                               message=ChatMessage(role="assistant", content=r[1]),
                               finish_reason=r[2])
                    for i, r in enumerate(results)
                ],
                usage=Usage(prompt_tokens=len(token_ids),
                            completion_tokens=total_out,
                            total_tokens=len(token_ids) + total_out),
            ).model_dump()
        stops = _stop_list(req.stop)
        tracker = StopStringTracker(stops)
        detok = IncrementalDetokenizer(tokenizer)
        text_ids: list[int] = []
        text_acc = ""
        lp_content: list[dict] = []
        finish = None
        try:
            async for out in engine.generate_stream(
                rid, token_ids, sp,
                prefill_addr=raw.headers.get("x-arks-prefill-addr"),
            ):
                text_ids.append(out.new_token_id)
                if out.logprob is not None:
                    lp_content.append({
                        "token": tokenizer.decode([out.new_token_id]),
                        "logprob": out.logprob,
                        "top_logprobs": [
                            {"token": tokenizer.decode([t]), "logprob": v}
                            for t, v in (out.top_logprobs or {}).items()
                        ],
                    })
                if stops:
                    emit, stopped = tracker.feed(detok.feed(out.new_token_id))
                    text_acc += emit
                    if stopped:
                        finish = "stop"
                        engine.abort(rid)
                        break
                if out.finished:
                    finish = out.finish_reason
        except asyncio.CancelledError:
            engine.abort(rid)  # client went away mid-request
            raise
        if stops and finish != "stop":
            emit, _ = tracker.feed(detok.flush())
            text_acc += emit + tracker.flush()
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
                    message=ChatMessage(
                        role="assistant",
                        content=text_acc if stops else tokenizer.decode(text_ids),
                    ),
                    finish_reason=finish or "stop",
                    logprobs={"content": lp_content} if lp_content else None,
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
        tracker = StopStringTracker(_stop_list(req.stop))
        detok = IncrementalDetokenizer(tokenizer)
        pending_lp: list[dict] = []  # per-token entries since the last emit
        try:
            async for out in engine.generate_stream(
            rid, token_ids, sp, prefill_addr=raw.headers.get("x-arks-prefill-addr")
        ):
                if await raw.is_disconnected():
                    engine.abort(rid)
                    return
                n_out += 1
                if out.logprob is not None:
                    pending_lp.append({
                        "token": tokenizer.decode([out.new_token_id]),
                        "logprob": out.logprob,
                        "top_logprobs": [
                            {"token": tokenizer.decode([t]), "logprob": v}
                            for t, v in (out.top_logprobs or {}).items()
                        ],
                    })
                text_in = detok.feed(out.new_token_id)
                if out.finished:
                    text_in += detok.flush()
                piece, stopped = tracker.feed(text_in)
                if stopped:
                    finish = "stop"
                    engine.abort(rid)
                elif out.finished:
                    finish = out.finish_reason
                    piece += tracker.flush()
                done = stopped or out.finished
                if piece or done:
                    chunk = ChatCompletionChunk(
                        id=rid, model=req.model, created=created,
                        choices=[ChatDeltaChoice(
                            delta={"content": piece},
                            logprobs=(
                                {"content": pending_lp} if pending_lp else None
                            ),
                            finish_reason=finish if done else None)],
                    )
                    pending_lp = []
                    yield f"data: {chunk.model_dump_json(exclude_none=True)}\n\n"
                if stopped:
                    break
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
        if len(token_ids) > engine.cfg.max_model_len:
            return _error(400, f"prompt is {len(token_ids)} tokens; "
                               f"max_model_len is {engine.cfg.max_model_len}")
        sp = _sampling(req, req.max_tokens)
        if req.echo:
            if req.n > 1 or req.stream:
                return _error(400, "echo is not supported with n > 1 or stream")
            # lm-eval-style prompt scoring: logits at every prompt position
            sp.prompt_logprobs = req.logprobs is not None
        rid = f"cmpl-{uuid.uuid4().hex}"
        if req.stream:
            if req.n > 1:
                return _error(400, "n > 1 is not supported with stream=true")
            return StreamingResponse(
                _completion_stream(engine, tokenizer, req, rid, token_ids, sp, raw),
                media_type="text/event-stream",
            )
        if req.n > 1:
            paddr = raw.headers.get("x-arks-prefill-addr")
            stops_n = _stop_list(req.stop)
            results = await asyncio.gather(*[
                _collect_choice(f"{rid}-{i}", token_ids, _choice_sp(sp, i),
                                stops_n, paddr)
                for i in range(req.n)
            ])
            total_out = sum(r[3] for r in results)
            return CompletionResponse(
                id=rid, model=req.model,
                choices=[
                    CompletionChoice(index=i, text=r[1], finish_reason=r[2])
                    for i, r in enumerate(results)
                ],
                usage=Usage(prompt_tokens=len(token_ids),
                            completion_tokens=total_out,
                            total_tokens=len(token_ids) + total_out),
            ).model_dump()
        stops = _stop_list(req.stop)
        tracker = StopStringTracker(stops)
        detok = IncrementalDetokenizer(tokenizer)
        out_ids: list[int] = []
        text_acc = ""
        finish = None
        token_lps: list[float] = []
        top_lps: list[dict] = []
        prompt_lps: list[float] | None = None
        try:
            async for out in engine.generate_stream(
                rid, token_ids, sp,
                prefill_addr=raw.headers.get("x-arks-prefill-addr"),
            ):
                out_ids.append(out.new_token_id)
                if out.prompt_logprobs is not None:
                    prompt_lps = out.prompt_logprobs
                if out.logprob is not None:
                    token_lps.append(out.logprob)
                    top_lps.append({
                        tokenizer.decode([t]): v
                        for t, v in (out.top_logprobs or {}).items()
                    })
                if stops:
                    emit, stopped = tracker.feed(detok.feed(out.new_token_id))
                    text_acc += emit
                    if stopped:
                        finish = "stop"
                        engine.abort(rid)
                        break
                if out.finished:
                    finish = out.finish_reason
        except asyncio.CancelledError:
            engine.abort(rid)  # client went away mid-request
            raise
        if stops and finish != "stop":
            emit, _ = tracker.feed(detok.flush())
            text_acc += emit + tracker.flush()
        text = text_acc if stops else tokenizer.decode(out_ids)
        lp_obj = None
        if token_lps:
            lp_obj = {
                "tokens": [tokenizer.decode([t]) for t in out_ids],
                "token_logprobs": list(token_lps),
                "top_logprobs": list(top_lps),
            }
        if req.echo:
            # OpenAI echo contract: prompt text precedes the completion and,
            # with logprobs, prompt tokens are scored (first one null)
            text = tokenizer.decode(token_ids) + text
            if req.logprobs is not None:
                lp_obj = {
                    "tokens": [tokenizer.decode([t]) for t in token_ids]
                    + (lp_obj["tokens"] if lp_obj else []),
                    "token_logprobs": [None] + (prompt_lps or [])
                    + (lp_obj["token_logprobs"] if lp_obj else []),
                    "top_logprobs": [None] * len(token_ids)
                    + (lp_obj["top_logprobs"] if lp_obj else []),
                }
        return CompletionResponse(
            id=rid, model=req.model,
            choices=[CompletionChoice(
                text=text,
                finish_reason=finish or "stop",
                logprobs=lp_obj)],
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
        tracker = StopStringTracker(_stop_list(req.stop))
        detok = IncrementalDetokenizer(tokenizer)
        try:
            async for out in engine.generate_stream(
            rid, token_ids, sp, prefill_addr=raw.headers.get("x-arks-prefill-addr")
        ):
                if await raw.is_disconnected():
                    engine.abort(rid)
                    return
                n_out += 1
                text_in = detok.feed(out.new_token_id)
                if out.finished:
                    text_in += detok.flush()
                piece, stopped = tracker.feed(text_in)
                if stopped:
                    finish = "stop"
                    engine.abort(rid)
                elif out.finished:
                    finish = out.finish_reason
                    piece += tracker.flush()
                else:
                    finish = None
                done = stopped or out.finished
                if piece or done:
                    chunk = {
                        "id": rid, "object": "text_completion", "created": created,
                        "model": req.model,
                        "choices": [{
                            "index": 0,
                            "text": piece,
                            "finish_reason": finish if done else None,
                        }],
                    }
                    yield f"data: {json.dumps(chunk)}\n\n"
                if stopped:
                    break
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
