"""PD-disaggregation router (first-party replacement for the sglang-router
the reference deploys at arksdisaggregatedapplication_controller.go:1630-1670).

Forwards OpenAI-API requests to a decode instance, attaching an
`x-arks-prefill-addr` header naming the prefill instance the decode engine
must pull KV pages from (arks_amd/server/disagg.py). Policies:

  cache_aware  - prefill choice is a stable hash of the prompt prefix, so
                 repeated/shared prompts land on the same prefill worker and
                 hit its radix/prefix cache (arks_amd/engine/kv_cache.py);
                 decode choice is least-loaded round-robin.
  round_robin  - both choices rotate.

Worker sets are mutable (set_workers) so a service-discovery loop can update
them; without prefill workers the router degrades to a plain weighted proxy
(non-PD mode).
"""

from __future__ import annotations

import hashlib
import itertools
import json

import httpx
from fastapi import FastAPI, Request, Response
from fastapi.responses import JSONResponse, StreamingResponse
from prometheus_client import CollectorRegistry, Counter, Gauge, generate_latest

HOP_HEADERS = {"host", "content-length", "connection", "accept-encoding"}


class RouterState:
    def __init__(self, prefill_urls: list[str], decode_urls: list[str],
                 policy: str = "cache_aware", transport=None):
        self.prefill_urls = list(prefill_urls)
        self.decode_urls = list(decode_urls)
        self.policy = policy
        self.transport = transport
        self._rr_prefill = itertools.count()
        self._rr_decode = itertools.count()
        self.inflight: dict[str, int] = {}  # decode url -> open requests
        self.registry = CollectorRegistry()
        self.requests_total = Counter(
            "arks_router_requests_total", "routed requests",
            labelnames=["decode", "prefill"], registry=self.registry,
        )
        self.inflight_gauge = Gauge(
            "arks_router_inflight", "open requests per decode worker",
            labelnames=["decode"], registry=self.registry,
        )

    def acquire(self, decode: str) -> None:
        self.inflight[decode] = self.inflight.get(decode, 0) + 1
        self.inflight_gauge.labels(decode=decode).set(self.inflight[decode])

    def release(self, decode: str) -> None:
        self.inflight[decode] = max(self.inflight.get(decode, 0) - 1, 0)
        self.inflight_gauge.labels(decode=decode).set(self.inflight[decode])

    def set_workers(self, prefill_urls: list[str] | None = None,
                    decode_urls: list[str] | None = None) -> None:
        if prefill_urls is not None:
            self.prefill_urls = list(prefill_urls)
        if decode_urls is not None:
            self.decode_urls = list(decode_urls)

    # --- policies ---
    def pick_decode(self) -> str:
        """cache_aware: least-loaded decode worker (open-request count,
        rotating tiebreak); round_robin: plain rotation."""
        if not self.decode_urls:
            raise LookupError("no decode workers")
        n = len(self.decode_urls)
        start = next(self._rr_decode) % n
        if self.policy == "round_robin":
            return self.decode_urls[start]
        order = self.decode_urls[start:] + self.decode_urls[:start]
        return min(order, key=lambda u: self.inflight.get(u, 0))

    def pick_prefill(self, prompt_key: str) -> str | None:
        if not self.prefill_urls:
            return None
        if self.policy == "cache_aware" and prompt_key:
            h = int.from_bytes(
                hashlib.blake2b(prompt_key.encode(), digest_size=8).digest(),
                "little",
            )
            return self.prefill_urls[h % len(self.prefill_urls)]
        return self.prefill_urls[next(self._rr_prefill) % len(self.prefill_urls)]


def _prompt_key(body: dict) -> str:
    """Stable prefix key for cache-aware prefill affinity."""
    if "messages" in body:
        txt = json.dumps(body["messages"])[:512]
    else:
        p = body.get("prompt", "")
        txt = p if isinstance(p, str) else json.dumps(p)
        txt = txt[:512]
    return txt


def create_router_app(state: RouterState) -> FastAPI:
    app = FastAPI(title="arks_amd-router")
    app.state.router = state

    def _base(url: str) -> str:
        return url if url.startswith("http") else f"http://{url}"

    async def _proxy(raw: Request, path: str):
        body_bytes = await raw.body()
        try:
            body = json.loads(body_bytes) if body_bytes else {}
        except json.JSONDecodeError:
            body = {}
        try:
            decode = state.pick_decode()
        except LookupError:
            return JSONResponse(status_code=503,
                                content={"error": "no decode workers ready"})
        prefill = state.pick_prefill(_prompt_key(body))
        headers = {
            k: v for k, v in raw.headers.items() if k.lower() not in HOP_HEADERS
        }
        if prefill:
            headers["x-arks-prefill-addr"] = _base(prefill)
        # connect-failure failover: service discovery lags a dead worker by
        # up to one watch/poll interval — retry distinct decode workers
        # before surfacing 502 (the reference's router relies on the same
        # discovery loop and has the same window)
        tried: set[str] = set()
        resp = client = None
        for _ in range(min(3, max(len(state.decode_urls), 1))):
            if decode in tried:
                try:
                    decode = state.pick_decode()
                except LookupError:
                    break
                if decode in tried:
                    continue
            tried.add(decode)
            state.acquire(decode)
            client = httpx.AsyncClient(
                transport=state.transport, base_url=_base(decode),
                timeout=600.0)
            req = client.build_request("POST", path, content=body_bytes,
                                       headers=headers)
            try:
                resp = await client.send(req, stream=True)
                break
            except (httpx.ConnectError, httpx.ConnectTimeout):
                state.release(decode)
                await client.aclose()
                resp = client = None
            except Exception:
                state.release(decode)
                await client.aclose()
                raise
        if resp is None:
            return JSONResponse(
                status_code=502,
                content={"error": "no reachable decode worker"})
        state.requests_total.labels(decode=decode, prefill=prefill or "").inc()

        async def relay():
            try:
                async for chunk in resp.aiter_raw():
                    yield chunk
            finally:
                state.release(decode)
                await resp.aclose()
                await client.aclose()

        return StreamingResponse(
            relay(), status_code=resp.status_code,
            media_type=resp.headers.get("content-type"),
        )

    @app.post("/v1/chat/completions")
    async def chat(raw: Request):
        return await _proxy(raw, "/v1/chat/completions")

    @app.post("/v1/completions")
    async def completions(raw: Request):
        return await _proxy(raw, "/v1/completions")

    @app.get("/v1/models")
    async def models():
        try:
            decode = state.pick_decode()
        except LookupError:
            return JSONResponse(status_code=503,
                                content={"error": "no decode workers ready"})
        async with httpx.AsyncClient(transport=state.transport,
                                     base_url=_base(decode), timeout=30.0) as c:
            r = await c.get("/v1/models")
            return Response(content=r.content, status_code=r.status_code,
                            media_type=r.headers.get("content-type"))

    @app.get("/health")
    async def health():
        ok = bool(state.decode_urls)
        return Response(status_code=200 if ok else 503)

    @app.get("/metrics")
    async def metrics():
        return Response(content=generate_latest(state.registry),
                        media_type="text/plain")

    return app
