"""The gateway data plane: OpenAI-front auth / rate-limit / quota proxy.

Native equivalent of the reference's Envoy ext_proc plugin
(pkg/gateway/gateway.go + handle_request.go + handle_response.go): here the
gateway IS the HTTP hop (ASGI reverse proxy) rather than an Envoy side-call,
but the request pipeline is the same, phase for phase:

  Bearer token -> qos lookup (401) -> model membership (400)
  -> streaming requires stream_options.include_usage (400)
  -> checkRateLimit + checkQuota (429) -> incr request counters
  -> inject model/namespace/username headers -> route to a ready backend
  -> count usage from the response (final SSE chunk / JSON body)
  -> incr token counters + quota usage -> Prometheus metrics.
"""

from __future__ import annotations

import json
import random
import time
from typing import Callable

import httpx
from prometheus_client import CollectorRegistry, Counter, Histogram, generate_latest, Gauge
from starlette.applications import Starlette
from starlette.requests import Request
from starlette.responses import JSONResponse, PlainTextResponse, Response, StreamingResponse
from starlette.routing import Route

from ..controlplane.store import Store
from .limiter import RULES, RateLimiter, TYPE_REQUEST, TYPE_TOKEN
from .provider import ConfigProvider
from .quota import QuotaService


class GatewayMetrics:
    def __init__(self, registry: CollectorRegistry | None = None):
        self.registry = registry or CollectorRegistry()
        self.requests_total = Counter(
            "gateway_requests_total", "requests",
            ["namespace", "user", "model", "status"], registry=self.registry,
        )
        self.request_duration = Histogram(
            "gateway_request_duration_seconds", "e2e duration",
            ["namespace", "user", "model"], registry=self.registry,
            buckets=(0.1, 0.5, 1, 2, 5, 10, 30, 60),
        )
        self.token_usage = Counter(
            "gateway_token_usage", "token usage",
            ["namespace", "user", "model", "type"], registry=self.registry,
        )
        self.rate_limit_hits = Counter(
            "gateway_rate_limit_hits_total", "429s",
            ["namespace", "user", "model", "rule"], registry=self.registry,
        )
        self.errors_total = Counter(
            "gateway_errors_total", "errors", ["kind"], registry=self.registry
        )
        # quota gauges (the reference left these as TODO stubs,
        # collector.go:58-75; populated here by the provider's 10 s
        # quota sync loop)
        self.quota_usage = Gauge(
            "gateway_quota_usage", "cumulative quota usage",
            ["namespace", "quota", "type"], registry=self.registry,
        )
        self.quota_limit = Gauge(
            "gateway_quota_limit", "configured quota limit",
            ["namespace", "quota", "type"], registry=self.registry,
        )


class OutlierDetector:
    """Passive outlier ejection for the no-Envoy mode: a backend with
    `threshold` CONSECUTIVE 5xx/connect errors is ejected for
    `ejection_s` (the reference delegates this to Envoy's
    BackendTrafficPolicy passive health check: 3x5xx -> 30 s)."""

    def __init__(self, threshold: int = 3, ejection_s: float = 30.0,
                 clock=time.time):
        self.threshold = threshold
        self.ejection_s = ejection_s
        self.clock = clock
        self._consecutive: dict[str, int] = {}
        self._ejected_until: dict[str, float] = {}

    def record(self, backend: str, ok: bool) -> None:
        if ok:
            self._consecutive.pop(backend, None)
            return
        n = self._consecutive.get(backend, 0) + 1
        self._consecutive[backend] = n
        if n >= self.threshold:
            self._ejected_until[backend] = self.clock() + self.ejection_s
            self._consecutive[backend] = 0

    def is_ejected(self, backend: str) -> bool:
        until = self._ejected_until.get(backend)
        if until is None:
            return False
        if until <= self.clock():
            del self._ejected_until[backend]
            return False
        return True


class BackendResolver:
    """Maps (namespace, model) -> base URL of a READY backend, by reading the
    HTTPRoute objects the endpoint controller generates (weighted pick,
    passive-outlier-aware)."""

    def __init__(self, store: Store,
                 url_for_service: Callable[[str, str], str] | None = None,
                 outliers: OutlierDetector | None = None):
        self.store = store
        self.url_for_service = url_for_service or (
            lambda ns, svc: f"http://{svc}.{ns}.svc:8080"
        )
        self.outliers = outliers or OutlierDetector()

    @staticmethod
    def _match_entry(m: dict, path: str, headers: dict[str, str],
                     method: str | None = None) -> bool:
        """One HTTPRouteMatch: method AND path AND all header conditions
        must hold (gateway-api semantics; absent fields match anything)."""
        want = m.get("method")
        if want and method and want.upper() != method.upper():
            return False
        pm = m.get("path") or {}
        if pm:
            val = pm.get("value", "/")
            typ = pm.get("type", "PathPrefix")
            if typ == "Exact":
                if path != val:
                    return False
            else:  # PathPrefix
                if not path.startswith(val):
                    return False
        for hm in m.get("headers", []) or []:
            if headers.get(hm.get("name", "").lower()) != hm.get("value"):
                return False
        return True

    def resolve(self, namespace: str, model: str, path: str = "/",
                headers: dict[str, str] | None = None,
                method: str | None = None) -> str | None:
        """Pick a backend from the endpoint's HTTPRoute, evaluating EVERY
        rule's match conditions (a rule matches when any of its match
        entries does — user MatchConfigs from the ArksEndpoint spec are
        honored, reference arksendpoint_controller.go:349-369); first
        matching rule with backendRefs wins."""
        route = self.store.get_opt("HTTPRoute", namespace, model)
        if route is None:
            return None
        hdrs = {k.lower(): v for k, v in (headers or {}).items()}
        hdrs.setdefault("namespace", namespace)
        hdrs.setdefault("model", model)
        refs = None
        for rule in route.get("spec", {}).get("rules", []):
            matches = rule.get("matches") or [{}]
            if any(self._match_entry(m, path, hdrs, method)
                   for m in matches):
                refs = rule.get("backendRefs", [])
                if refs:
                    break
        if not refs:
            return None
        # skip ejected backends (fall back to the full set if ALL are
        # ejected — Envoy's maxEjectionPercent analogue)
        live = [r for r in refs
                if not self.outliers.is_ejected(
                    self.url_for_service(namespace, r["name"]))]
        if live:
            refs = live
        weights = [max(int(r.get("weight", 1)), 0) for r in refs]
        total = sum(weights)
        if total <= 0:
            return None
        pick = random.uniform(0, total)
        acc = 0.0
        for r, w in zip(refs, weights):
            acc += w
            if pick <= acc:
                return self.url_for_service(namespace, r["name"])
        return self.url_for_service(namespace, refs[-1]["name"])


def _err(status: int, message: str, headers: dict | None = None) -> JSONResponse:
    # same JSON error shape as reference util.go:40-77; limit/quota errors
    # carry x-error-* headers naming the tripped rule (reference types.go)
    return JSONResponse(
        status_code=status,
        content={"error": {"message": message, "code": status}},
        headers=headers,
    )


def create_gateway_app(
    store: Store,
    limiter: RateLimiter | None = None,
    quota_service: QuotaService | None = None,
    resolver: BackendResolver | None = None,
    transport: httpx.AsyncBaseTransport | None = None,
    registry: CollectorRegistry | None = None,
) -> Starlette:
    limiter = limiter or RateLimiter()
    quota_service = quota_service or QuotaService()
    metrics = GatewayMetrics(registry)
    provider = ConfigProvider(store, quota_service, metrics=metrics)
    resolver = resolver or BackendResolver(store)
    client = httpx.AsyncClient(transport=transport, timeout=300.0)

    def _auth(request: Request) -> str | None:
        auth = request.headers.get("authorization", "")
        if not auth.lower().startswith("bearer "):
            return None
        return auth[7:].strip()

    async def proxy(request: Request) -> Response:
        t0 = time.time()
        token = _auth(request)
        if not token:
            metrics.errors_total.labels(kind="unauthorized").inc()
            return _err(401, "missing or malformed Authorization bearer token")
        body = await request.body()
        try:
            payload = json.loads(body)
        except Exception:
            return _err(400, "invalid JSON body")
        model = payload.get("model", "")
        stream = bool(payload.get("stream", False))
        include_usage = bool(
            (payload.get("stream_options") or {}).get("include_usage", False)
        )
        qos = provider.get_qos_by_token(token, model)
        if qos is None:
            metrics.errors_total.labels(kind="forbidden").inc()
            return _err(401, "invalid token or no QoS for this model")
        labels = dict(namespace=qos.namespace, user=qos.user, model=model)
        if model not in provider.get_model_list(qos.namespace):
            metrics.requests_total.labels(**labels, status="400").inc()
            return _err(400, f"model {model!r} not available")
        if stream and not include_usage:
            # the gateway cannot account streamed usage otherwise
            # (reference handle_request.go:156-171)
            metrics.requests_total.labels(**labels, status="400").inc()
            return _err(400, "streaming requires stream_options.include_usage=true")

        descriptors = qos.limit_descriptors()
        ok, rule = limiter.check_limit(descriptors, request=1)
        if not ok:
            metrics.rate_limit_hits.labels(**labels, rule=rule).inc()
            metrics.requests_total.labels(**labels, status="429").inc()
            return _err(429, f"rate limit exceeded: {rule}",
                        headers={"x-error-type": "rate-limit",
                                 "x-error-rule": str(rule)})
        qdesc = provider.get_quota_descriptors(qos)
        ok, qtype = quota_service.check(qdesc)
        if not ok:
            metrics.requests_total.labels(**labels, status="429").inc()
            return _err(429, f"quota exceeded: {qtype}",
                        headers={"x-error-type": "quota",
                                 "x-error-rule": str(qtype)})
        # incr request-type counters (rpm/rpd)
        limiter.do_limit(
            [d for d in descriptors if RULES[d.rule].type == TYPE_REQUEST], 1
        )

        base = resolver.resolve(
            qos.namespace, model, path=request.url.path,
            headers={"model": model, "namespace": qos.namespace,
                     "username": qos.user},
            method=request.method,
        )
        if base is None:
            metrics.requests_total.labels(**labels, status="503").inc()
            return _err(503, "no ready backend for model")

        def account_usage(usage: dict) -> None:
            pt = int(usage.get("prompt_tokens", 0))
            ct = int(usage.get("completion_tokens", 0))
            tt = int(usage.get("total_tokens", pt + ct))
            limiter.do_limit(
                [d for d in descriptors if RULES[d.rule].type == TYPE_TOKEN], tt
            )
            if qos.quota_name:
                quota_service.incr_usage(qos.namespace, qos.quota_name, "prompt", pt)
                quota_service.incr_usage(qos.namespace, qos.quota_name, "response", ct)
                quota_service.incr_usage(qos.namespace, qos.quota_name, "total", tt)
            metrics.token_usage.labels(**labels, type="input").inc(pt)
            metrics.token_usage.labels(**labels, type="output").inc(ct)

        # routing headers the HTTPRoute matches on (handle_request.go:208-231)
        fwd_headers = {
            "content-type": "application/json",
            "model": model,
            "namespace": qos.namespace,
            "username": qos.user,
        }
        url = base.rstrip("/") + request.url.path

        if not stream:
            try:
                resp = await client.post(url, content=body, headers=fwd_headers)
            except httpx.HTTPError as e:
                resolver.outliers.record(base, False)
                metrics.requests_total.labels(**labels, status="502").inc()
                return _err(502, f"backend error: {e}")
            resolver.outliers.record(base, resp.status_code < 500)
            status = str(resp.status_code)
            if resp.status_code == 200:
                try:
                    account_usage(resp.json().get("usage") or {})
                except Exception:
                    metrics.errors_total.labels(kind="usage_parse").inc()
            metrics.requests_total.labels(**labels, status=status).inc()
            metrics.request_duration.labels(**labels).observe(time.time() - t0)
            return Response(
                content=resp.content, status_code=resp.status_code,
                media_type=resp.headers.get("content-type"),
            )

        # streaming: relay SSE, parse the final usage chunk
        req = client.build_request("POST", url, content=body, headers=fwd_headers)
        try:
            upstream = await client.send(req, stream=True)
        except httpx.HTTPError as e:
            resolver.outliers.record(base, False)
            metrics.requests_total.labels(**labels, status="502").inc()
            return _err(502, f"backend error: {e}")
        resolver.outliers.record(base, upstream.status_code < 500)
        if upstream.status_code != 200:
            content = await upstream.aread()
            await upstream.aclose()
            metrics.requests_total.labels(**labels, status=str(upstream.status_code)).inc()
            return Response(content=content, status_code=upstream.status_code)

        async def relay():
            buf = b""
            try:
                async for chunk in upstream.aiter_bytes():
                    buf += chunk
                    while b"\n\n" in buf:
                        event, buf = buf.split(b"\n\n", 1)
                        if event.startswith(b"data: "):
                            data = event[len(b"data: "):]
                            if data.strip() != b"[DONE]":
                                try:
                                    obj = json.loads(data)
                                    # usage is in the final chunk with empty
                                    # choices (handle_response.go:113-133)
                                    if obj.get("usage") and not obj.get("choices"):
                                        account_usage(obj["usage"])
                                except Exception:
                                    pass
                        yield event + b"\n\n"
                if buf:
                    yield buf
            finally:
                await upstream.aclose()
                metrics.requests_total.labels(**labels, status="200").inc()
                metrics.request_duration.labels(**labels).observe(time.time() - t0)

        return StreamingResponse(relay(), media_type="text/event-stream")

    async def models(request: Request) -> Response:
        token = _auth(request)
        if not token:
            return _err(401, "missing bearer token")
        names = provider.get_models_by_token(token)
        return JSONResponse(
            {
                "object": "list",
                "data": [
                    {"id": n, "object": "model", "owned_by": "arks"} for n in names
                ],
            }
        )

    async def metrics_ep(request: Request) -> Response:
        return PlainTextResponse(generate_latest(metrics.registry).decode())

    async def health(request: Request) -> Response:
        return Response(status_code=200)

    app = Starlette(
        routes=[
            Route("/v1/chat/completions", proxy, methods=["POST"]),
            Route("/v1/completions", proxy, methods=["POST"]),
            Route("/v1/models", models, methods=["GET"]),
            Route("/metrics", metrics_ep, methods=["GET"]),
            Route("/health", health, methods=["GET"]),
        ]
    )
    app.state.provider = provider
    app.state.limiter = limiter
    app.state.quota_service = quota_service
    app.state.client = client
    return app
