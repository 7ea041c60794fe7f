"""Envoy ext_proc gRPC servicer — the reference gateway's data-plane
protocol (pkg/gateway/gateway.go:77-138): one bidirectional gRPC stream
per HTTP request, four phases (RequestHeaders -> RequestBody ->
ResponseHeaders -> ResponseBody), sharing the same limiter / quota /
config-provider pipeline as the ASGI proxy (arks_amd/gateway/app.py, the
no-Envoy mode).

Phase behavior mirrors the reference:
  RequestHeaders  — extract `Authorization: Bearer`, 401 ImmediateResponse
                    if absent; set x-went-into-req-headers, ClearRouteCache
                    (handle_request.go:37-79).
  RequestBody     — parse {model, stream, stream_options.include_usage};
                    QoS lookup (401), model membership (400), streaming
                    requires include_usage (400), rate-limit + quota checks
                    (429 + x-error-* headers), incr request counters;
                    inject routing headers model/namespace/username that
                    the generated HTTPRoutes match on
                    (handle_request.go:83-231).
  ResponseHeaders — capture :status (handle_response.go:50-57).
  ResponseBody    — streamed mode: parse SSE chunks, usage is in the final
                    chunk with empty choices; buffered mode: accumulate
                    until end_of_stream then parse {usage}; account token
                    rate limits + quota usage (handle_response.go:80-268).
A 500 upstream status means ResponseBody never arrives — per-request
state dies with the stream (gateway.go:117-121).

Envoy is wired to this service by the EnvoyExtensionPolicy in
deploy/gateway/envoy/ (request.body Buffered, response.body Streamed —
reference config/gateway/gateway-plugin/gateway-plugin.yaml).
"""

from __future__ import annotations

import json
import logging
import time
import uuid
from concurrent import futures

from .extproc_pb import (
    CommonResponse,
    HeaderMutation,
    HeaderValue,
    ImmediateResponse,
    ProcessingRequest,
    ProcessingResponse,
)
from .limiter import RULES, RateLimiter, TYPE_REQUEST, TYPE_TOKEN
from .provider import ConfigProvider
from .quota import QuotaService
from .sse import SSEUsageScanner

log = logging.getLogger("arks.gateway.extproc")

METHOD = "/envoy.service.ext_proc.v3.ExternalProcessor/Process"


def _error_json(message: str, code: int) -> bytes:
    # reference util.go:40-77 generateErrorResponse shape
    return json.dumps({"error": {"message": message, "code": code}}).encode()


def _immediate(code: int, message: str,
               extra: dict[str, str] | None = None) -> ProcessingResponse:
    hm = HeaderMutation(
        set_headers=[HeaderValue(key="content-type", value="application/json")]
        + [HeaderValue(key=k, value=v) for k, v in (extra or {}).items()]
    )
    return ProcessingResponse(
        immediate_response=ImmediateResponse(
            status_code=code, headers=hm, body=_error_json(message, code)
        )
    )


class ExtProcServicer:
    """The per-stream state machine. Wire it to grpc via
    `register(server)` or run standalone via `serve()`."""

    def __init__(self, provider: ConfigProvider, limiter: RateLimiter,
                 quota_service: QuotaService, metrics=None):
        self.provider = provider
        self.limiter = limiter
        self.quota = quota_service
        self.metrics = metrics

    # ---- the gRPC bidi handler (one stream == one HTTP request) ----
    def process(self, request_iterator, context):
        request_id = uuid.uuid4().hex
        t0 = time.time()
        state: dict = {"buffer": b""}
        for msg in request_iterator:
            if msg.request_headers is not None:
                yield self._on_request_headers(msg.request_headers, state)
            elif msg.request_body is not None:
                yield self._on_request_body(msg.request_body, state)
            elif msg.response_headers is not None:
                state["status"] = msg.response_headers.get(":status") or "0"
                yield ProcessingResponse(response_headers=CommonResponse())
            elif msg.response_body is not None:
                yield self._on_response_body(msg.response_body, state)
            else:
                # trailers/attributes phases: continue untouched
                yield ProcessingResponse(request_headers=CommonResponse())
        self._record(state, time.time() - t0, request_id)

    def _record(self, state: dict, dur: float, request_id: str) -> None:
        qos = state.get("qos")
        if self.metrics is None or qos is None:
            return
        labels = dict(namespace=qos.namespace, user=qos.user,
                      model=state.get("model", ""))
        self.metrics.requests_total.labels(
            **labels, status=str(state.get("status", "0"))).inc()
        self.metrics.request_duration.labels(**labels).observe(dur)

    # ---- phases ----
    def _on_request_headers(self, headers, state) -> ProcessingResponse:
        auth = headers.get("authorization") or ""
        if not auth.lower().startswith("bearer "):
            return _immediate(401, "missing or malformed Authorization bearer token")
        state["token"] = auth[7:].strip()
        # reference handle_request.go:61-79: marker header + route-cache
        # clear so the body-phase header injection can re-route
        return ProcessingResponse(
            request_headers=CommonResponse(
                header_mutation=HeaderMutation(
                    set_headers=[HeaderValue(key="x-went-into-req-headers",
                                             value="true")]
                ),
                clear_route_cache=True,
            )
        )

    def _on_request_body(self, body, state) -> ProcessingResponse:
        token = state.get("token")
        if token is None:
            return _immediate(401, "missing bearer token")
        try:
            payload = json.loads(body.body or b"{}")
        except Exception:
            return _immediate(400, "invalid JSON body")
        model = payload.get("model", "")
        stream = bool(payload.get("stream", False))
        include_usage = bool(
            (payload.get("stream_options") or {}).get("include_usage", False)
        )
        qos = self.provider.get_qos_by_token(token, model)
        if qos is None:
            return _immediate(401, "invalid token or no QoS for this model")
        if model not in self.provider.get_model_list(qos.namespace):
            return _immediate(400, f"model {model!r} not available")
        if stream and not include_usage:
            return _immediate(
                400, "streaming requires stream_options.include_usage=true")

        descriptors = qos.limit_descriptors()
        ok, rule = self.limiter.check_limit(descriptors, request=1)
        if not ok:
            if self.metrics is not None:
                self.metrics.rate_limit_hits.labels(
                    namespace=qos.namespace, user=qos.user, model=model,
                    rule=rule).inc()
            return _immediate(429, f"rate limit exceeded: {rule}",
                              extra={"x-error-type": "rate-limit",
                                     "x-error-rule": str(rule)})
        qdesc = self.provider.get_quota_descriptors(qos)
        ok, qtype = self.quota.check(qdesc)
        if not ok:
            return _immediate(429, f"quota exceeded: {qtype}",
                              extra={"x-error-type": "quota",
                                     "x-error-rule": str(qtype)})
        self.limiter.do_limit(
            [d for d in descriptors if RULES[d.rule].type == TYPE_REQUEST], 1
        )
        state.update(qos=qos, model=model, stream=stream,
                     descriptors=descriptors)
        # inject the routing headers the HTTPRoute rules match on
        # (handle_request.go:208-231 / arksendpoint_controller.go:349-369)
        return ProcessingResponse(
            request_body=CommonResponse(
                header_mutation=HeaderMutation(
                    set_headers=[
                        HeaderValue(key="model", value=model),
                        HeaderValue(key="namespace", value=qos.namespace),
                        HeaderValue(key="username", value=qos.user),
                    ]
                ),
                clear_route_cache=True,
            )
        )

    def _on_response_body(self, body, state) -> ProcessingResponse:
        qos = state.get("qos")
        if qos is not None and state.get("status", "200") == "200":
            if state.get("stream"):
                self._scan_sse(body.body, state)
            else:
                state["buffer"] += body.body
                if body.end_of_stream:
                    try:
                        obj = json.loads(state["buffer"] or b"{}")
                        if obj.get("usage"):
                            self._account(obj["usage"], state)
                    except Exception:
                        log.warning("unparseable upstream response body")
        return ProcessingResponse(response_body=CommonResponse())

    def _scan_sse(self, chunk: bytes, state: dict) -> None:
        # shared spec-correct scanner (gateway/sse.py): CRLF endings and
        # arbitrary fragmentation tolerated — a naive split on blank lines
        # silently dropped usage from upstreams emitting CRLF
        sc = state.get("sse_scanner")
        if sc is None:
            sc = state["sse_scanner"] = SSEUsageScanner(
                lambda usage: self._account(usage, state))
        sc.feed(chunk)

    def _account(self, usage: dict, state: dict) -> None:
        qos = state["qos"]
        pt = int(usage.get("prompt_tokens", 0))
        ct = int(usage.get("completion_tokens", 0))
        tt = int(usage.get("total_tokens", pt + ct))
        self.limiter.do_limit(
            [d for d in state["descriptors"]
             if RULES[d.rule].type == TYPE_TOKEN], tt,
        )
        if qos.quota_name:
            self.quota.incr_usage(qos.namespace, qos.quota_name, "prompt", pt)
            self.quota.incr_usage(qos.namespace, qos.quota_name, "response", ct)
            self.quota.incr_usage(qos.namespace, qos.quota_name, "total", tt)
        if self.metrics is not None:
            labels = dict(namespace=qos.namespace, user=qos.user,
                          model=state.get("model", ""))
            self.metrics.token_usage.labels(**labels, type="input").inc(pt)
            self.metrics.token_usage.labels(**labels, type="output").inc(ct)

    # ---- grpc wiring ----
    def register(self, server) -> None:
        import grpc

        handler = grpc.stream_stream_rpc_method_handler(
            self.process,
            request_deserializer=ProcessingRequest.decode,
            response_serializer=lambda m: m.encode(),
        )
        service = grpc.method_handlers_generic_handler(
            "envoy.service.ext_proc.v3.ExternalProcessor",
            {"Process": handler},
        )
        server.add_generic_rpc_handlers((service,))


def serve(provider: ConfigProvider, limiter: RateLimiter,
          quota_service: QuotaService, port: int = 50052, metrics=None,
          max_workers: int = 64):
    """Start the ext_proc gRPC server (reference default port 50052,
    cmd/gateway/main.go:44-135). Returns the grpc.Server."""
    import grpc

    server = grpc.server(futures.ThreadPoolExecutor(max_workers=max_workers))
    ExtProcServicer(provider, limiter, quota_service, metrics).register(server)
    bound = server.add_insecure_port(f"[::]:{port}")
    server.start()
    server.bound_port = bound  # for callers binding an ephemeral port
    log.info("ext_proc gRPC server on :%d", bound)
    return server
