"""Envoy ext_proc gRPC servicer tests: an in-process gRPC client drives
the four-phase state machine (RequestHeaders -> RequestBody ->
ResponseHeaders -> ResponseBody) against the real servicer + limiter +
quota + provider — the reference contract at pkg/gateway/gateway.go:77-138.
"""

import json

import grpc
import pytest

from arks_amd.controlplane import Store
from arks_amd.crd.types import parse_manifest
from arks_amd.gateway import RateLimiter, QuotaService
from arks_amd.gateway.extproc import METHOD, serve
from arks_amd.gateway.extproc_pb import (
    HeaderMap,
    HeaderValue,
    HttpBody,
    HttpHeaders,
    ProcessingRequest,
    ProcessingResponse,
)
from arks_amd.gateway.provider import ConfigProvider


TOKEN_YAML = """
apiVersion: arks.ai/v1
kind: ArksToken
metadata: {name: t1, namespace: default}
spec:
  token: sk-test-1
  qos:
    - arksEndpoint: {name: m1}
      rateLimits:
        - type: rpm
          value: 2
        - type: tpm
          value: 50
      quota: {name: q1}
"""

QUOTA_YAML = """
apiVersion: arks.ai/v1
kind: ArksQuota
metadata: {name: q1, namespace: default}
spec:
  quotas:
    - type: total
      value: 100
"""

ENDPOINT_YAML = """
apiVersion: arks.ai/v1
kind: ArksEndpoint
metadata: {name: m1, namespace: default}
spec: {}
"""


def mk_store():
    import yaml

    store = Store()
    for y in (TOKEN_YAML, QUOTA_YAML, ENDPOINT_YAML):
        store.apply(parse_manifest(yaml.safe_load(y)))
    return store


@pytest.fixture()
def stack():
    store = mk_store()
    limiter = RateLimiter()
    quota = QuotaService()
    provider = ConfigProvider(store, quota)
    server = serve(provider, limiter, quota, port=0)
    chan = grpc.insecure_channel(f"127.0.0.1:{server.bound_port}")
    yield chan, limiter, quota, provider
    chan.close()
    server.stop(0)


def _stream(chan):
    return chan.stream_stream(
        METHOD,
        request_serializer=lambda m: m.encode(),
        response_deserializer=ProcessingResponse.decode,
    )


def _headers_msg(pairs, eos=False):
    return ProcessingRequest(
        request_headers=HttpHeaders(
            headers=HeaderMap(
                headers=[HeaderValue(key=k, value=v) for k, v in pairs]
            ),
            end_of_stream=eos,
        )
    )


def _resp_headers_msg(status="200"):
    return ProcessingRequest(
        response_headers=HttpHeaders(
            headers=HeaderMap(headers=[HeaderValue(key=":status", value=status)])
        )
    )


def drive(chan, msgs):
    out = []
    it = _stream(chan)(iter(msgs))
    for r in it:
        out.append(r)
    return out


def test_missing_bearer_is_immediate_401(stack):
    chan, *_ = stack
    out = drive(chan, [_headers_msg([("host", "x")])])
    assert out[0].immediate_response is not None
    assert out[0].immediate_response.status_code == 401
    body = json.loads(out[0].immediate_response.body)
    assert body["error"]["code"] == 401


def test_full_request_flow_injects_routing_headers_and_accounts(stack):
    chan, limiter, quota, provider = stack
    body = json.dumps({"model": "m1", "stream": False}).encode()
    usage_body = json.dumps(
        {"model": "m1",
         "usage": {"prompt_tokens": 7, "completion_tokens": 5,
                   "total_tokens": 12}}
    ).encode()
    msgs = [
        _headers_msg([("authorization", "Bearer sk-test-1")]),
        ProcessingRequest(request_body=HttpBody(body=body, end_of_stream=True)),
        _resp_headers_msg("200"),
        ProcessingRequest(
            response_body=HttpBody(body=usage_body, end_of_stream=True)),
    ]
    out = drive(chan, msgs)
    assert len(out) == 4
    # phase 1: marker header + route-cache clear
    h1 = out[0].request_headers
    assert h1 is not None and h1.clear_route_cache
    keys1 = {h.key for h in h1.header_mutation.set_headers}
    assert "x-went-into-req-headers" in keys1
    # phase 2: routing header injection
    h2 = out[1].request_body
    injected = {h.key: h.value for h in h2.header_mutation.set_headers}
    assert injected == {"model": "m1", "namespace": "default",
                        "username": "t1"}
    # usage accounted into quota and tpm counters
    assert quota.get_usage("default", "q1", "total") == 12
    assert quota.get_usage("default", "q1", "prompt") == 7
    assert quota.get_usage("default", "q1", "response") == 5


def test_rate_limit_429_with_error_headers(stack):
    chan, *_ = stack
    body = json.dumps({"model": "m1"}).encode()

    def once():
        return drive(chan, [
            _headers_msg([("authorization", "Bearer sk-test-1")]),
            ProcessingRequest(
                request_body=HttpBody(body=body, end_of_stream=True)),
        ])

    once()
    once()
    out = once()  # rpm limit is 2
    imm = out[1].immediate_response
    assert imm is not None and imm.status_code == 429
    hdrs = {h.key: h.value for h in imm.headers.set_headers}
    assert hdrs.get("x-error-type") == "rate-limit"
    assert hdrs.get("x-error-rule") == "rpm"


def test_streamed_usage_final_chunk(stack):
    chan, limiter, quota, _ = stack
    body = json.dumps({"model": "m1", "stream": True,
                       "stream_options": {"include_usage": True}}).encode()
    chunk1 = b'data: {"choices":[{"delta":{"content":"hi"}}]}\n\n'
    final = (b'data: {"choices":[],"usage":{"prompt_tokens":3,'
             b'"completion_tokens":4,"total_tokens":7}}\n\n')
    done = b"data: [DONE]\n\n"
    msgs = [
        _headers_msg([("authorization", "Bearer sk-test-1")]),
        ProcessingRequest(request_body=HttpBody(body=body, end_of_stream=True)),
        _resp_headers_msg("200"),
        ProcessingRequest(response_body=HttpBody(body=chunk1)),
        ProcessingRequest(response_body=HttpBody(body=final)),
        ProcessingRequest(
            response_body=HttpBody(body=done, end_of_stream=True)),
    ]
    out = drive(chan, msgs)
    assert all(r.immediate_response is None for r in out)
    assert quota.get_usage("default", "q1", "total") == 7


def test_stream_without_include_usage_rejected(stack):
    chan, *_ = stack
    body = json.dumps({"model": "m1", "stream": True}).encode()
    out = drive(chan, [
        _headers_msg([("authorization", "Bearer sk-test-1")]),
        ProcessingRequest(request_body=HttpBody(body=body, end_of_stream=True)),
    ])
    imm = out[1].immediate_response
    assert imm is not None and imm.status_code == 400


def test_unknown_model_400_and_bad_token_401(stack):
    chan, *_ = stack
    out = drive(chan, [
        _headers_msg([("authorization", "Bearer sk-test-1")]),
        ProcessingRequest(request_body=HttpBody(
            body=json.dumps({"model": "nope"}).encode(), end_of_stream=True)),
    ])
    assert out[1].immediate_response.status_code == 401  # no qos for model
    out = drive(chan, [
        _headers_msg([("authorization", "Bearer sk-bogus")]),
        ProcessingRequest(request_body=HttpBody(
            body=json.dumps({"model": "m1"}).encode(), end_of_stream=True)),
    ])
    assert out[1].immediate_response.status_code == 401


def test_500_upstream_skips_accounting(stack):
    """Reference gateway.go:117-121: on a 5xx the ResponseBody phase never
    accounts usage."""
    chan, limiter, quota, _ = stack
    body = json.dumps({"model": "m1"}).encode()
    usage_body = json.dumps({"usage": {"total_tokens": 99}}).encode()
    drive(chan, [
        _headers_msg([("authorization", "Bearer sk-test-1")]),
        ProcessingRequest(request_body=HttpBody(body=body, end_of_stream=True)),
        _resp_headers_msg("500"),
        ProcessingRequest(
            response_body=HttpBody(body=usage_body, end_of_stream=True)),
    ])
    assert quota.get_usage("default", "q1", "total") == 0


def test_pb_roundtrip():
    """Wire-codec self-consistency for every message used on the stream."""
    req = ProcessingRequest(
        request_headers=HttpHeaders(
            headers=HeaderMap(headers=[
                HeaderValue(key="a", value="b"),
                HeaderValue(key="c", raw_value=b"\x00\xff"),
            ]),
            end_of_stream=True,
        )
    )
    back = ProcessingRequest.decode(req.encode())
    assert back.request_headers.get("a") == "b"
    assert back.request_headers.headers.headers[1].raw_value == b"\x00\xff"
    assert back.request_headers.end_of_stream
    body = ProcessingRequest(request_body=HttpBody(body=b"xyz"))
    assert ProcessingRequest.decode(body.encode()).request_body.body == b"xyz"


def test_streamed_usage_crlf_and_fragmented(stack):
    """SSE with CRLF line endings, a multi-line data field, and the usage
    chunk fragmented mid-line across two ext_proc body messages must still
    account (spec-correct SSE parsing, not naive split-on-\\n\\n)."""
    chan, limiter, quota, _ = stack
    body = json.dumps({"model": "m1", "stream": True,
                       "stream_options": {"include_usage": True}}).encode()
    chunk1 = b'data: {"choices":[{"delta":{"content":"hi"}}]}\r\n\r\n'
    final = (b'data: {"choices":[],"usage":{"prompt_tokens":5,\r\n'
             b'data: "completion_tokens":6,"total_tokens":11}}\r\n\r\n')
    # split the final event mid-line across two messages
    cut = 37
    msgs = [
        _headers_msg([("authorization", "Bearer sk-test-1")]),
        ProcessingRequest(request_body=HttpBody(body=body, end_of_stream=True)),
        _resp_headers_msg("200"),
        ProcessingRequest(response_body=HttpBody(body=chunk1)),
        ProcessingRequest(response_body=HttpBody(body=final[:cut])),
        ProcessingRequest(response_body=HttpBody(body=final[cut:])),
        ProcessingRequest(response_body=HttpBody(
            body=b"data: [DONE]\r\n\r\n", end_of_stream=True)),
    ]
    out = drive(chan, msgs)
    assert all(r.immediate_response is None for r in out)
    assert quota.get_usage("default", "q1", "total") == 11
