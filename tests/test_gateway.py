"""Gateway tests: limiter/quota semantics, and the full proxy pipeline in
front of the real engine server (ASGI-to-ASGI, CPU tiny model) — the
multi-tenant scenario of BASELINE.json config #5."""

import asyncio
import json

import httpx
import pytest

from arks_amd.config import EngineConfig
from arks_amd.controlplane import Operator, Store
from arks_amd.crd.types import parse_manifest
from arks_amd.gateway import (
    BackendResolver,
    LimitDescriptor,
    QuotaDescriptor,
    QuotaService,
    RateLimiter,
)
from arks_amd.gateway.app import create_gateway_app
from arks_amd.server.api import create_app
from arks_amd.server.async_engine import AsyncEngine
from arks_amd.server.tokenizer import ByteTokenizer


# ---------------- limiter unit tests ----------------
def test_fixed_window_limiter():
    now = [1000.0]
    rl = RateLimiter(clock=lambda: now[0])
    d = [LimitDescriptor("ns", "u", "m", "rpm", 2)]
    assert rl.check_limit(d)[0]
    rl.do_limit(d, 1)
    rl.do_limit(d, 1)
    ok, rule = rl.check_limit(d)
    assert not ok and rule == "rpm"
    now[0] += 61  # next minute window
    assert rl.check_limit(d)[0]


def test_token_rules_checked_at_zero_increment():
    now = [5000.0]
    rl = RateLimiter(clock=lambda: now[0])
    d = [LimitDescriptor("ns", "u", "m", "tpm", 100)]
    assert rl.check_limit(d)[0]
    rl.do_limit(d, 100)  # post-response accounting
    ok, rule = rl.check_limit(d)  # current(100) + 0 > 100 ? no
    assert ok
    rl.do_limit(d, 1)
    assert not rl.check_limit(d)[0]


def test_quota_cumulative_no_window():
    q = QuotaService()
    d = [QuotaDescriptor("ns", "qq", "total", 10)]
    assert q.check(d)[0]
    q.incr_usage("ns", "qq", "total", 10)
    assert q.check(d)[0]  # over only when current > limit
    q.incr_usage("ns", "qq", "total", 1)
    ok, t = q.check(d)
    assert not ok and t == "total"


# ---------------- full proxy pipeline ----------------
@pytest.fixture()
def stack():
    """Control-plane store with token/quota/endpoint + engine server app +
    gateway app wired via ASGI transports."""
    store = Store()
    op = Operator(store)
    for doc in [
        {
            "apiVersion": "arks.ai/v1",
            "kind": "ArksEndpoint",
            "metadata": {"name": "tiny", "namespace": "default"},
            "spec": {"defaultWeight": 1},
        },
        {
            "apiVersion": "arks.ai/v1",
            "kind": "ArksQuota",
            "metadata": {"name": "q1", "namespace": "default"},
            "spec": {"quotas": [{"type": "prompt", "value": 100000},
                                 {"type": "response", "value": 12},
                                 {"type": "total", "value": 100000}]},
        },
        {
            "apiVersion": "arks.ai/v1",
            "kind": "ArksToken",
            "metadata": {"name": "alice", "namespace": "default"},
            "spec": {
                "token": "sk-alice",
                "qos": [
                    {
                        "arksEndpoint": {"name": "tiny"},
                        "rateLimits": [
                            {"type": "rpm", "value": 4},
                            {"type": "tpm", "value": 1000},
                        ],
                        "quota": {"name": "q1"},
                    }
                ],
            },
        },
    ]:
        store.create(parse_manifest(doc))
    op.reconcile_until_stable()
    # a fake ready app so the endpoint routes somewhere
    route = store.get("HTTPRoute", "default", "tiny")
    route["spec"]["rules"][0]["backendRefs"] = [
        {"name": "arks-application-app1", "port": 8080, "weight": 1}
    ]
    store.update(route)

    cfg = EngineConfig(preset="tiny", device="cpu", kv_cache_blocks=256, max_model_len=512)
    engine = AsyncEngine(cfg, model_name="tiny")
    tok = ByteTokenizer(cfg.model_config().vocab_size, cfg.model_config().eos_token_id)
    server_app = create_app(engine, "tiny", tok)
    gw = create_gateway_app(
        store, transport=httpx.ASGITransport(app=server_app)
    )
    return store, server_app, gw


def run_stack(stack, fn):
    store, server_app, gw = stack

    async def go():
        async with server_app.router.lifespan_context(server_app):
            async with httpx.AsyncClient(
                transport=httpx.ASGITransport(app=gw), base_url="http://gw",
                timeout=60,
            ) as client:
                await fn(client)

    asyncio.new_event_loop().run_until_complete(go())


CHAT = {
    "model": "tiny",
    "messages": [{"role": "user", "content": "hello"}],
    "max_tokens": 3,
    "temperature": 0,
    "ignore_eos": True,
}


def test_auth_required(stack):
    async def fn(client):
        r = await client.post("/v1/chat/completions", json=CHAT)
        assert r.status_code == 401
        r = await client.post(
            "/v1/chat/completions", json=CHAT,
            headers={"Authorization": "Bearer sk-wrong"},
        )
        assert r.status_code == 401

    run_stack(stack, fn)


def test_proxied_completion_accounts_usage(stack):
    store, _, gw = stack

    async def fn(client):
        r = await client.post(
            "/v1/chat/completions", json=CHAT,
            headers={"Authorization": "Bearer sk-alice"},
        )
        assert r.status_code == 200, r.text
        usage = r.json()["usage"]
        assert usage["completion_tokens"] == 3
        qs = gw.state.quota_service
        assert qs.get_usage("default", "q1", "response") == 3
        assert qs.get_usage("default", "q1", "prompt") == usage["prompt_tokens"]
        assert qs.get_usage("default", "q1", "total") == usage["total_tokens"]

    run_stack(stack, fn)


def test_stream_requires_include_usage_and_accounts(stack):
    store, _, gw = stack

    async def fn(client):
        bad = dict(CHAT, stream=True)
        r = await client.post(
            "/v1/chat/completions", json=bad,
            headers={"Authorization": "Bearer sk-alice"},
        )
        assert r.status_code == 400

        good = dict(CHAT, stream=True, stream_options={"include_usage": True})
        async with client.stream(
            "POST", "/v1/chat/completions", json=good,
            headers={"Authorization": "Bearer sk-alice"},
        ) as resp:
            assert resp.status_code == 200
            lines = [l async for l in resp.aiter_lines() if l.startswith("data: ")]
        assert lines[-1] == "data: [DONE]"
        final = json.loads(lines[-2][6:])
        assert final["usage"]["completion_tokens"] == 3
        assert gw.state.quota_service.get_usage("default", "q1", "response") == 3

    run_stack(stack, fn)


def test_rpm_limit_enforced(stack):
    async def fn(client):
        h = {"Authorization": "Bearer sk-alice"}
        for i in range(4):
            r = await client.post("/v1/chat/completions", json=CHAT, headers=h)
            assert r.status_code == 200, (i, r.text)
        r = await client.post("/v1/chat/completions", json=CHAT, headers=h)
        assert r.status_code == 429
        assert "rpm" in r.json()["error"]["message"]

    run_stack(stack, fn)


def test_quota_exhaustion_blocks(stack):
    store, _, gw = stack

    async def fn(client):
        h = {"Authorization": "Bearer sk-alice"}
        # response quota is 12 -> 4 requests x 3 tokens hit it...
        codes = []
        for _ in range(4):
            r = await client.post("/v1/chat/completions", json=CHAT, headers=h)
            codes.append(r.status_code)
        assert codes == [200, 200, 200, 200]
        # usage now 12 (= limit, not over). One more passes precheck? current > limit is false.
        # rpm(4) also exhausted; bump window by using a fresh limiter state:
        gw.state.limiter.store._data.clear()
        r = await client.post("/v1/chat/completions", json=CHAT, headers=h)
        assert r.status_code == 200
        gw.state.limiter.store._data.clear()
        r = await client.post("/v1/chat/completions", json=CHAT, headers=h)
        assert r.status_code == 429  # 15 > 12
        assert "quota" in r.json()["error"]["message"]

    run_stack(stack, fn)


def test_models_endpoint_and_metrics(stack):
    async def fn(client):
        r = await client.get("/v1/models", headers={"Authorization": "Bearer sk-alice"})
        assert [m["id"] for m in r.json()["data"]] == ["tiny"]
        r = await client.post(
            "/v1/chat/completions", json=CHAT,
            headers={"Authorization": "Bearer sk-alice"},
        )
        assert r.status_code == 200
        m = await client.get("/metrics")
        assert "gateway_requests_total" in m.text
        assert "gateway_token_usage" in m.text

    run_stack(stack, fn)


def test_quota_sync_loop_updates_cr_status(stack):
    store, _, gw = stack

    async def fn(client):
        r = await client.post(
            "/v1/chat/completions", json=CHAT,
            headers={"Authorization": "Bearer sk-alice"},
        )
        assert r.status_code == 200
        gw.state.provider.sync_quota_usage()
        quota = store.get("ArksQuota", "default", "q1")
        used = {e.type: e.used for e in quota.status.quota_status}
        assert used["response"] == 3
        # crash recovery: wipe live counters, sync pushes CR values back
        gw.state.quota_service.store._usage.clear()
        gw.state.provider.sync_quota_usage()
        assert gw.state.quota_service.get_usage("default", "q1", "response") == 3

    run_stack(stack, fn)


def test_backend_resolver_multi_rule_matching():
    """Multi-rule HTTPRoutes: rules are evaluated with their path/header
    match conditions (user MatchConfigs), not just rules[0]."""
    from arks_amd.gateway import BackendResolver

    store = Store()
    store.apply({
        "apiVersion": "gateway.networking.k8s.io/v1",
        "kind": "HTTPRoute",
        "metadata": {"name": "m", "namespace": "default"},
        "spec": {"rules": [
            # static rule for a specific path prefix
            {"matches": [{"path": {"type": "PathPrefix",
                                   "value": "/v1/special"}}],
             "backendRefs": [{"name": "svc-special", "weight": 1}]},
            # dynamic rule matched on the injected routing headers
            {"matches": [{"path": {"type": "PathPrefix", "value": "/"},
                          "headers": [{"name": "namespace",
                                       "value": "default"},
                                      {"name": "model", "value": "m"}]}],
             "backendRefs": [{"name": "svc-app", "weight": 1}]},
        ]},
    })
    r = BackendResolver(store, url_for_service=lambda ns, svc: svc)
    assert r.resolve("default", "m", path="/v1/special/x") == "svc-special"
    assert r.resolve("default", "m", path="/v1/chat/completions") == "svc-app"
    # header mismatch on the dynamic rule -> no backend
    assert r.resolve("default", "m", path="/v1/chat/completions",
                     headers={"model": "other"}) is None


def test_quota_gauges_exported():
    """gateway_quota_usage/limit gauges (TODO stubs in the reference,
    collector.go:58-75) are populated by the quota sync loop."""
    from prometheus_client import generate_latest

    from arks_amd.gateway.app import create_gateway_app
    from arks_amd.crd.types import parse_manifest

    store = Store()
    store.apply(parse_manifest({
        "apiVersion": "arks.ai/v1", "kind": "ArksQuota",
        "metadata": {"name": "qg", "namespace": "default"},
        "spec": {"quotas": [{"type": "total", "value": 500}]},
    }))
    gw = create_gateway_app(store)
    gw.state.quota_service.incr_usage("default", "qg", "total", 42)
    gw.state.provider.sync_quota_usage()
    text = generate_latest(gw.state.provider.metrics.registry).decode()
    assert 'gateway_quota_usage{namespace="default",quota="qg",type="total"} 42.0' in text
    assert 'gateway_quota_limit{namespace="default",quota="qg",type="total"} 500.0' in text


def test_passive_outlier_ejection():
    """3 consecutive 5xx eject a backend for 30 s; traffic shifts to the
    remaining backend; ejection expires (reference BackendTrafficPolicy
    semantics for the no-Envoy mode)."""
    from arks_amd.gateway.app import BackendResolver, OutlierDetector

    now = [1000.0]
    det = OutlierDetector(threshold=3, ejection_s=30.0, clock=lambda: now[0])
    store = Store()
    store.apply({
        "apiVersion": "gateway.networking.k8s.io/v1",
        "kind": "HTTPRoute",
        "metadata": {"name": "m", "namespace": "default"},
        "spec": {"rules": [{"backendRefs": [
            {"name": "a", "weight": 1}, {"name": "b", "weight": 1},
        ]}]},
    })
    r = BackendResolver(store, url_for_service=lambda ns, svc: svc,
                        outliers=det)
    det.record("a", False)
    det.record("a", False)
    assert not det.is_ejected("a")  # below threshold
    det.record("a", True)           # success resets the streak
    det.record("a", False)
    det.record("a", False)
    det.record("a", False)
    assert det.is_ejected("a")
    picks = {r.resolve("default", "m") for _ in range(20)}
    assert picks == {"b"}
    now[0] += 31  # ejection expires
    assert not det.is_ejected("a")
    picks = {r.resolve("default", "m") for _ in range(50)}
    assert picks == {"a", "b"}
    # all ejected -> fall back to the full set (never 0 backends)
    det.record("a", False); det.record("a", False); det.record("a", False)
    det.record("b", False); det.record("b", False); det.record("b", False)
    assert r.resolve("default", "m") in {"a", "b"}


def test_backend_resolver_weight_proportional():
    """defaultWeight routing (reference HTTPRoute backendRef weights): a
    3:1 weighting splits traffic ~3:1."""
    import random

    from arks_amd.controlplane import Store
    from arks_amd.gateway import BackendResolver

    random.seed(7)
    store = Store()
    store.apply({
        "apiVersion": "gateway.networking.k8s.io/v1",
        "kind": "HTTPRoute",
        "metadata": {"name": "m", "namespace": "default"},
        "spec": {"rules": [{"backendRefs": [
            {"name": "heavy", "weight": 3}, {"name": "light", "weight": 1},
        ]}]},
    })
    r = BackendResolver(store, url_for_service=lambda ns, svc: svc)
    n = 4000
    heavy = sum(r.resolve("default", "m") == "heavy" for _ in range(n))
    assert 0.70 < heavy / n < 0.80, f"heavy fraction {heavy / n}"
    # zero-weight backends never picked
    store.apply({
        "apiVersion": "gateway.networking.k8s.io/v1",
        "kind": "HTTPRoute",
        "metadata": {"name": "z", "namespace": "default"},
        "spec": {"rules": [{"backendRefs": [
            {"name": "on", "weight": 1}, {"name": "off", "weight": 0},
        ]}]},
    })
    assert {r.resolve("default", "z") for _ in range(50)} == {"on"}


def test_backend_resolver_method_matching():
    """gateway-api method matching: a rule restricted to GET must not
    capture POST traffic (reference MatchConfigs carry `method`)."""
    from arks_amd.controlplane import Store
    from arks_amd.gateway import BackendResolver

    store = Store()
    store.apply({
        "apiVersion": "gateway.networking.k8s.io/v1",
        "kind": "HTTPRoute",
        "metadata": {"name": "m", "namespace": "default"},
        "spec": {"rules": [
            {"matches": [{"method": "GET",
                          "path": {"type": "PathPrefix", "value": "/"}}],
             "backendRefs": [{"name": "get-only", "weight": 1}]},
            {"matches": [{"method": "POST",
                          "path": {"type": "PathPrefix", "value": "/"}}],
             "backendRefs": [{"name": "post-svc", "weight": 1}]},
        ]},
    })
    r = BackendResolver(store, url_for_service=lambda ns, svc: svc)
    assert r.resolve("default", "m", method="POST") == "post-svc"
    assert r.resolve("default", "m", method="GET") == "get-only"
    # method omitted by the caller: first rule wins (match-all fallback)
    assert r.resolve("default", "m") == "get-only"
