"""Router k8s pod discovery against a fake API server (httpx MockTransport)."""

import asyncio

import httpx

from arks_amd.router.app import RouterState
from arks_amd.router.discovery import KubePodDiscovery


def _pod(name, ip, ready=True, labels=None):
    return {
        "metadata": {"name": name, "labels": labels or {}},
        "status": {
            "podIP": ip,
            "conditions": [{"type": "Ready",
                            "status": "True" if ready else "False"}],
        },
    }


def test_discovery_updates_worker_lists():
    pods = {
        "role=prefill": [_pod("p1", "10.0.0.1"), _pod("p2", "10.0.0.2", ready=False)],
        "role=decode": [_pod("d1", "10.0.0.3")],
    }

    def handler(request: httpx.Request) -> httpx.Response:
        sel = request.url.params.get("labelSelector")
        return httpx.Response(200, json={"items": pods.get(sel, [])})

    disc = KubePodDiscovery(
        namespace="ns", prefill_selector="role=prefill",
        decode_selector="role=decode", port=8080,
        api_base="https://fake", interval_s=0.01,
        transport=httpx.MockTransport(handler),
    )
    state = RouterState([], [])

    async def go():
        task = asyncio.get_running_loop().create_task(disc.run(state))
        for _ in range(100):
            await asyncio.sleep(0.01)
            if state.prefill_urls:
                break
        disc.stop()
        await task

    asyncio.new_event_loop().run_until_complete(go())
    # only READY pods are registered
    assert state.prefill_urls == ["http://10.0.0.1:8080"]
    assert state.decode_urls == ["http://10.0.0.3:8080"]


def test_least_loaded_decode_policy():
    from arks_amd.router.app import RouterState

    st = RouterState([], ["http://d1", "http://d2", "http://d3"],
                     policy="cache_aware")
    # no load: rotation spreads picks
    picks = {st.pick_decode() for _ in range(3)}
    assert len(picks) == 3
    # d1 and d2 busy -> d3 wins regardless of rotation
    st.acquire("http://d1")
    st.acquire("http://d1")
    st.acquire("http://d2")
    assert all(st.pick_decode() == "http://d3" for _ in range(4))
    # release evens things out again
    st.release("http://d1")
    st.release("http://d1")
    st.release("http://d2")
    assert len({st.pick_decode() for _ in range(3)}) == 3
    # round_robin ignores load
    rr = RouterState([], ["http://a", "http://b"], policy="round_robin")
    rr.acquire("http://a")
    assert {rr.pick_decode(), rr.pick_decode()} == {"http://a", "http://b"}
