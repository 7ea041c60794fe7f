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
