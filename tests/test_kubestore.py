"""KubeStore against a fake Kubernetes REST API (httpx MockTransport):
CRUD round-trip through the pydantic types, status subresource writes, and
the resync diff loop."""

import json

import httpx

from arks_amd.controlplane.kubestore import KubeStore
from arks_amd.crd.types import ArksModel, ArksModelSpec, ModelPhase, ObjectMeta


class FakeKube:
    """Minimal namespaced REST server over a dict."""

    def __init__(self):
        self.objects: dict[str, dict] = {}
        self.rv = 0

    def handler(self, request: httpx.Request) -> httpx.Response:
        path = request.url.path
        method = request.method
        status_sub = path.endswith("/status")
        if status_sub:
            path = path[: -len("/status")]
        parts = [p for p in path.split("/") if p]
        # .../namespaces/<ns>/<plural>[/<name>]
        if "namespaces" in parts:
            i = parts.index("namespaces")
            ns, plural = parts[i + 1], parts[i + 2]
            name = parts[i + 3] if len(parts) > i + 3 else None
        else:
            ns, plural, name = None, parts[-1], None
        if method == "GET" and name is None:
            items = [o for k, o in self.objects.items()
                     if k.startswith(f"{plural}/") and
                     (ns is None or o["metadata"]["namespace"] == ns)]
            return httpx.Response(200, json={"items": items})
        key = f"{plural}/{ns}/{name}"
        if method == "GET":
            if key not in self.objects:
                return httpx.Response(404, json={})
            return httpx.Response(200, json=self.objects[key])
        if method == "POST":
            body = json.loads(request.content)
            name = body["metadata"]["name"]
            key = f"{plural}/{ns}/{name}"
            if key in self.objects:
                return httpx.Response(409, json={})
            self.rv += 1
            body["metadata"]["resourceVersion"] = str(self.rv)
            body["metadata"].setdefault("namespace", ns)
            self.objects[key] = body
            return httpx.Response(201, json=body)
        if method == "PUT":
            if key not in self.objects:
                return httpx.Response(404, json={})
            body = json.loads(request.content)
            self.rv += 1
            body["metadata"]["resourceVersion"] = str(self.rv)
            if status_sub:
                cur = dict(self.objects[key])
                cur["status"] = body.get("status")
                cur["metadata"]["resourceVersion"] = str(self.rv)
                self.objects[key] = cur
                return httpx.Response(200, json=cur)
            body.setdefault("status", self.objects[key].get("status"))
            self.objects[key] = body
            return httpx.Response(200, json=body)
        if method == "DELETE":
            if self.objects.pop(key, None) is None:
                return httpx.Response(404, json={})
            return httpx.Response(200, json={})
        return httpx.Response(405)


def mk_store():
    fake = FakeKube()
    transport = httpx.MockTransport(fake.handler)
    store = KubeStore(api_base="https://fake", token="t", verify=False,
                      transport=transport)
    return fake, store


def test_crud_roundtrip_pydantic():
    fake, store = mk_store()
    m = ArksModel(
        metadata=ObjectMeta(name="qwen", namespace="ns1"),
        spec=ArksModelSpec(model="Qwen/Qwen2.5-7B-Instruct"),
    )
    store.create(m)
    got = store.get("ArksModel", "ns1", "qwen")
    assert isinstance(got, ArksModel)
    assert got.spec.model == "Qwen/Qwen2.5-7B-Instruct"
    assert got.metadata.resource_version == 1

    got.status.phase = ModelPhase.READY
    store.update(got)
    again = store.get("ArksModel", "ns1", "qwen")
    assert again.status.phase == ModelPhase.READY
    assert store.list("ArksModel", "ns1")[0].metadata.name == "qwen"

    store.delete("ArksModel", "ns1", "qwen")
    assert store.get_opt("ArksModel", "ns1", "qwen") is None


def test_dict_kinds_roundtrip():
    fake, store = mk_store()
    pod = {"apiVersion": "v1", "kind": "Pod",
           "metadata": {"name": "p1", "namespace": "ns1"},
           "spec": {"containers": []}}
    store.create(pod)
    got = store.get("Pod", "ns1", "p1")
    assert isinstance(got, dict) and got["metadata"]["name"] == "p1"
    # apply = update-or-create
    pod["spec"]["restartPolicy"] = "Never"
    store.apply(pod)
    assert store.get("Pod", "ns1", "p1")["spec"]["restartPolicy"] == "Never"


def test_resync_events():
    fake, store = mk_store()
    events = []
    store.subscribe(lambda e, o: events.append((e, o)))
    m = ArksModel(metadata=ObjectMeta(name="m1", namespace="d"),
                  spec=ArksModelSpec(model="x/y"))
    store.create(m)
    store.resync_once()
    assert any(e == "ADDED" for e, _ in events)
    events.clear()
    store.resync_once()
    assert events == []  # no changes -> no events
    got = store.get("ArksModel", "d", "m1")
    got.status.phase = ModelPhase.READY
    store.update(got)
    store.resync_once()
    assert any(e == "MODIFIED" for e, _ in events)
    events.clear()
    store.delete("ArksModel", "d", "m1")
    store.resync_once()
    assert any(e == "DELETED" for e, _ in events)
