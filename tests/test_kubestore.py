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
        self.conflicts_to_inject = 0  # chaos: next N PUTs 409 regardless
        # watch support: (rv, plural, event-dict) log
        self.events: list[tuple[int, str, dict]] = []

    def _record(self, etype: str, plural: str, obj: dict) -> None:
        self.events.append((self.rv, plural, {"type": etype, "object": obj}))

    def handler(self, request: httpx.Request) -> httpx.Response:
        path = request.url.path
        method = request.method
        query = dict(
            kv.split("=", 1) for kv in str(request.url.query, "utf-8").split("&")
            if "=" in kv
        ) if request.url.query else {}
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
        if method == "GET" and name is None and query.get("watch") == "1":
            # one-shot watch: emit events after resourceVersion, then end
            # the stream (the client's watch loop re-issues)
            since = int(query.get("resourceVersion") or 0)
            evs = [e for (erv, epl, e) in self.events
                   if epl == plural and erv > since]
            body = "\n".join(json.dumps(e) for e in evs)
            return httpx.Response(200, text=body)
        if method == "GET" and name is None:
            items = [o for k, o in self.objects.items()
                     if k.startswith(f"{plural}/") and
                     (ns is None or o["metadata"]["namespace"] == ns)]
            return httpx.Response(
                200, json={"items": items,
                           "metadata": {"resourceVersion": str(self.rv)}})
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
            self._record("ADDED", plural, body)
            return httpx.Response(201, json=body)
        if method == "PUT":
            if key not in self.objects:
                return httpx.Response(404, json={})
            body = json.loads(request.content)
            # real API-server optimistic concurrency: a stale
            # resourceVersion (or an injected chaos conflict) is a 409
            sent_rv = body.get("metadata", {}).get("resourceVersion")
            cur_rv = self.objects[key]["metadata"].get("resourceVersion")
            if self.conflicts_to_inject > 0:
                self.conflicts_to_inject -= 1
                return httpx.Response(409, json={"reason": "Conflict"})
            if sent_rv is not None and sent_rv != cur_rv:
                return httpx.Response(409, json={"reason": "Conflict"})
            self.rv += 1
            body["metadata"]["resourceVersion"] = str(self.rv)
            if status_sub:
                cur = dict(self.objects[key])
                cur["status"] = body.get("status")
                cur["metadata"]["resourceVersion"] = str(self.rv)
                self.objects[key] = cur
                self._record("MODIFIED", plural, cur)
                return httpx.Response(200, json=cur)
            body.setdefault("status", self.objects[key].get("status"))
            self.objects[key] = body
            self._record("MODIFIED", plural, body)
            return httpx.Response(200, json=body)
        if method == "DELETE":
            gone = self.objects.pop(key, None)
            if gone is None:
                return httpx.Response(404, json={})
            self.rv += 1
            gone = dict(gone)
            gone["metadata"] = {**gone["metadata"],
                                "resourceVersion": str(self.rv)}
            self._record("DELETED", plural, gone)
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


def test_watch_stream_delivers_events_without_polling():
    """The watch path (informer equivalent): after the initial seed list,
    mutations arrive as ADDED/MODIFIED/DELETED watch events — no list+diff
    polling involved."""
    fake, store = mk_store()
    events = []
    store.subscribe(lambda e, o: events.append((e, getattr(
        o, "metadata", None) and o.metadata.name or o["metadata"]["name"])))

    m = ArksModel(
        metadata=ObjectMeta(name="m1", namespace="default"),
        spec=ArksModelSpec(model="org/m"),
    )
    store.create(m)
    # seed: list emits ADDED, returns the collection resourceVersion
    rv = store._seed_kind("ArksModel")
    assert ("ADDED", "m1") in events
    events.clear()

    # mutate: the next watch call must deliver MODIFIED without any list
    got = store.get("ArksModel", "default", "m1")
    got.status.phase = ModelPhase.READY
    store.update(got)
    rv2 = store._watch_kind_once("ArksModel", rv, timeout_s=1)
    assert rv2 is not None and int(rv2) > int(rv)
    assert ("MODIFIED", "m1") in events
    events.clear()

    store.delete("ArksModel", "default", "m1")
    rv3 = store._watch_kind_once("ArksModel", rv2, timeout_s=1)
    assert ("DELETED", "m1") in events
    assert rv3 is not None


def test_watch_seed_emits_deletes_for_vanished_objects():
    fake, store = mk_store()
    events = []
    store.subscribe(lambda e, o: events.append(e))
    m = ArksModel(
        metadata=ObjectMeta(name="mgone", namespace="default"),
        spec=ArksModelSpec(model="org/m"),
    )
    store.create(m)
    store._seed_kind("ArksModel")
    # delete behind the store's back (another client), reseed
    fake.objects.clear()
    store._seed_kind("ArksModel")
    assert "DELETED" in events


def test_watch_error_event_triggers_relist_and_recovers():
    """A watch ERROR event (the wire form of 410 Gone: compaction dropped
    our resourceVersion) makes _watch_kind_once return None; the caller
    relists via _seed_kind and the store converges on current state
    (reference informer semantics)."""
    fake, store = mk_store()
    events = []
    store.subscribe(lambda e, o: events.append((e, getattr(
        o, "metadata", None) and o.metadata.name or o["metadata"]["name"])))

    m = ArksModel(
        metadata=ObjectMeta(name="m1", namespace="default"),
        spec=ArksModelSpec(model="org/m"),
    )
    store.create(m)
    rv = store._seed_kind("ArksModel")
    events.clear()

    # inject an ERROR watch event at the head of the stream (410 Gone)
    fake.rv += 1
    fake._record("ERROR", "arksmodels", {
        "kind": "Status", "code": 410, "reason": "Gone",
        "metadata": {}})
    assert store._watch_kind_once("ArksModel", rv, timeout_s=1) is None

    # while we were "disconnected", another client created m2
    m2 = ArksModel(
        metadata=ObjectMeta(name="m2", namespace="default"),
        spec=ArksModelSpec(model="org/m2"),
    )
    store.create(m2)
    events.clear()
    rv2 = store._seed_kind("ArksModel")  # the relist
    assert rv2 is not None
    assert ("ADDED", "m2") in events


def test_update_retries_on_conflict():
    """Optimistic-concurrency semantics (real API server / envtest): a 409
    between the store's refresh and PUT is retried with a fresh
    resourceVersion (reference RetryOnConflict); persistent conflicts
    surface as Conflict."""
    import pytest as _pytest

    from arks_amd.controlplane.kubestore import Conflict

    fake, store = mk_store()
    m = ArksModel(
        metadata=ObjectMeta(name="c1", namespace="default"),
        spec=ArksModelSpec(model="org/m"),
    )
    store.create(m)
    got = store.get("ArksModel", "default", "c1")
    got.status.phase = ModelPhase.READY
    fake.conflicts_to_inject = 2  # two racing writers, then success
    out = store.update(got)
    assert out.status.phase == ModelPhase.READY
    # a conflict storm (more than the retry budget) surfaces
    got2 = store.get("ArksModel", "default", "c1")
    fake.conflicts_to_inject = 99
    with _pytest.raises(Conflict):
        store.update(got2)
    fake.conflicts_to_inject = 0


def test_stale_resource_version_put_rejected():
    """FakeKube enforces resourceVersion like a real API server: a raw PUT
    carrying a stale rv is rejected with 409 (this is what forces the
    store's refresh-and-retry path to exist)."""
    fake, store = mk_store()
    m = ArksModel(
        metadata=ObjectMeta(name="c2", namespace="default"),
        spec=ArksModelSpec(model="org/m"),
    )
    store.create(m)
    key = "arksmodels/default/c2"
    stale = dict(fake.objects[key])
    # another writer bumps the object
    cur = store.get("ArksModel", "default", "c2")
    store.update(cur)
    # raw PUT with the old rv must 409
    import httpx as _httpx
    req = _httpx.Request(
        "PUT",
        "https://fake/apis/arks.ai/v1/namespaces/default/arksmodels/c2",
        json=stale)
    resp = fake.handler(req)
    assert resp.status_code == 409
