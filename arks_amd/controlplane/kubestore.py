"""Real-cluster Store backend: the same CRUD surface as
arks_amd.controlplane.store.Store, implemented against the Kubernetes API
server over REST (httpx, serviceaccount auth — no client library needed).

This is what makes the operator deployable (reference cmd/main.go runs a
controller-runtime manager; here Operator + KubeStore fill that role):
  * arks.ai kinds round-trip through the pydantic types (arks_amd.crd.types)
    so reconcilers see the same objects as with the in-memory store;
  * workload kinds (Pod/Service/PVC/Deployment/LWS/RBGS/HTTPRoute) stay raw
    dicts;
  * status is written through the /status subresource when present;
  * events come from real watch streams (run_watch: one list to seed,
    then long-lived `?watch=1` requests per kind, relist on 410 Gone),
    with the list+diff resync loop kept as a low-frequency safety net.
"""

from __future__ import annotations

import os
import threading
import time
from typing import Any, Callable

import httpx

from ..crd import types as T
from .store import Conflict, NotFound

SA_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"

# kind -> (api prefix, plural, namespaced)
API_MAP: dict[str, tuple[str, str]] = {
    "Pod": ("/api/v1", "pods"),
    "Service": ("/api/v1", "services"),
    "PersistentVolumeClaim": ("/api/v1", "persistentvolumeclaims"),
    "Secret": ("/api/v1", "secrets"),
    "Deployment": ("/apis/apps/v1", "deployments"),
    "ServiceAccount": ("/api/v1", "serviceaccounts"),
    "Role": ("/apis/rbac.authorization.k8s.io/v1", "roles"),
    "RoleBinding": ("/apis/rbac.authorization.k8s.io/v1", "rolebindings"),
    "LeaderWorkerSet": ("/apis/leaderworkerset.x-k8s.io/v1", "leaderworkersets"),
    "RoleBasedGroupSet": ("/apis/workloads.x-k8s.io/v1alpha1", "rolebasedgroupsets"),
    "HTTPRoute": ("/apis/gateway.networking.k8s.io/v1", "httproutes"),
    "Lease": ("/apis/coordination.k8s.io/v1", "leases"),
    "ArksModel": ("/apis/arks.ai/v1", "arksmodels"),
    "ArksApplication": ("/apis/arks.ai/v1", "arksapplications"),
    "ArksDisaggregatedApplication": ("/apis/arks.ai/v1",
                                     "arksdisaggregatedapplications"),
    "ArksEndpoint": ("/apis/arks.ai/v1", "arksendpoints"),
    "ArksToken": ("/apis/arks.ai/v1", "arkstokens"),
    "ArksQuota": ("/apis/arks.ai/v1", "arksquotas"),
}

ARKS_TYPES: dict[str, type] = {
    "ArksModel": T.ArksModel,
    "ArksApplication": T.ArksApplication,
    "ArksDisaggregatedApplication": T.ArksDisaggregatedApplication,
    "ArksEndpoint": T.ArksEndpoint,
    "ArksToken": T.ArksToken,
    "ArksQuota": T.ArksQuota,
}

WATCHED_KINDS = [
    "ArksModel", "ArksApplication", "ArksDisaggregatedApplication",
    "ArksEndpoint", "Pod", "LeaderWorkerSet", "RoleBasedGroupSet",
    "Deployment",
]


def _to_dict(obj: Any) -> dict:
    if isinstance(obj, dict):
        return obj
    d = obj.model_dump(by_alias=True, exclude_none=True)
    kind = type(obj).__name__
    d["apiVersion"] = "arks.ai/v1"
    d["kind"] = kind
    meta = d.get("metadata", {})
    # internal metadata back to K8s shape
    rv = meta.pop("resourceVersion", 0)
    if rv:
        meta["resourceVersion"] = str(rv)
    meta.pop("deletionTimestamp", None)
    d["metadata"] = meta
    return d


def _from_dict(kind: str, d: dict) -> Any:
    cls = ARKS_TYPES.get(kind)
    if cls is None:
        return d
    meta = dict(d.get("metadata", {}))
    rv = meta.get("resourceVersion")
    if isinstance(rv, str) and rv.isdigit():
        meta["resourceVersion"] = int(rv)
    elif not isinstance(rv, int):
        meta.pop("resourceVersion", None)
    dt = meta.get("deletionTimestamp")
    if isinstance(dt, str):
        meta["deletionTimestamp"] = time.time()  # presence is what matters
    obj = cls.model_validate({**d, "metadata": meta})
    return obj


class KubeStore:
    """Store-compatible CRUD against a real API server."""

    def __init__(self, api_base: str | None = None, token: str | None = None,
                 verify: Any = None, transport=None, timeout: float = 15.0):
        host = os.environ.get("KUBERNETES_SERVICE_HOST", "kubernetes.default.svc")
        port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
        base = api_base or f"https://{host}:{port}"
        headers = {"Content-Type": "application/json"}
        token_path = os.path.join(SA_DIR, "token")
        if token is None and os.path.exists(token_path):
            with open(token_path) as f:
                token = f.read().strip()
        if token:
            headers["Authorization"] = f"Bearer {token}"
        if verify is None:
            ca = os.path.join(SA_DIR, "ca.crt")
            verify = ca if os.path.exists(ca) else False
        self._client = httpx.Client(base_url=base, headers=headers,
                                    verify=verify, transport=transport,
                                    timeout=timeout)
        self._watchers: list[Callable[[str, Any], None]] = []
        self._stop = threading.Event()
        self._known: dict[tuple, str] = {}  # (kind, ns, name) -> resourceVersion

    # -- paths --
    @staticmethod
    def _path(kind: str, namespace: str, name: str | None = None) -> str:
        prefix, plural = API_MAP[kind]
        p = f"{prefix}/namespaces/{namespace}/{plural}"
        return f"{p}/{name}" if name else p

    # -- events --
    def subscribe(self, fn: Callable[[str, Any], None]) -> None:
        self._watchers.append(fn)

    def _notify(self, event: str, obj: Any) -> None:
        for fn in list(self._watchers):
            fn(event, obj)

    # -- CRUD (Store interface) --
    def create(self, obj: Any) -> Any:
        d = _to_dict(obj)
        kind = d["kind"]
        ns = d["metadata"].get("namespace", "default")
        r = self._client.post(self._path(kind, ns), json=d)
        if r.status_code == 409:
            raise Conflict(r.text)
        r.raise_for_status()
        return _from_dict(kind, r.json())

    def get(self, kind: str, namespace: str, name: str) -> Any:
        r = self._client.get(self._path(kind, namespace, name))
        if r.status_code == 404:
            raise NotFound(f"{kind}/{namespace}/{name}")
        r.raise_for_status()
        return _from_dict(kind, r.json())

    def get_opt(self, kind: str, namespace: str, name: str) -> Any | None:
        try:
            return self.get(kind, namespace, name)
        except NotFound:
            return None

    def update(self, obj: Any) -> Any:
        d = _to_dict(obj)
        kind = d["kind"]
        meta = d["metadata"]
        ns = meta.get("namespace", "default")
        name = meta["name"]
        path = self._path(kind, ns, name)
        status = d.pop("status", None)
        # refresh-and-retry on 409 (the reference's RetryOnConflict:
        # reconcilers are level-triggered and idempotent, so re-reading the
        # live resourceVersion and re-applying our mutation is safe)
        for attempt in range(5):
            live = self._client.get(path)
            if live.status_code == 404:
                raise NotFound(f"{kind}/{ns}/{name}")
            live.raise_for_status()
            live_obj = live.json()
            meta["resourceVersion"] = live_obj["metadata"]["resourceVersion"]
            r = self._client.put(path, json=d)
            if r.status_code == 409:
                if attempt == 4:
                    raise Conflict(r.text)
                continue
            r.raise_for_status()
            break
        out = r.json()
        if status is not None:
            sub = dict(out)
            sub["status"] = status
            rs = self._client.put(f"{path}/status", json=sub)
            if rs.status_code not in (404, 405):  # kinds without status sub
                rs.raise_for_status()
                out = rs.json()
        return _from_dict(kind, out)

    def apply(self, obj: Any) -> Any:
        try:
            return self.update(obj)
        except NotFound:
            return self.create(obj)

    def delete(self, kind: str, namespace: str, name: str) -> None:
        r = self._client.delete(self._path(kind, namespace, name))
        if r.status_code not in (200, 202, 404):
            r.raise_for_status()

    def finalize(self, obj: Any) -> None:
        """Called by reconcilers after removing the last finalizer: the
        update() above already persisted the empty finalizer list, so the
        API server completes the deletion itself."""

    def mark_deleted(self, kind: str, namespace: str, name: str) -> Any | None:
        self.delete(kind, namespace, name)
        return self.get_opt(kind, namespace, name)

    def list(self, kind: str, namespace: str | None = None) -> list:
        if namespace is not None:
            r = self._client.get(self._path(kind, namespace))
            r.raise_for_status()
            return [_from_dict(kind, i) for i in r.json().get("items", [])]
        prefix, plural = API_MAP[kind]
        r = self._client.get(f"{prefix}/{plural}")
        r.raise_for_status()
        return [_from_dict(kind, i) for i in r.json().get("items", [])]

    # -- resync loop (drives Operator's queue like informers would) --
    def resync_once(self) -> None:
        seen: set[tuple] = set()
        for kind in WATCHED_KINDS:
            try:
                items = self.list(kind)
            except Exception:
                continue
            for obj in items:
                m = obj["metadata"] if isinstance(obj, dict) else obj.metadata
                ns = (m.get("namespace", "default") if isinstance(m, dict)
                      else m.namespace)
                name = m.get("name", "") if isinstance(m, dict) else m.name
                rv = str(m.get("resourceVersion", "") if isinstance(m, dict)
                         else m.resource_version)
                key = (kind, ns, name)
                seen.add(key)
                if self._known.get(key) != rv:
                    event = "MODIFIED" if key in self._known else "ADDED"
                    self._known[key] = rv
                    self._notify(event, obj)
        for key in [k for k in self._known if k not in seen]:
            kind, ns, name = key
            del self._known[key]
            self._notify("DELETED", {"kind": kind,
                                     "metadata": {"namespace": ns, "name": name}})

    def run_resync(self, interval_s: float = 5.0) -> None:
        while not self._stop.is_set():
            self.resync_once()
            self._stop.wait(interval_s)

    # -- watch streams (the informer path; the resync loop above is the
    #    fallback for servers without watch support) --
    def _list_raw(self, kind: str) -> tuple[list[dict], str]:
        """List a kind cluster-wide, returning (items, list resourceVersion)."""
        prefix, plural = API_MAP[kind]
        r = self._client.get(f"{prefix}/{plural}")
        r.raise_for_status()
        body = r.json()
        return (body.get("items", []),
                str(body.get("metadata", {}).get("resourceVersion", "")))

    def _seed_kind(self, kind: str) -> str:
        """Initial list: emit ADDED/MODIFIED/DELETED diffs vs the known
        state, return the collection resourceVersion to watch from."""
        items, rv = self._list_raw(kind)
        seen = set()
        for raw in items:
            m = raw.get("metadata", {})
            key = (kind, m.get("namespace", "default"), m.get("name", ""))
            seen.add(key)
            orv = str(m.get("resourceVersion", ""))
            if self._known.get(key) != orv:
                event = "MODIFIED" if key in self._known else "ADDED"
                self._known[key] = orv
                self._notify(event, _from_dict(kind, raw))
        for key in [k for k in self._known
                    if k[0] == kind and k not in seen]:
            _, ns, name = key
            del self._known[key]
            self._notify("DELETED", {"kind": kind,
                                     "metadata": {"namespace": ns,
                                                  "name": name}})
        return rv

    def _watch_kind_once(self, kind: str, rv: str,
                         timeout_s: float = 300.0) -> str | None:
        """One watch request from resourceVersion `rv`; streams
        {type, object} JSON lines and feeds the event bus. Returns the
        last seen resourceVersion, or None on 410 Gone / error (caller
        must relist)."""
        import json as _json

        prefix, plural = API_MAP[kind]
        url = (f"{prefix}/{plural}?watch=1&allowWatchBookmarks=true"
               f"&timeoutSeconds={int(timeout_s)}&resourceVersion={rv}")
        try:
            with self._client.stream("GET", url, timeout=timeout_s + 15) as r:
                if r.status_code != 200:
                    return None
                for line in r.iter_lines():
                    if self._stop.is_set():
                        return rv
                    if not line.strip():
                        continue
                    try:
                        ev = _json.loads(line)
                    except ValueError:
                        return None
                    etype = ev.get("type", "")
                    obj = ev.get("object", {})
                    orv = str(obj.get("metadata", {})
                              .get("resourceVersion", rv))
                    if etype == "BOOKMARK":
                        rv = orv
                        continue
                    if etype == "ERROR":
                        return None  # usually 410 Gone: relist
                    m = obj.get("metadata", {})
                    key = (kind, m.get("namespace", "default"),
                           m.get("name", ""))
                    rv = orv
                    if etype == "DELETED":
                        self._known.pop(key, None)
                        self._notify("DELETED", obj)
                    elif etype in ("ADDED", "MODIFIED"):
                        self._known[key] = orv
                        self._notify(etype, _from_dict(kind, obj))
        except Exception:
            return None
        return rv

    def _watch_loop(self, kind: str) -> None:
        while not self._stop.is_set():
            try:
                rv = self._seed_kind(kind)
            except Exception:
                self._stop.wait(2.0)
                continue
            while not self._stop.is_set():
                rv2 = self._watch_kind_once(kind, rv)
                if rv2 is None:
                    break  # relist
                rv = rv2

    def run_watch(self) -> list[threading.Thread]:
        """Start one watch thread per watched kind (the controller-runtime
        informer equivalent — no polling between events). Threads are
        daemons; stop() ends them at their next event/timeout."""
        threads = []
        for kind in WATCHED_KINDS:
            t = threading.Thread(target=self._watch_loop, args=(kind,),
                                 daemon=True, name=f"watch-{kind}")
            t.start()
            threads.append(t)
        return threads

    def stop(self) -> None:
        self._stop.set()
