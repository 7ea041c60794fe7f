"""In-memory object store — the controllers' state backend.

Plays the role of the K8s API server for reconciler logic: typed CRs and
generated workload manifests (dicts) live here, keyed (kind, namespace,
name), with resourceVersion bumps and deletion timestamps + finalizer
semantics. Unit tests drive reconcilers against it directly; a standalone
(non-k8s) deployment uses it as the live state; an httpx-backed real
API-server client can implement the same interface later.
"""

from __future__ import annotations

import threading
from typing import Any, Callable


class Conflict(Exception):
    pass


class NotFound(Exception):
    pass


def _meta(obj: Any) -> Any:
    return obj["metadata"] if isinstance(obj, dict) else obj.metadata


def _key(kind: str, namespace: str, name: str) -> tuple:
    return (kind, namespace, name)


def obj_kind(obj: Any) -> str:
    return obj["kind"] if isinstance(obj, dict) else obj.kind


class Store:
    def __init__(self):
        self._lock = threading.RLock()
        self._objects: dict[tuple, Any] = {}
        self._watchers: list[Callable[[str, Any], None]] = []

    # -- events --
    def subscribe(self, fn: Callable[[str, Any], None]) -> None:
        self._watchers.append(fn)

    def _notify(self, event: str, obj: Any) -> None:
        for fn in list(self._watchers):
            fn(event, obj)

    # -- CRUD --
    def create(self, obj: Any) -> Any:
        with self._lock:
            m = _meta(obj)
            k = _key(obj_kind(obj), self._ns(m), self._name(m))
            if k in self._objects:
                raise Conflict(f"{k} exists")
            self._bump(m)
            self._objects[k] = obj
        self._notify("ADDED", obj)
        return obj

    def get(self, kind: str, namespace: str, name: str) -> Any:
        with self._lock:
            obj = self._objects.get(_key(kind, namespace, name))
            if obj is None:
                raise NotFound(f"{kind}/{namespace}/{name}")
            return obj

    def get_opt(self, kind: str, namespace: str, name: str) -> Any | None:
        try:
            return self.get(kind, namespace, name)
        except NotFound:
            return None

    def update(self, obj: Any) -> Any:
        with self._lock:
            m = _meta(obj)
            k = _key(obj_kind(obj), self._ns(m), self._name(m))
            if k not in self._objects:
                raise NotFound(str(k))
            self._bump(m)
            self._objects[k] = obj
        self._notify("MODIFIED", obj)
        return obj

    def apply(self, obj: Any) -> Any:
        """create-or-update (server-side-apply-ish)."""
        with self._lock:
            m = _meta(obj)
            k = _key(obj_kind(obj), self._ns(m), self._name(m))
            exists = k in self._objects
        return self.update(obj) if exists else self.create(obj)

    def mark_deleted(self, kind: str, namespace: str, name: str) -> Any | None:
        """Set deletionTimestamp (finalizer-gated delete, like K8s)."""
        import time

        with self._lock:
            obj = self.get_opt(kind, namespace, name)
            if obj is None:
                return None
            m = _meta(obj)
            if isinstance(m, dict):
                m["deletionTimestamp"] = m.get("deletionTimestamp") or time.time()
                fins = m.get("finalizers") or []
            else:
                m.deletion_timestamp = m.deletion_timestamp or time.time()
                fins = m.finalizers
            if not fins:
                del self._objects[_key(kind, namespace, name)]
                self._notify("DELETED", obj)
                return None
        self._notify("MODIFIED", obj)
        return obj

    def finalize(self, obj: Any) -> None:
        """Remove from store once finalizers are gone and deletion pending."""
        with self._lock:
            m = _meta(obj)
            k = _key(obj_kind(obj), self._ns(m), self._name(m))
            self._objects.pop(k, None)
        self._notify("DELETED", obj)

    def delete(self, kind: str, namespace: str, name: str) -> None:
        with self._lock:
            obj = self._objects.pop(_key(kind, namespace, name), None)
        if obj is not None:
            self._notify("DELETED", obj)

    def list(self, kind: str, namespace: str | None = None) -> list:
        with self._lock:
            return [
                o
                for (k, ns, _), o in self._objects.items()
                if k == kind and (namespace is None or ns == namespace)
            ]

    # -- helpers --
    @staticmethod
    def _ns(m) -> str:
        return m.get("namespace", "default") if isinstance(m, dict) else m.namespace

    @staticmethod
    def _name(m) -> str:
        return m.get("name", "") if isinstance(m, dict) else m.name

    @staticmethod
    def _bump(m) -> None:
        if isinstance(m, dict):
            m["resourceVersion"] = int(m.get("resourceVersion", 0)) + 1
        else:
            m.resource_version += 1
