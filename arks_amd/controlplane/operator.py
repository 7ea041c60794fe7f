"""Operator: wires store events to the reconcilers (the controller-runtime
manager equivalent — reference cmd/main.go:255-301).

`reconcile_until_stable()` drains the work queue synchronously (tests,
batch); `run()` is the long-running threaded form (standalone deployments).
"""

from __future__ import annotations

import queue
import threading
import time

from ..crd.types import served_model_name
from .reconcilers import (
    ArksApplicationReconciler,
    ArksDisaggregatedApplicationReconciler,
    ArksEndpointReconciler,
    ArksModelReconciler,
)
from .store import Store, obj_kind
from . import metrics as opmetrics


class Operator:
    def __init__(self, store: Store):
        self.store = store
        self.model_rec = ArksModelReconciler(store)
        self.app_rec = ArksApplicationReconciler(store)
        self.endpoint_rec = ArksEndpointReconciler(store)
        self.disagg_rec = ArksDisaggregatedApplicationReconciler(store)
        self._queue: queue.Queue = queue.Queue()
        self._stop = threading.Event()
        store.subscribe(self._on_event)

    # --- event routing (reference Watches/Owns wiring) ---
    def _on_event(self, event: str, obj) -> None:
        kind = obj_kind(obj)
        m = obj["metadata"] if isinstance(obj, dict) else obj.metadata
        ns = m.get("namespace", "default") if isinstance(m, dict) else m.namespace
        name = m.get("name", "") if isinstance(m, dict) else m.name
        if kind in ("ArksModel", "ArksApplication", "ArksEndpoint",
                    "ArksDisaggregatedApplication"):
            self._queue.put((kind, ns, name))
        if kind == "ArksModel":
            # requeue apps gated on this model (reference requestsForModel :1063)
            for app in self.store.list("ArksApplication", ns):
                if app.spec.model.get("name") == name:
                    self._queue.put(("ArksApplication", ns, app.metadata.name))
            for dapp in self.store.list("ArksDisaggregatedApplication", ns):
                if dapp.spec.model.get("name") == name:
                    self._queue.put(("ArksDisaggregatedApplication", ns, dapp.metadata.name))
        if kind == "Pod" and name.startswith("arks-worker-"):
            self._queue.put(("ArksModel", ns, name[len("arks-worker-"):]))
        if kind in ("LeaderWorkerSet", "RoleBasedGroupSet", "Deployment"):
            # route by ownerReferences (reference field-index wiring,
            # arksapplication_controller.go:1091-1134) — name-splitting
            # mis-routes hyphenated app names
            owners = (
                m.get("ownerReferences", []) if isinstance(m, dict) else []
            )
            routed = False
            for ref in owners:
                okind = ref.get("kind")
                if okind in ("ArksApplication", "ArksDisaggregatedApplication"):
                    self._queue.put((okind, ns, ref.get("name", "")))
                    routed = True
            if not routed:
                # adopted/legacy objects without owner refs: fall back to
                # name-prefix routing
                base = (name.rsplit("-", 1)[0]
                        if kind != "RoleBasedGroupSet" else name)
                self._queue.put(("ArksApplication", ns, name))
                self._queue.put(("ArksDisaggregatedApplication", ns, base))
        if kind in ("ArksApplication", "ArksDisaggregatedApplication"):
            # endpoint watches app readiness (reference filterApp :119-168)
            sname = (
                served_model_name(obj)
                if kind == "ArksApplication"
                else (obj.spec.served_model_name or obj.spec.model.get("name", ""))
            )
            if self.store.get_opt("ArksEndpoint", ns, sname) is not None:
                self._queue.put(("ArksEndpoint", ns, sname))

    def _dispatch(self, kind: str, ns: str, name: str):
        rec = {
            "ArksModel": self.model_rec,
            "ArksApplication": self.app_rec,
            "ArksEndpoint": self.endpoint_rec,
            "ArksDisaggregatedApplication": self.disagg_rec,
        }.get(kind)
        if rec is None:
            return None
        opmetrics.workqueue_depth.labels("operator").set(self._queue.qsize())
        t0 = time.perf_counter()
        try:
            delay = rec.reconcile(ns, name)
        except Exception:
            opmetrics.reconcile_errors.labels(kind).inc()
            opmetrics.reconcile_total.labels(kind, "error").inc()
            raise
        opmetrics.reconcile_time.labels(kind).observe(time.perf_counter() - t0)
        result = ("success" if delay is None
                  else "requeue" if delay == 0 else "requeue_after")
        opmetrics.reconcile_total.labels(kind, result).inc()
        return delay

    def reconcile_until_stable(self, max_iters: int = 200) -> int:
        """Drain the queue; follow zero-delay requeues. Returns iterations."""
        n = 0
        seen_requeue: set = set()
        while n < max_iters:
            try:
                kind, ns, name = self._queue.get_nowait()
            except queue.Empty:
                break
            n += 1
            delay = self._dispatch(kind, ns, name)
            if delay == 0:
                self._queue.put((kind, ns, name))
            elif delay is not None and (kind, ns, name) not in seen_requeue:
                # timed requeues retried once per drain (tests flip external
                # state between drains)
                seen_requeue.add((kind, ns, name))
        return n

    def run(self, poll_interval: float = 1.0, workers: int = 8) -> None:
        """Long-running form: `workers` reconcile threads drain the queue
        (the reference runs MaxConcurrentReconciles=30 per controller;
        reconcilers here are level-triggered and idempotent). The same
        (kind, ns, name) never reconciles concurrently: an in-flight key
        is re-queued by its finishing worker instead."""
        inflight: set = set()
        lock = threading.Lock()

        def requeue_later(item, delay):
            t = threading.Timer(delay, lambda: self._queue.put(item))
            t.daemon = True
            t.start()

        def worker():
            while not self._stop.is_set():
                try:
                    item = self._queue.get(timeout=poll_interval)
                except queue.Empty:
                    continue
                with lock:
                    if item in inflight:
                        # coalesce: someone is reconciling this key now —
                        # run it again shortly after they finish
                        requeue_later(item, 0.05)
                        continue
                    inflight.add(item)
                try:
                    delay = self._dispatch(*item)
                except Exception as exc:  # noqa: BLE001
                    # error backoff requeue (controller-runtime retries a
                    # failed Reconcile with rate-limited backoff)
                    print(f"reconcile {item} failed: {exc!r}", flush=True)
                    delay = None
                    requeue_later(item, 5.0)
                finally:
                    with lock:
                        inflight.discard(item)
                if delay == 0:
                    self._queue.put(item)
                elif delay is not None:
                    requeue_later(item, delay)

        threads = [threading.Thread(target=worker, daemon=True,
                                    name=f"reconcile-{i}")
                   for i in range(max(workers, 1))]
        for t in threads:
            t.start()
        for t in threads:
            t.join()

    def stop(self):
        self._stop.set()
