"""QoS config provider (reference pkg/gateway/qosconfig/).

Reads ArksToken / ArksQuota / ArksEndpoint CRs from the control-plane Store
(the informer-cache equivalent), indexes tokens by spec.token, and runs the
10 s quota sync loop: usage -> ArksQuota.status.quotaStatus, and CR -> store
write-back when the counter store lost data (crash recovery — reference
qosconfig/arks_impl.go:171-300).
"""

from __future__ import annotations

import threading
from dataclasses import dataclass, field

from ..controlplane.store import Store
from ..crd.types import ArksQuota, ArksToken, QuotaStatusEntry
from .limiter import LimitDescriptor, RULES
from .quota import QuotaDescriptor, QuotaService


@dataclass
class UserQos:
    user: str  # token CR name
    namespace: str
    model: str
    quota_name: str = ""
    rate_limits: list[tuple[str, int]] = field(default_factory=list)  # (rule, value)

    def limit_descriptors(self) -> list[LimitDescriptor]:
        return [
            LimitDescriptor(self.namespace, self.user, self.model, rule, value)
            for rule, value in self.rate_limits
            if rule in RULES
        ]


class ConfigProvider:
    def __init__(self, store: Store, quota_service: QuotaService,
                 metrics=None):
        self.store = store
        self.quota_service = quota_service
        self.metrics = metrics  # GatewayMetrics (quota gauges)
        self._stop = threading.Event()

    # --- lookups (token indexed by spec.token — arks_impl.go:59-73) ---
    def find_token(self, token: str) -> ArksToken | None:
        for t in self.store.list("ArksToken"):
            if t.spec.token == token:
                return t
        return None

    def get_qos_by_token(self, token: str, model: str) -> UserQos | None:
        t = self.find_token(token)
        if t is None:
            return None
        for qos in t.spec.qos:
            if qos.endpoint_name == model:
                return UserQos(
                    user=t.metadata.name,
                    namespace=t.metadata.namespace,
                    model=model,
                    quota_name=qos.quota.get("name", ""),
                    rate_limits=[(r.type, r.value) for r in qos.rate_limits],
                )
        return None

    def get_model_list(self, namespace: str) -> list[str]:
        return [e.metadata.name for e in self.store.list("ArksEndpoint", namespace)]

    def get_models_by_token(self, token: str) -> list[str]:
        t = self.find_token(token)
        if t is None:
            return []
        return [q.endpoint_name for q in t.spec.qos if q.endpoint_name]

    def get_quota_descriptors(self, qos: UserQos) -> list[QuotaDescriptor]:
        if not qos.quota_name:
            return []
        quota: ArksQuota | None = self.store.get_opt(
            "ArksQuota", qos.namespace, qos.quota_name
        )
        if quota is None:
            return []
        return [
            QuotaDescriptor(qos.namespace, qos.quota_name, q.type, q.value)
            for q in quota.spec.quotas
        ]

    # --- quota usage sync (arks_impl.go:217-300) ---
    def sync_quota_usage(self) -> None:
        for quota in self.store.list("ArksQuota"):
            ns, name = quota.metadata.namespace, quota.metadata.name
            changed = False
            for q in quota.spec.quotas:
                live = self.quota_service.get_usage(ns, name, q.type)
                if self.metrics is not None:
                    self.metrics.quota_usage.labels(
                        namespace=ns, quota=name, type=q.type).set(live)
                    self.metrics.quota_limit.labels(
                        namespace=ns, quota=name, type=q.type).set(q.value)
                entry = next(
                    (e for e in quota.status.quota_status if e.type == q.type), None
                )
                if entry is None:
                    entry = QuotaStatusEntry(type=q.type, used=0)
                    quota.status.quota_status.append(entry)
                    changed = True
                if live < entry.used:
                    # counter store lost data: push CR value back (recovery)
                    self.quota_service.set_usage(ns, name, q.type, entry.used)
                elif live > entry.used:
                    entry.used = live
                    changed = True
            if changed:
                self.store.update(quota)

    def start_sync_loop(self, interval_s: float = 10.0) -> threading.Thread:
        def loop():
            while not self._stop.wait(interval_s):
                self.sync_quota_usage()

        t = threading.Thread(target=loop, daemon=True)
        t.start()
        return t

    def stop(self):
        self._stop.set()
