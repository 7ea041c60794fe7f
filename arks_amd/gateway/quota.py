"""Cumulative quota service (reference pkg/gateway/quota/): lifetime
counters (no window/TTL) keyed namespace/quota-name/type; over-limit when
current > limit."""

from __future__ import annotations

import threading
from dataclasses import dataclass

QUOTA_TYPES = ("prompt", "response", "total")


@dataclass(frozen=True)
class QuotaDescriptor:
    namespace: str
    quota_name: str
    type: str  # prompt | response | total
    limit: int


class QuotaService:
    KEY_PREFIX = "arks-quota"

    def __init__(self):
        self._lock = threading.Lock()
        self._usage: dict[str, int] = {}

    def _key(self, ns: str, name: str, type_: str) -> str:
        return f"{self.KEY_PREFIX}:ns={ns}:quota={name}:type={type_}"

    def incr_usage(self, ns: str, name: str, type_: str, amount: int) -> int:
        with self._lock:
            k = self._key(ns, name, type_)
            self._usage[k] = self._usage.get(k, 0) + amount
            return self._usage[k]

    def set_usage(self, ns: str, name: str, type_: str, value: int) -> None:
        with self._lock:
            self._usage[self._key(ns, name, type_)] = value

    def get_usage(self, ns: str, name: str, type_: str) -> int:
        with self._lock:
            return self._usage.get(self._key(ns, name, type_), 0)

    def check(self, descriptors: list[QuotaDescriptor]) -> tuple[bool, str | None]:
        """Over-limit when current > limit (reference quota/redis_impl.go:101)."""
        for d in descriptors:
            if self.get_usage(d.namespace, d.quota_name, d.type) > d.limit:
                return False, d.type
        return True, None
