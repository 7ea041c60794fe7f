"""Lease-based leader election for the operator.

The reference operator runs a controller-runtime manager with leader
election id ``e4ada7ad.arks.ai`` (reference cmd/main.go:198-216) so only
one of N operator replicas reconciles at a time. Same semantics here over
a ``coordination.k8s.io/v1`` Lease through the Store interface:

  * acquire when the Lease is absent, expired, or already ours;
  * renew every ``renew_s`` while leading;
  * a candidate steals the Lease ``lease_s`` after the last renewal;
  * losing the Lease (another holder appears) reports leadership lost —
    the entrypoint exits so the pod restarts as a follower.
"""

from __future__ import annotations

import threading
import time
from datetime import datetime, timezone

from .store import Conflict, NotFound

DEFAULT_LEASE_NAME = "e4ada7ad.arks.ai"


def _now_iso(clock) -> str:
    return datetime.fromtimestamp(clock(), tz=timezone.utc).strftime(
        "%Y-%m-%dT%H:%M:%S.%fZ"
    )


def _parse_iso(s: str) -> float:
    return datetime.strptime(s, "%Y-%m-%dT%H:%M:%S.%fZ").replace(
        tzinfo=timezone.utc
    ).timestamp()


class LeaderElector:
    """One candidate's view of the election. `store` is any Store
    implementation (in-memory or KubeStore); `clock` is injectable for
    tests (defaults to time.time)."""

    def __init__(self, store, identity: str, namespace: str = "arks-system",
                 name: str = DEFAULT_LEASE_NAME, lease_s: float = 15.0,
                 renew_s: float = 5.0, clock=time.time):
        self.store = store
        self.identity = identity
        self.namespace = namespace
        self.name = name
        self.lease_s = lease_s
        self.renew_s = renew_s
        self.clock = clock
        self.is_leader = False
        self._stop = threading.Event()

    # -- single protocol step ------------------------------------------------
    def _lease_dict(self, transitions: int, acquire_time: str | None = None):
        now = _now_iso(self.clock)
        return {
            "apiVersion": "coordination.k8s.io/v1",
            "kind": "Lease",
            "metadata": {"name": self.name, "namespace": self.namespace},
            "spec": {
                "holderIdentity": self.identity,
                "leaseDurationSeconds": int(self.lease_s),
                "acquireTime": acquire_time or now,
                "renewTime": now,
                "leaseTransitions": transitions,
            },
        }

    def try_acquire(self) -> bool:
        """One acquire-or-renew attempt; updates self.is_leader."""
        try:
            cur = self.store.get_opt("Lease", self.namespace, self.name)
        except Exception:
            cur = None
        if cur is None:
            try:
                self.store.create(self._lease_dict(transitions=0))
                self.is_leader = True
                return True
            except Conflict:
                self.is_leader = False
                return False
        spec = cur.get("spec", {}) if isinstance(cur, dict) else {}
        holder = spec.get("holderIdentity") or ""
        renew = spec.get("renewTime")
        expired = True
        if renew:
            try:
                expired = self.clock() - _parse_iso(renew) > float(
                    spec.get("leaseDurationSeconds", self.lease_s)
                )
            except ValueError:
                expired = True
        if holder == self.identity:
            lease = self._lease_dict(
                transitions=int(spec.get("leaseTransitions", 0)),
                acquire_time=spec.get("acquireTime"),
            )
        elif holder and not expired:
            self.is_leader = False
            return False
        else:  # vacant or expired -> steal
            lease = self._lease_dict(
                transitions=int(spec.get("leaseTransitions", 0)) + 1
            )
        lease["metadata"]["resourceVersion"] = (
            cur.get("metadata", {}).get("resourceVersion", 0)
        )
        try:
            self.store.update(lease)
            self.is_leader = True
            return True
        except (Conflict, NotFound):
            self.is_leader = False
            return False

    # -- blocking loops ------------------------------------------------------
    def acquire(self, poll_s: float | None = None) -> None:
        """Block until this candidate becomes leader (or stop() is called)."""
        poll = poll_s if poll_s is not None else self.renew_s
        while not self._stop.is_set():
            if self.try_acquire():
                return
            self._stop.wait(poll)

    def run_renew(self, on_lost=None) -> None:
        """Renew until stopped; calls on_lost() once if leadership is lost
        (another holder took the Lease after our renewals failed)."""
        while not self._stop.is_set():
            self._stop.wait(self.renew_s)
            if self._stop.is_set():
                return
            if not self.try_acquire():
                if on_lost is not None:
                    on_lost()
                return

    def release(self) -> None:
        """Best-effort: vacate the Lease so a follower takes over quickly."""
        self._stop.set()
        if not self.is_leader:
            return
        try:
            cur = self.store.get_opt("Lease", self.namespace, self.name)
            if cur and cur.get("spec", {}).get("holderIdentity") == self.identity:
                cur["spec"]["holderIdentity"] = ""
                self.store.update(cur)
        except Exception:
            pass
        self.is_leader = False

    def stop(self) -> None:
        self._stop.set()
