"""Fixed-window rate limiter (reference pkg/gateway/ratelimiter/).

Same semantics as the Redis implementation: window key =
`prefix:ns=..:user=..:model=..:<rule>:<windowStart>` with windowStart =
now truncated to the window; CheckLimit is over-limit when
current + request > limit; DoLimit increments and sets expiry window+jitter.
The store is pluggable: InMemoryCounterStore here; a Redis(RESP)-backed
store can implement the same three methods.
"""

from __future__ import annotations

import threading
import time
from dataclasses import dataclass

# Hard-coded rule table (reference ratelimiter/rate_limiter.go:31-68).
MINUTE = 60.0
DAY = 86400.0


@dataclass(frozen=True)
class Rule:
    name: str  # rpm | rpd | tpm | tpd
    type: str  # "request" | "token"
    window: float


RULES: dict[str, Rule] = {
    "rpm": Rule("rpm", "request", MINUTE),
    "rpd": Rule("rpd", "request", DAY),
    "tpm": Rule("tpm", "token", MINUTE),
    "tpd": Rule("tpd", "token", DAY),
}

TYPE_REQUEST = "request"
TYPE_TOKEN = "token"


class InMemoryCounterStore:
    """Counter store with per-key expiry (the Redis stand-in)."""

    def __init__(self):
        self._lock = threading.Lock()
        self._data: dict[str, tuple[int, float]] = {}  # key -> (value, expires)

    def incr_by(self, key: str, amount: int, expire_s: float, now: float) -> int:
        with self._lock:
            val, exp = self._data.get(key, (0, 0.0))
            if exp and exp <= now:
                val = 0
            val += amount
            exp = exp if exp > now else now + expire_s
            self._data[key] = (val, exp)
            return val

    def get(self, key: str, now: float) -> int:
        with self._lock:
            val, exp = self._data.get(key, (0, 0.0))
            if exp and exp <= now:
                return 0
            return val

    def set(self, key: str, value: int, now: float, expire_s: float | None = None) -> None:
        with self._lock:
            exp = now + expire_s if expire_s else float("inf")
            self._data[key] = (value, exp)

    def sweep(self, now: float) -> None:
        with self._lock:
            dead = [k for k, (_, e) in self._data.items() if e <= now]
            for k in dead:
                del self._data[k]


class RedisCounterStore:
    """Redis(RESP)-backed counter store — the reference's production
    backend (pkg/gateway/ratelimiter/redis_impl.go:115-165): pipelined
    INCRBY + TTL, EXPIRE(window + jitter <= 1 s) set only when the key has
    no TTL yet. Any number of gateway replicas share one budget."""

    def __init__(self, client):
        self.client = client  # arks_amd.gateway.resp.RespClient

    def incr_by(self, key: str, amount: int, expire_s: float, now: float) -> int:
        import random

        val, ttl = self.client.pipeline(
            [("INCRBY", key, amount), ("TTL", key)]
        )
        if isinstance(val, Exception):
            raise val
        if isinstance(ttl, int) and ttl < 0:
            self.client.command(
                "EXPIRE", key, int(expire_s + random.uniform(0.0, 1.0))
            )
        return int(val)

    def get(self, key: str, now: float) -> int:
        v = self.client.command("GET", key)
        return int(v) if v is not None else 0

    def set(self, key: str, value: int, now: float,
            expire_s: float | None = None) -> None:
        if expire_s:
            self.client.command("SET", key, value, "EX", int(expire_s))
        else:
            self.client.command("SET", key, value)


@dataclass(frozen=True)
class LimitDescriptor:
    namespace: str
    user: str
    model: str
    rule: str  # rpm/rpd/tpm/tpd
    limit: int


class RateLimiter:
    KEY_PREFIX = "arks-ratelimiter"

    def __init__(self, store: InMemoryCounterStore | None = None, clock=time.time):
        self.store = store or InMemoryCounterStore()
        self.clock = clock

    def _key(self, d: LimitDescriptor, now: float) -> str:
        rule = RULES[d.rule]
        window_start = int(now // rule.window * rule.window)
        return (
            f"{self.KEY_PREFIX}:ns={d.namespace}:user={d.user}:model={d.model}"
            f":{d.rule}:{window_start}"
        )

    def check_limit(self, descriptors: list[LimitDescriptor], request: int = 1
                    ) -> tuple[bool, str | None]:
        """True when ALL descriptors admit `request` more units.
        Token rules are checked at 0 increment pre-request
        (reference check.go:108-156)."""
        now = self.clock()
        for d in descriptors:
            inc = request if RULES[d.rule].type == TYPE_REQUEST else 0
            cur = self.store.get(self._key(d, now), now)
            if cur + inc > d.limit:
                return False, d.rule
        return True, None

    def do_limit(self, descriptors: list[LimitDescriptor], amount: int) -> None:
        """Increment counters (request rules by 1 pre-request; token rules by
        token count post-response — reference check.go:31-59)."""
        now = self.clock()
        for d in descriptors:
            rule = RULES[d.rule]
            self.store.incr_by(self._key(d, now), amount, rule.window + 1.0, now)
