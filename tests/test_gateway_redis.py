"""Redis(RESP)-backed limiter/quota stores against an in-process
mini-Redis — the reference's production backend semantics
(pkg/gateway/ratelimiter/redis_impl.go, quota/redis_impl.go), including
two gateway-side limiter instances sharing one budget."""

import time

import pytest

from arks_amd.gateway import (
    LimitDescriptor,
    QuotaDescriptor,
    QuotaService,
    RateLimiter,
)
from arks_amd.gateway.limiter import RedisCounterStore
from arks_amd.gateway.quota import RedisQuotaStore
from arks_amd.gateway.resp import RespClient, RespError

import sys, os
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from miniredis import MiniRedis


@pytest.fixture()
def redis():
    with MiniRedis() as r:
        yield r


def test_resp_client_basics(redis):
    c = RespClient("127.0.0.1", redis.port)
    assert c.ping()
    assert c.command("GET", "nope") is None
    assert c.command("SET", "k", 5) == "OK"
    assert c.command("GET", "k") == b"5"
    assert c.command("INCRBY", "k", 3) == 8
    assert c.command("TTL", "k") == -1
    assert c.command("EXPIRE", "k", 100) == 1
    assert 0 < c.command("TTL", "k") <= 100
    # pipelining returns per-command replies in order
    out = c.pipeline([("INCRBY", "p", 2), ("TTL", "p"), ("GET", "p")])
    assert out[0] == 2 and out[1] == -1 and out[2] == b"2"
    with pytest.raises(RespError):
        c.command("BOGUS")


def test_redis_limiter_window_semantics(redis):
    c = RespClient("127.0.0.1", redis.port)
    rl = RateLimiter(store=RedisCounterStore(c))
    d = [LimitDescriptor("ns", "u", "m", "rpm", 2)]
    assert rl.check_limit(d)[0]
    rl.do_limit(d, 1)
    rl.do_limit(d, 1)
    ok, rule = rl.check_limit(d)
    assert not ok and rule == "rpm"
    # the key carries a TTL of window + jitter (reference redis_impl.go:151)
    keys = [k for k in redis.server.state.data if b":rpm:" in k]
    assert keys
    ttl = c.command("TTL", keys[0])
    assert 0 < ttl <= 61


def test_two_limiter_replicas_share_one_budget(redis):
    """Two gateway instances over one Redis must see each other's
    increments — the multi-replica property the in-memory store lacks."""
    a = RateLimiter(store=RedisCounterStore(RespClient("127.0.0.1", redis.port)))
    b = RateLimiter(store=RedisCounterStore(RespClient("127.0.0.1", redis.port)))
    d = [LimitDescriptor("ns", "u", "m", "rpm", 3)]
    a.do_limit(d, 1)
    b.do_limit(d, 1)
    a.do_limit(d, 1)
    ok, rule = b.check_limit(d)
    assert not ok and rule == "rpm"


def test_redis_quota_cumulative_shared(redis):
    qa = QuotaService(store=RedisQuotaStore(RespClient("127.0.0.1", redis.port)))
    qb = QuotaService(store=RedisQuotaStore(RespClient("127.0.0.1", redis.port)))
    d = [QuotaDescriptor("ns", "qq", "total", 10)]
    qa.incr_usage("ns", "qq", "total", 6)
    qb.incr_usage("ns", "qq", "total", 5)
    ok, t = qa.check(d)
    assert not ok and t == "total"
    # set_usage writes through (the provider's CR crash-recovery path)
    qa.set_usage("ns", "qq", "total", 0)
    assert qb.get_usage("ns", "qq", "total") == 0
    assert qb.check(d)[0]


def test_resp_client_reconnects(redis):
    c = RespClient("127.0.0.1", redis.port)
    assert c.command("SET", "x", 1) == "OK"
    # kill the client's socket under it; next command must reconnect
    c._sock.close()
    assert c.command("GET", "x") == b"1"


def test_redis_limiter_token_rules(redis):
    c = RespClient("127.0.0.1", redis.port)
    rl = RateLimiter(store=RedisCounterStore(c))
    d = [LimitDescriptor("ns", "u", "m", "tpm", 100)]
    rl.do_limit(d, 100)
    assert rl.check_limit(d)[0]  # token rules check at 0 increment
    rl.do_limit(d, 1)
    assert not rl.check_limit(d)[0]


def test_resp_parser_fragmented_and_nested():
    """Protocol-level: replies fragmented byte-by-byte across packets,
    nested arrays, null bulks/arrays, and in-pipeline errors all parse
    (the client buffers partial reads — resp.py _read_line/_read_exact)."""
    import socket
    import threading

    # crafted reply stream for one pipeline of 6 commands
    payload = (b"+OK\r\n"
               b":42\r\n"
               b"$-1\r\n"
               b"*-1\r\n"
               b"-ERR boom\r\n"
               b"*3\r\n$3\r\nfoo\r\n:7\r\n*2\r\n+a\r\n$0\r\n\r\n")

    srv = socket.socket()
    srv.bind(("127.0.0.1", 0))
    srv.listen(1)

    def serve():
        conn, _ = srv.accept()
        conn.recv(65536)  # the pipelined commands (content irrelevant)
        for i in range(len(payload)):  # dribble one byte per send
            conn.sendall(payload[i:i + 1])
        conn.close()

    t = threading.Thread(target=serve, daemon=True)
    t.start()
    from arks_amd.gateway.resp import RespClient, RespError

    c = RespClient("127.0.0.1", srv.getsockname()[1], timeout=5.0)
    out = c.pipeline([("PING",), ("X",), ("GET", "m"), ("LRANGE", "k"),
                      ("BAD",), ("NEST",)])
    assert out[0] == "OK"
    assert out[1] == 42
    assert out[2] is None
    assert out[3] is None
    assert isinstance(out[4], RespError) and "boom" in str(out[4])
    assert out[5] == [b"foo", 7, ["a", b""]]
    c.close()
    srv.close()
    t.join(timeout=5)
