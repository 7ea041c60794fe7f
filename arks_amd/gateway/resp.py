"""Minimal RESP2 (Redis Serialization Protocol) client.

The gateway's rate-limit/quota state lives in Redis in the reference
deployment (pkg/gateway/ratelimiter/redis_impl.go, quota/redis_impl.go);
this is a dependency-free socket client speaking the subset the gateway
needs (GET/SET/INCRBY/EXPIRE/TTL/PING/DEL), with command pipelining to
match the reference's pipelined check/incr round trips.

Thread-safe: one connection guarded by a lock (the gateway's asyncio loop
calls through run_in_executor; contention is a couple of commands per
request). Reconnects once on a broken connection.
"""

from __future__ import annotations

import socket
import threading


class RespError(RuntimeError):
    """Server-side error reply (-ERR ...)."""


class RespClient:
    def __init__(self, host: str = "127.0.0.1", port: int = 6379,
                 timeout: float = 2.0):
        self.host = host
        self.port = port
        self.timeout = timeout
        self._lock = threading.Lock()
        self._sock: socket.socket | None = None
        self._buf = b""

    # ---- connection ----
    def _connect(self) -> socket.socket:
        s = socket.create_connection((self.host, self.port), self.timeout)
        s.settimeout(self.timeout)
        self._buf = b""
        return s

    def close(self) -> None:
        with self._lock:
            if self._sock is not None:
                try:
                    self._sock.close()
                finally:
                    self._sock = None

    # ---- protocol ----
    @staticmethod
    def _encode(args: tuple) -> bytes:
        out = [b"*%d\r\n" % len(args)]
        for a in args:
            b = a if isinstance(a, bytes) else str(a).encode()
            out.append(b"$%d\r\n%s\r\n" % (len(b), b))
        return b"".join(out)

    def _read_line(self, sock: socket.socket) -> bytes:
        while b"\r\n" not in self._buf:
            chunk = sock.recv(65536)
            if not chunk:
                raise ConnectionError("redis connection closed")
            self._buf += chunk
        line, self._buf = self._buf.split(b"\r\n", 1)
        return line

    def _read_exact(self, sock: socket.socket, n: int) -> bytes:
        while len(self._buf) < n + 2:
            chunk = sock.recv(65536)
            if not chunk:
                raise ConnectionError("redis connection closed")
            self._buf += chunk
        data, self._buf = self._buf[:n], self._buf[n + 2:]
        return data

    def _read_reply(self, sock: socket.socket):
        line = self._read_line(sock)
        t, rest = line[:1], line[1:]
        if t == b"+":
            return rest.decode()
        if t == b"-":
            raise RespError(rest.decode())
        if t == b":":
            return int(rest)
        if t == b"$":
            n = int(rest)
            if n == -1:
                return None
            return self._read_exact(sock, n)
        if t == b"*":
            n = int(rest)
            if n == -1:
                return None
            return [self._read_reply(sock) for _ in range(n)]
        raise RespError(f"unexpected reply type {line!r}")

    # ---- public API ----
    def pipeline(self, cmds: list[tuple]) -> list:
        """Send commands in one write, read all replies (server errors are
        returned in-place as RespError instances, matching go-redis
        pipeline semantics)."""
        payload = b"".join(self._encode(c) for c in cmds)
        with self._lock:
            for attempt in (0, 1):
                sock = self._sock
                try:
                    if sock is None:
                        sock = self._sock = self._connect()
                    sock.sendall(payload)
                    out = []
                    for _ in cmds:
                        try:
                            out.append(self._read_reply(sock))
                        except RespError as e:
                            out.append(e)
                    return out
                except (OSError, ConnectionError):
                    # Broken connection: drop it and retry once. Note the
                    # at-least-once caveat shared with go-redis retries: a
                    # send that died after the server applied an INCRBY is
                    # re-applied — for rate limiting that over-counts
                    # (conservative direction), never under-counts.
                    try:
                        if self._sock is not None:
                            self._sock.close()
                    finally:
                        self._sock = None
                    if attempt:
                        raise
        raise AssertionError("unreachable")

    def command(self, *args):
        out = self.pipeline([args])[0]
        if isinstance(out, RespError):
            raise out
        return out

    def ping(self) -> bool:
        try:
            return self.command("PING") == "PONG"
        except (OSError, ConnectionError, RespError):
            return False
