"""A miniature in-process Redis (RESP2) server for gateway store tests —
the stand-in for the reference's real-Redis test dependency
(pkg/gateway/ratelimiter/redis_impl_test.go runs against $REDIS_ADDR).
Supports: PING GET SET(EX) INCRBY EXPIRE TTL DEL FLUSHALL, with key expiry.
"""

from __future__ import annotations

import socket
import socketserver
import threading
import time


class _State:
    def __init__(self):
        self.lock = threading.Lock()
        self.data: dict[bytes, bytes] = {}
        self.expiry: dict[bytes, float] = {}

    def _alive(self, key: bytes, now: float) -> bool:
        exp = self.expiry.get(key)
        if exp is not None and exp <= now:
            self.data.pop(key, None)
            self.expiry.pop(key, None)
            return False
        return key in self.data


class _Handler(socketserver.BaseRequestHandler):
    def _read_command(self, buf: bytearray) -> list[bytes] | None:
        sock = self.request

        def read_line():
            while b"\r\n" not in buf:
                chunk = sock.recv(65536)
                if not chunk:
                    return None
                buf.extend(chunk)
            i = buf.index(b"\r\n")
            line = bytes(buf[:i])
            del buf[: i + 2]
            return line

        line = read_line()
        if line is None:
            return None
        assert line[:1] == b"*", line
        n = int(line[1:])
        args = []
        for _ in range(n):
            hdr = read_line()
            if hdr is None:
                return None
            assert hdr[:1] == b"$"
            ln = int(hdr[1:])
            while len(buf) < ln + 2:
                chunk = sock.recv(65536)
                if not chunk:
                    return None
                buf.extend(chunk)
            args.append(bytes(buf[:ln]))
            del buf[: ln + 2]
        return args

    def handle(self):
        st: _State = self.server.state  # type: ignore[attr-defined]
        buf = bytearray()
        while True:
            try:
                cmd = self._read_command(buf)
            except (ConnectionError, OSError):
                return
            if cmd is None:
                return
            self.request.sendall(self._dispatch(st, cmd))

    @staticmethod
    def _dispatch(st: _State, cmd: list[bytes]) -> bytes:
        name = cmd[0].upper()
        now = time.time()
        with st.lock:
            if name == b"PING":
                return b"+PONG\r\n"
            if name == b"FLUSHALL":
                st.data.clear()
                st.expiry.clear()
                return b"+OK\r\n"
            if name == b"GET":
                k = cmd[1]
                if not st._alive(k, now):
                    return b"$-1\r\n"
                v = st.data[k]
                return b"$%d\r\n%s\r\n" % (len(v), v)
            if name == b"SET":
                k, v = cmd[1], cmd[2]
                st.data[k] = v
                st.expiry.pop(k, None)
                if len(cmd) >= 5 and cmd[3].upper() == b"EX":
                    st.expiry[k] = now + int(cmd[4])
                return b"+OK\r\n"
            if name == b"INCRBY":
                k = cmd[1]
                cur = int(st.data[k]) if st._alive(k, now) else 0
                cur += int(cmd[2])
                st.data[k] = str(cur).encode()
                return b":%d\r\n" % cur
            if name == b"EXPIRE":
                k = cmd[1]
                if not st._alive(k, now):
                    return b":0\r\n"
                st.expiry[k] = now + int(cmd[2])
                return b":1\r\n"
            if name == b"TTL":
                k = cmd[1]
                if not st._alive(k, now):
                    return b":-2\r\n"
                exp = st.expiry.get(k)
                if exp is None:
                    return b":-1\r\n"
                return b":%d\r\n" % max(0, int(exp - now))
            if name == b"DEL":
                n = 0
                for k in cmd[1:]:
                    if st._alive(k, now):
                        del st.data[k]
                        st.expiry.pop(k, None)
                        n += 1
                return b":%d\r\n" % n
        return b"-ERR unknown command '%s'\r\n" % name


class MiniRedis:
    """Threaded mini-Redis; use as a context manager. `.port` is the bound
    TCP port on 127.0.0.1."""

    def __init__(self):
        class Srv(socketserver.ThreadingTCPServer):
            allow_reuse_address = True
            daemon_threads = True

        self.server = Srv(("127.0.0.1", 0), _Handler)
        self.server.state = _State()  # type: ignore[attr-defined]
        self.port = self.server.server_address[1]
        self._thread = threading.Thread(
            target=self.server.serve_forever, daemon=True
        )

    def __enter__(self):
        self._thread.start()
        return self

    def __exit__(self, *exc):
        self.server.shutdown()
        self.server.server_close()
        return False
