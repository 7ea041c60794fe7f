"""Incremental SSE usage scanner, shared by both gateway data planes (the
ASGI proxy relay and the Envoy ext_proc servicer).

Spec-correct parsing (the reference uses openai-go's ssestream): CRLF line
endings tolerated, multi-line `data:` fields joined with \\n, comments and
other fields ignored, arbitrary fragmentation across feed() calls. The
OpenAI usage chunk is the final event with a `usage` object and empty
`choices` (reference handle_response.go:113-133).
"""

from __future__ import annotations

import json
from typing import Callable


class SSEUsageScanner:
    def __init__(self, on_usage: Callable[[dict], None]):
        self.buf = b""
        self._data: list[bytes] = []
        self.on_usage = on_usage

    def feed(self, chunk: bytes) -> None:
        self.buf += chunk
        while b"\n" in self.buf:
            line, self.buf = self.buf.split(b"\n", 1)
            line = line.rstrip(b"\r")
            if line.startswith(b"data:"):
                self._data.append(line[len(b"data:"):].lstrip())
            elif line == b"":
                parts, self._data = self._data, []
                if not parts:
                    continue
                data = b"\n".join(parts)
                if data.strip() == b"[DONE]":
                    continue
                try:
                    obj = json.loads(data)
                except Exception:
                    continue
                if obj.get("usage") and not obj.get("choices"):
                    self.on_usage(obj["usage"])
