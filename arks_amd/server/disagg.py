"""Prefill/decode disaggregation wire protocol.

The reference's ArksDisaggregatedApplication runs SGLang prefill and decode
fleets with `--disaggregation-mode prefill|decode` plus an external KV
transfer (reference arksdisaggregatedapplication_controller.go:1672-1724).
Here both halves are first-party:

  POST /disagg/prefill  (on a prefill instance)
      JSON {request_id, token_ids, sampling{...}}
  --> 200, headers x-arks-first-token / x-arks-finish-reason /
      x-arks-kv-shape (comma-separated ints), body = raw bf16 KV pages in
      model_runner.extract_kv layout [L, 2, nblocks, Hkv, 16, D].

The decode instance pulls this over TCP, injects the pages into its own
paged cache, and continues the decode loop (engine.add_prefilled). On one
MI355X node the transfer rides loopback; cross-node it is bounded by the
fabric, and the layout is a single contiguous buffer so an RDMA transport
can replace httpx without touching the engine.

TP>1 on either side: extract_kv all-gathers the KV head shards so the wire
tensor always carries the FULL head set, and inject_kv takes each rank's
slice — prefill and decode instances may run different TP degrees.
"""

from __future__ import annotations

import torch
from fastapi import Request, Response

from ..engine import SamplingParams

KV_DTYPE = torch.bfloat16


def encode_kv(kv: torch.Tensor) -> tuple[str, bytes]:
    """-> (shape header, raw bytes). kv must be the extract_kv host tensor."""
    kv = kv.contiguous()
    shape = ",".join(str(s) for s in kv.shape)
    return shape, kv.view(torch.uint16).numpy().tobytes()


def decode_kv(shape_header: str, body: bytes) -> torch.Tensor:
    shape = tuple(int(s) for s in shape_header.split(","))
    t = torch.frombuffer(bytearray(body), dtype=torch.uint16).view(shape)
    return t.view(KV_DTYPE)


def add_prefill_routes(app, engine) -> None:
    """Register the prefill-side transfer endpoint on a FastAPI app."""

    @app.post("/disagg/prefill")
    async def disagg_prefill(raw: Request):
        req = await raw.json()
        sp = SamplingParams(**req.get("sampling", {}))
        first, reason, kv = await engine.disagg_prefill(
            req["request_id"], req["token_ids"], sp
        )
        shape, body = encode_kv(kv)
        return Response(
            content=body,
            media_type="application/octet-stream",
            headers={
                "x-arks-first-token": str(first),
                "x-arks-finish-reason": reason or "",
                "x-arks-kv-shape": shape,
            },
        )


async def remote_prefill(addr: str, request_id: str, token_ids: list[int],
                         sampling: SamplingParams, transport=None):
    """Decode-side client: run the prompt on `addr`, return
    (first_token, finish_reason, kv tensor)."""
    import httpx

    base = addr if addr.startswith("http") else f"http://{addr}"
    async with httpx.AsyncClient(transport=transport, base_url=base,
                                 timeout=300.0) as client:
        r = await client.post("/disagg/prefill", json={
            "request_id": request_id,
            "token_ids": token_ids,
            "sampling": sampling.__dict__.copy(),
        })
        r.raise_for_status()
        kv = decode_kv(r.headers["x-arks-kv-shape"], r.content)
        return (
            int(r.headers["x-arks-first-token"]),
            r.headers.get("x-arks-finish-reason") or None,
            kv,
        )
