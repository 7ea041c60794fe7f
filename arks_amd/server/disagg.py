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

# wire dtypes: bf16 pages (default) or fp8 e4m3 pages (kv_cache_dtype=fp8)
_WIRE = {
    "bfloat16": (torch.bfloat16, torch.uint16),
    "float8_e4m3fn": (torch.float8_e4m3fn, torch.uint8),
}


def encode_kv(kv: torch.Tensor) -> tuple[str, bytes]:
    """-> (shape header incl. dtype, raw bytes) for the extract_kv host
    tensor. Header: "<dtype>:<d0>,<d1>,..."."""
    kv = kv.contiguous()
    name = str(kv.dtype).removeprefix("torch.")
    _, raw = _WIRE[name]
    shape = ",".join(str(s) for s in kv.shape)
    return f"{name}:{shape}", kv.view(raw).numpy().tobytes()


def decode_kv(shape_header: str, body: bytes) -> torch.Tensor:
    if ":" in shape_header:
        name, shape_s = shape_header.split(":", 1)
    else:  # older peers sent bare shapes (bf16 implied)
        name, shape_s = "bfloat16", shape_header
    dtype, raw = _WIRE[name]
    shape = tuple(int(s) for s in shape_s.split(","))
    t = torch.frombuffer(bytearray(body), dtype=raw).view(shape)
    return t.view(dtype)


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
