"""Tensor-parallel process-group management over RCCL/xGMI.

One process per GPU; torch.distributed backend "nccl" IS RCCL on ROCm. The
TP group is the whole world for a single serving instance (the LWS
leader/worker topology of the reference maps each inference group to one
node's xGMI-connected GPUs — SURVEY.md §5 'Distributed communication
backend'). Falls back to gloo on CPU so the distributed path is testable
without hardware.
"""

from __future__ import annotations

import os
from datetime import timedelta

import torch
import torch.distributed as dist

_TP_GROUP = None
_TP_RANK = 0
_TP_WORLD = 1


def init_tp(
    tp_size: int | None = None,
    rank: int | None = None,
    backend: str | None = None,
    master_addr: str = "127.0.0.1",
    master_port: int = 29500,
    timeout_s: int = 600,
) -> None:
    """Initialize the TP process group. Reads torchrun env (RANK/WORLD_SIZE/
    MASTER_ADDR/MASTER_PORT) when present."""
    global _TP_GROUP, _TP_RANK, _TP_WORLD
    world = int(os.environ.get("WORLD_SIZE", tp_size or 1))
    if world <= 1:
        _TP_GROUP, _TP_RANK, _TP_WORLD = None, 0, 1
        return
    r = int(os.environ.get("RANK", rank if rank is not None else 0))
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", master_addr)
        os.environ.setdefault("MASTER_PORT", str(master_port))
        dist.init_process_group(
            backend=backend,
            rank=r,
            world_size=world,
            timeout=timedelta(seconds=timeout_s),
        )
    if torch.cuda.is_available():
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", r)))
    _TP_GROUP = dist.group.WORLD
    _TP_RANK = dist.get_rank()
    _TP_WORLD = dist.get_world_size()


def destroy_tp() -> None:
    global _TP_GROUP, _TP_RANK, _TP_WORLD
    if dist.is_initialized():
        dist.destroy_process_group()
    _TP_GROUP, _TP_RANK, _TP_WORLD = None, 0, 1


def get_tp_group():
    return _TP_GROUP


def get_tp_rank() -> int:
    return _TP_RANK


def get_tp_world_size() -> int:
    return _TP_WORLD


_P2P_AR = None  # lazily built OneShotAllReduce (or False after a failure)


def _p2p_ar():
    """Lazy one-shot p2p all-reduce setup (first CUDA all-reduce call,
    which happens during eager warmup — before any hipGraph capture)."""
    global _P2P_AR
    if _P2P_AR is None:
        try:
            if dist.get_backend(_TP_GROUP) == "nccl":
                from .p2p import OneShotAllReduce

                _P2P_AR = OneShotAllReduce(
                    _TP_GROUP, _TP_RANK, _TP_WORLD,
                    torch.device("cuda", torch.cuda.current_device()),
                )
            else:
                _P2P_AR = False
        except Exception:
            _P2P_AR = False
    return _P2P_AR if _P2P_AR else None


def tp_all_reduce(x: torch.Tensor) -> torch.Tensor:
    """Sum all-reduce across the TP group (no-op at TP=1). Small bf16 CUDA
    tensors take the one-shot p2p xGMI kernel (parallel/p2p.py) when its
    init-time self-check passed; everything else goes through RCCL."""
    if _TP_WORLD > 1:
        if x.is_cuda:
            ar = _p2p_ar()
            if ar is not None and ar.usable(x):
                # returns a NEW tensor (all callers consume the return)
                return ar(x)
        dist.all_reduce(x, op=dist.ReduceOp.SUM, group=_TP_GROUP)
    return x


def tp_all_gather(x: torch.Tensor, dim: int = -1) -> torch.Tensor:
    """All-gather shards along `dim` (no-op at TP=1)."""
    if _TP_WORLD <= 1:
        return x
    # fp8 has no collective support on gloo (nor NCCL in common torch
    # versions) — ship the bytes and view back.
    fp8 = x.dtype == torch.float8_e4m3fn
    wire = x.contiguous().view(torch.uint8) if fp8 else x.contiguous()
    parts = [torch.empty_like(wire) for _ in range(_TP_WORLD)]
    dist.all_gather(parts, wire, group=_TP_GROUP)
    if fp8:
        parts = [p.view(torch.float8_e4m3fn) for p in parts]
    return torch.cat(parts, dim=dim)


def tp_broadcast_object(obj, src: int = 0):
    """Broadcast a picklable object from the driver rank (no-op at TP=1)."""
    if _TP_WORLD <= 1:
        return obj
    buf = [obj]
    dist.broadcast_object_list(buf, src=src, group=_TP_GROUP)
    return buf[0]


def tp_broadcast_tensor(x: torch.Tensor | None, src: int = 0) -> torch.Tensor:
    """Broadcast a tensor from `src` (shape/dtype via a small object
    broadcast, payload via dist.broadcast on the backend's device — avoids
    pickling multi-GB KV pages). Non-src ranks pass None. No-op at TP=1."""
    if _TP_WORLD <= 1:
        return x
    meta = None
    if _TP_RANK == src:
        meta = (tuple(x.shape), str(x.dtype).removeprefix("torch."))
    meta = tp_broadcast_object(meta, src)
    dev = "cuda" if dist.get_backend(_TP_GROUP) == "nccl" else "cpu"
    dtype = getattr(torch, meta[1])
    if _TP_RANK == src:
        t = x.to(dev).contiguous()
    else:
        t = torch.empty(meta[0], dtype=dtype, device=dev)
    # fp8 has no collective support on gloo — ship the bytes
    wire = t.view(torch.uint8) if dtype == torch.float8_e4m3fn else t
    dist.broadcast(wire, src=src, group=_TP_GROUP)
    return t
