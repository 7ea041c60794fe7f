"""One-shot p2p all-reduce over xGMI for small TP tensors.

MI355X has no switch: 7 point-to-point xGMI links per GPU, so RCCL's ring
all-reduce pays (world-1) serialized link hops — painful for the small
per-token tensors of tensor-parallel decode (SURVEY.md §5 recommends a
hand-written path below ~1 MiB). OneShotAllReduce implements the classic
small-message scheme (every rank pushes into every peer's mailbox, then
reduces locally — see ops/csrc/allreduce.hip) with:

  * IPC-mapped mailboxes (hipIpcGetMemHandle/hipIpcOpenMemHandle,
    exchanged once over the existing torch.distributed group);
  * device-resident sequence numbers, so calls are hipGraph-capture-safe
    (replays keep incrementing);
  * an init-time SELF-CHECK against dist.all_reduce — any mismatch or
    setup failure quietly falls back to RCCL, so a driver-side multi-GPU
    run can never be broken by this path.
"""

from __future__ import annotations

import logging
import os

import torch
import torch.distributed as dist

log = logging.getLogger("arks.p2p")

MAX_ELEMS = 1 << 19  # 512k bf16 = 1 MiB payload cap (ring wins above)
AR_MAX_BLOCKS = 64


class OneShotAllReduce:
    """Constructed once per TP group (CUDA + nccl backend, world 2/4/8).
    `available` is False when setup or the self-check failed."""

    def __init__(self, group, rank: int, world: int, device):
        from .. import ops

        self.group = group
        self.rank = rank
        self.world = world
        self.device = device
        self.available = False
        self._opened: list[int] = []
        if world not in (2, 4, 8) or not torch.cuda.is_available():
            return
        if os.environ.get("ARKS_P2P_ALLREDUCE", "1") != "1":
            return
        try:
            nat = ops._native()
            # raw hipMalloc allocations: an IPC handle for a torch
            # caching-allocator tensor maps the underlying block, not the
            # tensor's offset within it
            # x2: parity double-buffered mailbox halves (see
            # allreduce.hip's consecutive-call race note)
            self._mail_ptr, mail_h = nat.ipc_alloc(2 * 8 * MAX_ELEMS * 2)
            self._flag_ptr, flag_h = nat.ipc_alloc(world * AR_MAX_BLOCKS * 8)
            self._owned = [self._mail_ptr, self._flag_ptr]
            self.seq = torch.zeros(1, dtype=torch.int64, device=device)
            my = (bytes(mail_h), bytes(flag_h))
            handles: list = [None] * world
            dist.all_gather_object(handles, my, group=group)
            self.mail_ptrs, self.flag_ptrs = [], []
            for p, (mh, fh) in enumerate(handles):
                if p == rank:
                    self.mail_ptrs.append(self._mail_ptr)
                    self.flag_ptrs.append(self._flag_ptr)
                else:
                    mp = nat.ipc_open(mh)
                    fp = nat.ipc_open(fh)
                    self._opened += [mp, fp]
                    self.mail_ptrs.append(mp)
                    self.flag_ptrs.append(fp)
            self._nat = nat
            self.available = self._self_check()
            if not self.available:
                log.warning("p2p all-reduce self-check failed; using RCCL")
        except Exception as e:  # any setup failure -> RCCL
            log.warning("p2p all-reduce unavailable (%s); using RCCL", e)
            self.available = False

    def _self_check(self) -> bool:
        """Validate the full path against dist.all_reduce once at init."""
        x = torch.randn(4096, dtype=torch.bfloat16, device=self.device)
        x += self.rank  # rank-distinct payload
        mine = self(x.clone(), force=True)
        ref = x.clone()
        dist.all_reduce(ref, group=self.group)
        torch.cuda.synchronize()
        return bool(torch.allclose(mine.float(), ref.float(),
                                   atol=2e-2, rtol=2e-2))

    def usable(self, x: torch.Tensor) -> bool:
        return (self.available and x.is_cuda
                and x.dtype == torch.bfloat16
                and x.numel() <= MAX_ELEMS and x.numel() % 8 == 0)

    def __call__(self, x: torch.Tensor, force: bool = False) -> torch.Tensor:
        out = torch.empty_like(x)
        self._nat.one_shot_allreduce(
            out.view(-1), x.contiguous().view(-1), self.mail_ptrs,
            self.flag_ptrs, self.seq, self.rank,
        )
        return out.view_as(x)

    def close(self) -> None:
        for p in self._opened:
            try:
                self._nat.ipc_close(p)
            except Exception:
                pass
        self._opened = []
        for p in getattr(self, "_owned", []):
            try:
                self._nat.ipc_alloc_free(p)
            except Exception:
                pass
        self._owned = []
