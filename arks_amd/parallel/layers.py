"""Tensor-parallel linear layers (Megatron-style column/row split).

GEMMs go through F.linear (hipBLASLt on ROCm — the guide's 'plain library
GEMM' path); the fused hot ops around them are the hand-written kernels in
arks_amd.ops. Weight loading maps full HF tensors to local shards via each
layer's `shard()`.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from .comm import get_tp_rank, get_tp_world_size, tp_all_gather, tp_all_reduce


def quantize_module_fp8(mod: nn.Module) -> None:
    """Switch a linear layer to W8A8 fp8 (OCP e4m3, per-output-channel weight
    scales, dynamic per-token activation scales — ops.quant_fp8_rows).
    The bf16 weight parameter is dropped on CUDA to free HBM; on CPU a
    quant-dequant copy emulates the GPU numerics for tests."""
    w = mod.weight.data
    amax = w.float().abs().amax(dim=1).clamp(min=1e-6)
    scale = 448.0 / amax
    w8 = (w.float() * scale[:, None]).clamp(-448.0, 448.0).to(torch.float8_e4m3fn)
    mod.register_buffer("weight_fp8", w8, persistent=False)
    mod.register_buffer(
        "w_inv_scale", (1.0 / scale).float()[None, :], persistent=False
    )
    if w.is_cuda:
        del mod._parameters["weight"]
        mod.weight = None
    else:
        mod.register_buffer(
            "weight_qdq",
            (w8.float() * (1.0 / scale)[:, None]).to(w.dtype),
            persistent=False,
        )
    mod.fp8 = True


def _fp8_linear_prequant(mod: nn.Module, x8, sa) -> torch.Tensor:
    """fp8 GEMM on an already-quantized activation (the producing kernel —
    rmsnorm_fp8 / silu_mul_fp8 — emitted x8 + per-token scales)."""
    if x8.is_cuda:
        return torch._scaled_mm(
            x8, mod.weight_fp8.t(), scale_a=sa[:, None], scale_b=mod.w_inv_scale,
            bias=mod.bias, out_dtype=torch.bfloat16,
        )
    xd = (x8.float() * sa[:, None]).to(torch.bfloat16)
    return F.linear(xd, mod.weight_qdq, mod.bias)


def _fp8_linear(mod: nn.Module, x: torch.Tensor) -> torch.Tensor:
    from .. import ops

    if x.is_cuda:
        x8, sa = ops.quant_fp8_rows(x)
        return torch._scaled_mm(
            x8, mod.weight_fp8.t(), scale_a=sa[:, None], scale_b=mod.w_inv_scale,
            bias=mod.bias, out_dtype=x.dtype,
        )
    # CPU emulation: quant-dequant both operands, accumulate in f32
    x8, sa = ops.quant_fp8_rows(x)
    xd = (x8.float() * sa[:, None]).to(x.dtype)
    return F.linear(xd, mod.weight_qdq, mod.bias)


class ColumnParallelLinear(nn.Module):
    """Y = X W^T with W row-sharded over TP ranks (output features split).
    Output stays sharded (gather_output=False semantics)."""

    def __init__(self, in_features: int, out_features: int, bias: bool,
                 dtype: torch.dtype = torch.bfloat16):
        super().__init__()
        tp = get_tp_world_size()
        assert out_features % tp == 0, (out_features, tp)
        self.in_features = in_features
        self.out_features = out_features
        self.out_per_rank = out_features // tp
        self.weight = nn.Parameter(
            torch.empty(self.out_per_rank, in_features, dtype=dtype),
            requires_grad=False,
        )
        self.bias = (
            nn.Parameter(torch.empty(self.out_per_rank, dtype=dtype), requires_grad=False)
            if bias
            else None
        )

    def shard(self, full: torch.Tensor, param: str = "weight") -> torch.Tensor:
        r = get_tp_rank()
        return full[r * self.out_per_rank : (r + 1) * self.out_per_rank]

    fp8 = False

    def forward(self, x) -> torch.Tensor:
        if isinstance(x, tuple):  # pre-quantized by the producing kernel
            return _fp8_linear_prequant(self, *x)
        if self.fp8:
            return _fp8_linear(self, x)
        from .. import ops

        return ops.linear_bf16(x, self.weight, self.bias)


class QKVParallelLinear(ColumnParallelLinear):
    """Fused QKV projection; q/k/v head groups are sharded per rank.
    Weight layout per rank: [q_shard | k_shard | v_shard]."""

    def __init__(self, hidden: int, head_dim: int, num_q_heads: int,
                 num_kv_heads: int, bias: bool, dtype=torch.bfloat16):
        tp = get_tp_world_size()
        assert num_q_heads % tp == 0 and num_kv_heads % tp == 0
        self.head_dim = head_dim
        self.nq, self.nkv = num_q_heads, num_kv_heads
        self.nq_local = num_q_heads // tp
        self.nkv_local = num_kv_heads // tp
        out = (self.nq_local + 2 * self.nkv_local) * head_dim * tp  # per-rank x tp
        super().__init__(hidden, out, bias, dtype)

    def shard_qkv_parts(self, q_full, k_full, v_full) -> list[torch.Tensor]:
        """This rank's q/k/v slices in fused order (views — callers copy
        each part straight into the fused parameter, no host-side cat)."""
        r = get_tp_rank()
        hd = self.head_dim
        return [
            q_full[r * self.nq_local * hd: (r + 1) * self.nq_local * hd],
            k_full[r * self.nkv_local * hd: (r + 1) * self.nkv_local * hd],
            v_full[r * self.nkv_local * hd: (r + 1) * self.nkv_local * hd],
        ]

    def shard_qkv(self, q_full, k_full, v_full) -> torch.Tensor:
        """Build this rank's fused weight (or bias) from full q/k/v tensors."""
        return torch.cat(self.shard_qkv_parts(q_full, k_full, v_full), dim=0)

    def split_qkv(self, qkv: torch.Tensor):
        hd = self.head_dim
        q, k, v = qkv.split(
            [self.nq_local * hd, self.nkv_local * hd, self.nkv_local * hd], dim=-1
        )
        return q, k, v


class MergedColumnParallelLinear(ColumnParallelLinear):
    """Two column-parallel projections fused (gate_proj|up_proj)."""

    def __init__(self, in_features: int, each_out: int, bias: bool,
                 dtype=torch.bfloat16):
        self.each_out = each_out
        super().__init__(in_features, 2 * each_out, bias, dtype)

    def shard_merged_parts(self, gate_full, up_full) -> list[torch.Tensor]:
        r = get_tp_rank()
        tp = get_tp_world_size()
        per = self.each_out // tp
        return [gate_full[r * per: (r + 1) * per],
                up_full[r * per: (r + 1) * per]]

    def shard_merged(self, gate_full: torch.Tensor, up_full: torch.Tensor) -> torch.Tensor:
        return torch.cat(self.shard_merged_parts(gate_full, up_full), dim=0)


class RowParallelLinear(nn.Module):
    """Y = sum_ranks X_shard W_shard^T, W column-sharded (input features
    split); all-reduce over the TP group after the local GEMM."""

    def __init__(self, in_features: int, out_features: int, bias: bool,
                 dtype: torch.dtype = torch.bfloat16):
        super().__init__()
        tp = get_tp_world_size()
        assert in_features % tp == 0, (in_features, tp)
        self.in_per_rank = in_features // tp
        self.weight = nn.Parameter(
            torch.empty(out_features, self.in_per_rank, dtype=dtype),
            requires_grad=False,
        )
        # bias applied once (post-reduce) on every rank identically
        self.bias = (
            nn.Parameter(torch.empty(out_features, dtype=dtype), requires_grad=False)
            if bias
            else None
        )

    def shard(self, full: torch.Tensor, param: str = "weight") -> torch.Tensor:
        if param == "bias":
            return full
        r = get_tp_rank()
        return full[:, r * self.in_per_rank : (r + 1) * self.in_per_rank]

    fp8 = False

    def forward(self, x) -> torch.Tensor:
        if isinstance(x, tuple) or self.fp8:
            bias = self.bias
            self.bias = None  # bias must be applied post-reduce, once
            try:
                y = (_fp8_linear_prequant(self, *x) if isinstance(x, tuple)
                     else _fp8_linear(self, x))
            finally:
                self.bias = bias
        else:
            from .. import ops

            y = ops.linear_bf16(x, self.weight)
        y = tp_all_reduce(y)
        if self.bias is not None:
            y = y + self.bias
        return y


class ParallelLMHead(nn.Module):
    """Vocab-sharded output projection; logits are all-gathered so every rank
    samples identically."""

    def __init__(self, hidden: int, vocab: int, dtype=torch.bfloat16):
        super().__init__()
        tp = get_tp_world_size()
        # pad vocab shard up so it divides evenly
        self.vocab = vocab
        self.vocab_padded = ((vocab + tp - 1) // tp) * tp
        self.per_rank = self.vocab_padded // tp
        self.weight = nn.Parameter(
            torch.empty(self.per_rank, hidden, dtype=dtype), requires_grad=False
        )

    def shard(self, full: torch.Tensor, param: str = "weight") -> torch.Tensor:
        r = get_tp_rank()
        lo, hi = r * self.per_rank, (r + 1) * self.per_rank
        if hi <= full.shape[0]:
            return full[lo:hi]
        pad = torch.full(
            (hi - min(hi, full.shape[0]), full.shape[1]), 0, dtype=full.dtype
        )
        return torch.cat([full[lo:], pad], dim=0)

    fp8 = False

    def forward(self, x) -> torch.Tensor:
        if isinstance(x, tuple) or self.fp8:
            self.bias = None
            logits = (_fp8_linear_prequant(self, *x) if isinstance(x, tuple)
                      else _fp8_linear(self, x))
        else:
            from .. import ops

            logits = ops.linear_bf16(x, self.weight)
        logits = tp_all_gather(logits, dim=-1)
        return logits[..., : self.vocab]
