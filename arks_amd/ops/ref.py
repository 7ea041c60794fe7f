"""Reference implementations of every arks_amd custom op, in plain PyTorch.

These are the numerics oracle for the HIP kernels (tests compare the gfx950
kernels against these run in fp32) and the compute path for CPU-only unit
tests of the engine/scheduler/model layers. They are NOT a runtime fallback:
on a GPU box the HIP extension is required and ops fail loudly without it
(see arks_amd/ops/__init__.py).

Conventions shared with the kernels:
- KV cache layout: [num_blocks, num_kv_heads, block_size, head_dim]
- block_tables: int32 [num_seqs, max_blocks_per_seq]
- slot_mapping: int64 flat slot index = block_id * block_size + offset
"""

from __future__ import annotations

import math

import torch


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    """RMSNorm over the last dim. Accumulates in fp32 like the kernel."""
    dtype = x.dtype
    xf = x.float()
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    out = xf * torch.rsqrt(var + eps) * weight.float()
    return out.to(dtype)


def fused_add_rmsnorm(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6
) -> tuple[torch.Tensor, torch.Tensor]:
    """residual += x; out = rmsnorm(residual). Returns (out, new_residual)."""
    new_residual = (residual.float() + x.float()).to(x.dtype)
    return rmsnorm(new_residual, weight, eps), new_residual


def silu_mul(gate_up: torch.Tensor) -> torch.Tensor:
    """SwiGLU activation: input [..., 2*d] as [gate | up] -> silu(gate) * up."""
    d = gate_up.shape[-1] // 2
    gate = gate_up[..., :d].float()
    up = gate_up[..., d:].float()
    return (torch.nn.functional.silu(gate) * up).to(gate_up.dtype)


def _scaled_inv_freq(inv_freq: torch.Tensor, rope_scaling: dict | None,
                     base: float, head_dim: int) -> tuple[torch.Tensor, float]:
    """Apply an HF rope_scaling config; returns (inv_freq, mscale).
    Supports the llama3 ramp, YaRN (NTK-by-parts + attention scaling) and
    plain linear scaling — matching transformers' rope_utils semantics."""
    if not rope_scaling:
        return inv_freq, 1.0
    rtype = rope_scaling.get("rope_type") or rope_scaling.get("type")
    factor = float(rope_scaling.get("factor", 1.0))
    if rtype == "linear":
        return inv_freq / factor, 1.0
    if rtype == "llama3":
        lo = float(rope_scaling.get("low_freq_factor", 1.0))
        hi = float(rope_scaling.get("high_freq_factor", 4.0))
        orig = float(rope_scaling.get("original_max_position_embeddings", 8192))
        wavelen = 2 * math.pi / inv_freq
        scaled = torch.where(wavelen > orig / lo, inv_freq / factor, inv_freq)
        smooth = (orig / wavelen - lo) / (hi - lo)
        mid = (1 - smooth) * inv_freq / factor + smooth * inv_freq
        is_mid = (wavelen <= orig / lo) & (wavelen >= orig / hi)
        return torch.where(is_mid, mid, scaled), 1.0
    if rtype == "yarn":
        orig = float(rope_scaling.get("original_max_position_embeddings", 4096))
        beta_fast = float(rope_scaling.get("beta_fast", 32.0))
        beta_slow = float(rope_scaling.get("beta_slow", 1.0))

        def corr_dim(num_rot: float) -> float:
            return (head_dim * math.log(orig / (num_rot * 2 * math.pi))) / (
                2 * math.log(base))

        low = max(math.floor(corr_dim(beta_fast)), 0)
        high = min(math.ceil(corr_dim(beta_slow)), head_dim // 2 - 1)
        ramp = torch.clamp(
            (torch.arange(head_dim // 2, dtype=torch.float64,
                          device=inv_freq.device) - low) / max(high - low, 1),
            0, 1,
        )
        # ramp==0 (high-freq dims) -> extrapolate (keep original);
        # ramp==1 (long wavelengths) -> interpolate (divide by factor)
        out = (inv_freq / factor) * ramp + inv_freq * (1 - ramp)
        mscale = float(rope_scaling.get("attention_factor") or
                       (0.1 * math.log(factor) + 1.0))
        return out, mscale
    return inv_freq, 1.0


def rope_cos_sin_cache(
    head_dim: int,
    max_positions: int,
    base: float = 10000.0,
    dtype: torch.dtype = torch.float32,
    device=None,
    rope_scaling: dict | None = None,
) -> torch.Tensor:
    """Precomputed [max_positions, head_dim] cache: first half cos, second half sin
    (per rotary pair), matching HF neox-style RoPE (incl. llama3 / yarn /
    linear rope_scaling)."""
    inv_freq = 1.0 / (
        base ** (torch.arange(0, head_dim, 2, dtype=torch.float64, device=device) / head_dim)
    )
    inv_freq, mscale = _scaled_inv_freq(inv_freq, rope_scaling, base, head_dim)
    t = torch.arange(max_positions, dtype=torch.float64, device=device)
    freqs = torch.outer(t, inv_freq)  # [P, head_dim/2]
    cache = torch.cat([freqs.cos() * mscale, freqs.sin() * mscale], dim=-1)
    return cache.to(dtype)


def rope_apply(
    positions: torch.Tensor,  # [num_tokens] int
    q: torch.Tensor,  # [num_tokens, num_q_heads * head_dim]
    k: torch.Tensor,  # [num_tokens, num_kv_heads * head_dim]
    cos_sin: torch.Tensor,  # [max_pos, head_dim]
    head_dim: int,
) -> tuple[torch.Tensor, torch.Tensor]:
    """Neox-style (rotate-half) RoPE applied out-of-place to q and k."""

    def _rot(x: torch.Tensor) -> torch.Tensor:
        n = x.shape[0]
        xf = x.float().view(n, -1, head_dim)
        half = head_dim // 2
        cs = cos_sin.float()[positions]  # [n, head_dim]
        cos = cs[:, :half].unsqueeze(1)  # [n, 1, half]
        sin = cs[:, half:].unsqueeze(1)
        x1 = xf[..., :half]
        x2 = xf[..., half:]
        o1 = x1 * cos - x2 * sin
        o2 = x2 * cos + x1 * sin
        return torch.cat([o1, o2], dim=-1).view(n, -1).to(x.dtype)

    return _rot(q), _rot(k)


def reshape_and_cache(
    k: torch.Tensor,  # [num_tokens, num_kv_heads, head_dim]
    v: torch.Tensor,
    k_cache: torch.Tensor,  # [num_blocks, num_kv_heads, block_size, head_dim]
    v_cache: torch.Tensor,
    slot_mapping: torch.Tensor,  # [num_tokens] int64
) -> None:
    block_size = k_cache.shape[2]
    block_ids = slot_mapping // block_size
    offsets = slot_mapping % block_size
    k_cache[block_ids, :, offsets] = k.to(k_cache.dtype)
    v_cache[block_ids, :, offsets] = v.to(v_cache.dtype)


def attention_prefill_varlen(
    q: torch.Tensor,  # [total_tokens, num_q_heads, head_dim]
    k: torch.Tensor,  # [total_tokens, num_kv_heads, head_dim]
    v: torch.Tensor,
    cu_seqlens: torch.Tensor,  # [num_seqs + 1] int32
    scale: float,
    window: int = 0,
) -> torch.Tensor:
    """Causal attention over packed variable-length sequences (full prefill:
    keys == the packed k/v of the same forward). GQA by head repetition.
    window > 0 = sliding-window attention."""
    num_q_heads = q.shape[1]
    num_kv_heads = k.shape[1]
    rep = num_q_heads // num_kv_heads
    out = torch.empty_like(q)
    for i in range(cu_seqlens.numel() - 1):
        s, e = int(cu_seqlens[i]), int(cu_seqlens[i + 1])
        qi = q[s:e].float().transpose(0, 1)  # [H, L, D]
        ki = k[s:e].float().repeat_interleave(rep, dim=1).transpose(0, 1)
        vi = v[s:e].float().repeat_interleave(rep, dim=1).transpose(0, 1)
        L = e - s
        scores = torch.matmul(qi, ki.transpose(-1, -2)) * scale  # [H, L, L]
        mask = torch.triu(
            torch.full((L, L), float("-inf"), device=q.device), diagonal=1
        )
        if window > 0:
            mask = mask + torch.tril(
                torch.full((L, L), float("-inf"), device=q.device),
                diagonal=-window,
            )
        scores = scores + mask
        p = torch.softmax(scores, dim=-1)
        o = torch.matmul(p, vi)  # [H, L, D]
        out[s:e] = o.transpose(0, 1).to(q.dtype)
    return out


def attention_decode_paged(
    q: torch.Tensor,  # [num_seqs, num_q_heads, head_dim]
    k_cache: torch.Tensor,  # [num_blocks, num_kv_heads, block_size, head_dim]
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,  # [num_seqs, max_blocks] int32
    seq_lens: torch.Tensor,  # [num_seqs] int32 (length INCLUDING current token)
    scale: float,
    window: int = 0,
) -> torch.Tensor:
    """Single-token decode attention against the paged KV cache.
    window > 0 = sliding-window (the query sees the last `window` keys)."""
    num_seqs, num_q_heads, head_dim = q.shape
    num_kv_heads = k_cache.shape[1]
    block_size = k_cache.shape[2]
    rep = num_q_heads // num_kv_heads
    out = torch.empty_like(q)
    for i in range(num_seqs):
        L = int(seq_lens[i])
        nblocks = (L + block_size - 1) // block_size
        bt = block_tables[i, :nblocks].long()
        # gather [L, num_kv_heads, head_dim]
        ks = k_cache[bt].permute(0, 2, 1, 3).reshape(nblocks * block_size, num_kv_heads, head_dim)[:L]
        vs = v_cache[bt].permute(0, 2, 1, 3).reshape(nblocks * block_size, num_kv_heads, head_dim)[:L]
        kf = ks.float().repeat_interleave(rep, dim=1)  # [L, H, D]
        vf = vs.float().repeat_interleave(rep, dim=1)
        qi = q[i].float()  # [H, D]
        scores = torch.einsum("hd,lhd->hl", qi, kf) * scale
        if window > 0 and L > window:
            scores[:, : L - window] = float("-inf")
        p = torch.softmax(scores, dim=-1)
        o = torch.einsum("hl,lhd->hd", p, vf)
        out[i] = o.to(q.dtype)
    return out


def attention_extend_paged(
    q: torch.Tensor,  # [total_new_tokens, num_q_heads, head_dim]
    k_cache: torch.Tensor,  # [num_blocks, num_kv_heads, block_size, head_dim]
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,  # [num_seqs, max_blocks] int32
    kv_lens: torch.Tensor,  # [num_seqs] int32 total kv length per seq
    cu_seqlens_q: torch.Tensor,  # [num_seqs + 1] int32 (packed new tokens)
    scale: float,
    window: int = 0,
) -> torch.Tensor:
    """Causal attention of each sequence's new tokens over its full paged KV
    (prefix caching / chunked prefill). New token j of seq i sits at global
    position kv_len - q_len + j. window > 0 = sliding-window attention
    (each query sees only the last `window` key positions)."""
    num_q_heads, head_dim = q.shape[1], q.shape[2]
    num_kv_heads = k_cache.shape[1]
    block_size = k_cache.shape[2]
    rep = num_q_heads // num_kv_heads
    out = torch.empty_like(q)
    for i in range(cu_seqlens_q.numel() - 1):
        s, e = int(cu_seqlens_q[i]), int(cu_seqlens_q[i + 1])
        qn = e - s
        L = int(kv_lens[i])
        off = L - qn
        nblocks = (L + block_size - 1) // block_size
        bt = block_tables[i, :nblocks].long()
        ks = k_cache[bt].permute(0, 2, 1, 3).reshape(nblocks * block_size, num_kv_heads, head_dim)[:L]
        vs = v_cache[bt].permute(0, 2, 1, 3).reshape(nblocks * block_size, num_kv_heads, head_dim)[:L]
        kf = ks.float().repeat_interleave(rep, dim=1).transpose(0, 1)  # [H, L, D]
        vf = vs.float().repeat_interleave(rep, dim=1).transpose(0, 1)
        qi = q[s:e].float().transpose(0, 1)  # [H, qn, D]
        scores = torch.matmul(qi, kf.transpose(-1, -2)) * scale  # [H, qn, L]
        kpos = torch.arange(L, device=q.device)
        qpos = torch.arange(off, L, device=q.device)
        mask = kpos[None, :] > qpos[:, None]  # [qn, L] True = hidden
        if window > 0:
            mask |= kpos[None, :] <= qpos[:, None] - window
        scores = scores.masked_fill(mask, float("-inf"))
        p = torch.softmax(scores, dim=-1)
        o = torch.matmul(p, vf)  # [H, qn, D]
        out[s:e] = o.transpose(0, 1).to(q.dtype)
    return out


def moe_topk(router_logits: torch.Tensor, k: int, renorm: bool):
    """softmax + top-k (+ renorm): (weights [T,k] f32, ids [T,k] i32)."""
    probs = torch.softmax(router_logits.float(), dim=-1)
    weights, selected = probs.topk(k, dim=-1)
    if renorm:
        weights = weights / weights.sum(dim=-1, keepdim=True)
    return weights, selected.to(torch.int32)


def moe_mix(y: torch.Tensor, weights: torch.Tensor, ids: torch.Tensor,
            expert_base: int = 0) -> torch.Tensor:
    """Weighted mix of per-(local-)expert dense outputs y [El, T, H]."""
    El, T, H = y.shape
    le = ids.long() - expert_base
    mine = (le >= 0) & (le < El)
    le = le.clamp(0, El - 1)
    t_idx = torch.arange(T, device=y.device)[:, None].expand_as(le)
    gathered = y[le, t_idx].float()  # [T, k, H]
    w = torch.where(mine, weights, torch.zeros_like(weights))
    return torch.einsum("tkh,tk->th", gathered, w.float()).to(y.dtype)


def moe_mix_rows(y: torch.Tensor, weights: torch.Tensor,
                 rows: torch.Tensor) -> torch.Tensor:
    """out[t] = sum_j weights[t,j] * y[rows[t,j]] over a flat [R, H]
    expert-output buffer."""
    gathered = y[rows.long()].float()  # [T, k, H]
    return torch.einsum("tkh,tk->th", gathered, weights.float()).to(y.dtype)


def quant_fp8_rows(x: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
    """Per-row symmetric quantization to OCP e4m3 (range +-448)."""
    amax = x.float().abs().amax(dim=1).clamp(min=1e-6)
    scale = 448.0 / amax
    q = (x.float() * scale[:, None]).clamp(-448.0, 448.0).to(torch.float8_e4m3fn)
    return q, (1.0 / scale)


def greedy_sample(logits: torch.Tensor) -> torch.Tensor:
    """[num_seqs, vocab] -> [num_seqs] int64 argmax."""
    return logits.float().argmax(dim=-1)


def gumbel_sample(
    logits: torch.Tensor,  # [num_seqs, vocab]
    temperatures: torch.Tensor,  # [num_seqs] float; 0 => greedy
    uniform: torch.Tensor,  # [num_seqs, vocab] uniform(0,1) noise
) -> torch.Tensor:
    """Exact categorical sampling via the Gumbel-max trick:
    argmax(logits/T + Gumbel) ~ softmax(logits/T). T==0 rows are greedy."""
    lf = logits.float()
    t = temperatures.float().clamp(min=0.0).unsqueeze(1)
    gumbel = -torch.log(-torch.log(uniform.float().clamp(1e-20, 1.0)))
    scaled = torch.where(t > 0, lf / t.clamp(min=1e-8) + gumbel, lf)
    return scaled.argmax(dim=-1)


def softmax_scale(head_dim: int) -> float:
    return 1.0 / math.sqrt(head_dim)
