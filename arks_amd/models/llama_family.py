"""Qwen2 / Llama decoder family for the arks_amd engine.

One implementation covers both architectures (Qwen2 = Llama + QKV bias +
different rope_theta/vocab). HF-config-driven (arks_amd.config.ModelConfig);
weight names follow the HF checkpoint layout so safetensors load directly.

Compute path: F.linear GEMMs (hipBLASLt) + arks_amd.ops HIP kernels for
RMSNorm / RoPE / paged attention / SwiGLU; TP via column/row-parallel shards
with one RCCL all-reduce per attention and per MLP block.
"""

from __future__ import annotations

import math

import torch
import torch.nn as nn

from .. import ops
from ..config import ModelConfig
from ..engine.forward_batch import ForwardBatch
from ..parallel.comm import get_tp_rank, get_tp_world_size
from ..parallel.layers import (
    MergedColumnParallelLinear,
    ParallelLMHead,
    QKVParallelLinear,
    RowParallelLinear,
)


class RMSNorm(nn.Module):
    fp8_out = False  # emit (fp8, scales) for a W8A8 consumer GEMM

    def __init__(self, hidden: int, eps: float, dtype=torch.bfloat16):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden, dtype=dtype), requires_grad=False)
        self.eps = eps

    def forward(self, x, residual=None):
        if self.fp8_out:
            if residual is None:
                return ops.rmsnorm_fp8(x, self.weight, self.eps)
            q, s, res = ops.fused_add_rmsnorm_fp8(x, residual, self.weight, self.eps)
            return (q, s), res
        if residual is None:
            return ops.rmsnorm(x, self.weight, self.eps)
        return ops.fused_add_rmsnorm(x, residual, self.weight, self.eps)


class Attention(nn.Module):
    def __init__(self, cfg: ModelConfig, dtype=torch.bfloat16,
                 layer_idx: int = 0):
        super().__init__()
        tp = get_tp_world_size()
        self.head_dim = cfg.head_dim
        self.nq_local = cfg.num_attention_heads // tp
        self.nkv_local = cfg.num_key_value_heads // tp
        self.scale = 1.0 / math.sqrt(cfg.head_dim)
        # per-layer sliding window (HF Qwen2 max_window_layers/layer_types
        # semantics; Mistral = every layer; 0 = full attention). Masking
        # happens inside the kernels.
        self.window = cfg.layer_window(layer_idx)
        # SmolLM3 NoPE layers attend without positional encoding: k/v are
        # cached unrotated (HF modeling_smollm3.py:200-226 semantics)
        self.use_rope = cfg.layer_uses_rope(layer_idx)
        self.qkv_proj = QKVParallelLinear(
            cfg.hidden_size, cfg.head_dim, cfg.num_attention_heads,
            cfg.num_key_value_heads, bias=cfg.attention_bias, dtype=dtype,
        )
        self.o_proj = RowParallelLinear(
            cfg.num_attention_heads * cfg.head_dim, cfg.hidden_size, bias=False,
            dtype=dtype,
        )
        # Qwen3: per-head RMSNorm on q and k before RoPE
        self.q_norm = (
            RMSNorm(cfg.head_dim, cfg.rms_norm_eps, dtype) if cfg.qk_norm else None
        )
        self.k_norm = (
            RMSNorm(cfg.head_dim, cfg.rms_norm_eps, dtype) if cfg.qk_norm else None
        )

    def forward(self, x, batch: ForwardBatch, kv_cache) -> torch.Tensor:
        T = (x[0] if isinstance(x, tuple) else x).shape[0]
        qkv = self.qkv_proj(x)
        # Strided views into the fused QKV buffer — RoPE/cache/attention
        # kernels take row strides, so no .contiguous() copies on the hot path.
        q, k, v = self.qkv_proj.split_qkv(qkv)
        if self.q_norm is not None:
            hd = self.head_dim
            q = self.q_norm(
                q.contiguous().view(-1, hd)).view(T, self.nq_local * hd)
            k = self.k_norm(
                k.contiguous().view(-1, hd)).view(T, self.nkv_local * hd)
            v = v.contiguous()  # reshape_and_cache wants one shared kv stride
        k_cache, v_cache = kv_cache
        if self.use_rope:
            q, k = ops.rope_and_cache(
                batch.positions, q, k, v, k_cache, v_cache,
                batch.slot_mapping, self._cos_sin, self.head_dim,
            )
        else:
            # NoPE layer: cache k/v as-is, q unrotated
            ops.reshape_and_cache(
                k.unflatten(-1, (self.nkv_local, self.head_dim)),
                v.unflatten(-1, (self.nkv_local, self.head_dim)),
                k_cache, v_cache, batch.slot_mapping,
            )
        k = k.unflatten(-1, (self.nkv_local, self.head_dim))
        v = v.unflatten(-1, (self.nkv_local, self.head_dim))
        q = q.unflatten(-1, (self.nq_local, self.head_dim))
        if batch.is_prefill:
            if batch.block_tables is not None:
                # Prefix-cache hit in the batch: new tokens attend over the
                # full paged KV (cached prefix + the rows just written above).
                out = ops.attention_extend_paged(
                    q, k_cache, v_cache, batch.block_tables, batch.seq_lens,
                    batch.cu_seqlens, batch.seq_lens_list, self.scale,
                    window=self.window, tiles=batch.ext_tiles,
                )
            else:
                out = ops.attention_prefill_varlen(
                    q, k, v, batch.cu_seqlens, batch.seq_lens_list, self.scale,
                    window=self.window,
                )
        else:
            out = ops.attention_decode_paged(
                q, k_cache, v_cache, batch.block_tables, batch.seq_lens,
                self.scale, window=self.window,
            )
        return self.o_proj(out.view(T, -1))


class MLP(nn.Module):
    def __init__(self, cfg: ModelConfig, dtype=torch.bfloat16):
        super().__init__()
        self.gate_up_proj = MergedColumnParallelLinear(
            cfg.hidden_size, cfg.intermediate_size, bias=False, dtype=dtype
        )
        self.down_proj = RowParallelLinear(
            cfg.intermediate_size, cfg.hidden_size, bias=False, dtype=dtype
        )

    fp8 = False

    def forward(self, x):
        gate_up = self.gate_up_proj(x)
        if self.fp8:
            return self.down_proj(ops.silu_mul_fp8(gate_up))
        return self.down_proj(ops.silu_mul(gate_up))


_ARANGE_CACHE: dict[tuple[int, int, str], torch.Tensor] = {}


def _arange_interleave(T: int, k: int, device) -> torch.Tensor:
    """Cached [0,0,..,1,1,..] token-index vector — identical for every MoE
    layer of a forward, so build it once per (T, k) instead of 48 times."""
    key = (T, k, str(device))
    t = _ARANGE_CACHE.get(key)
    if t is None:
        if len(_ARANGE_CACHE) > 256:
            _ARANGE_CACHE.clear()
        t = torch.arange(T, device=device).repeat_interleave(k)
        _ARANGE_CACHE[key] = t
    return t


class MoEMLP(nn.Module):
    """Mixtral-style sparse MoE block, expert-parallel over the TP group:
    each rank owns num_local_experts/tp whole experts (an expert's FFN is
    never split), computes its experts' contributions for the tokens routed
    to them, and one all-reduce sums the partial outputs — the same single
    collective per block as the dense RowParallel MLP. The router (gate) is
    replicated so every rank routes identically."""

    fp8 = False  # MoE experts stay bf16 (v1)

    def __init__(self, cfg: ModelConfig, dtype=torch.bfloat16):
        super().__init__()
        tp = get_tp_world_size()
        E = cfg.num_local_experts
        assert E % tp == 0, (E, tp)
        self.num_experts = E
        self.top_k = cfg.num_experts_per_tok
        self.local_experts = E // tp
        self.expert_base = get_tp_rank() * self.local_experts
        self.hidden = cfg.hidden_size
        self.inter = cfg.moe_intermediate_size or cfg.intermediate_size
        self.norm_topk = cfg.norm_topk_prob
        self.gate = nn.Parameter(
            torch.empty(E, cfg.hidden_size, dtype=dtype), requires_grad=False
        )
        # fused [gate|up] and down per local expert, stored PRE-TRANSPOSED
        # ([in, out] per expert) so every MoE GEMM is an NN strided bmm /
        # mm with contiguous operands: strided bmm with a transposed-VIEW B
        # memory-faults on this ROCm stack at prefill shapes
        # (scripts/probe_bmm_fault.py — all tb=True variants die, contiguous
        # B passes), and NN needs no per-call transpose copies.
        self.w13 = nn.Parameter(
            torch.empty(self.local_experts, self.hidden, 2 * self.inter,
                        dtype=dtype), requires_grad=False,
        )
        self.w2 = nn.Parameter(
            torch.empty(self.local_experts, self.inter, self.hidden,
                        dtype=dtype), requires_grad=False,
        )
        # Qwen2-MoE shared expert (dense, TP-sharded like a normal MLP) with
        # a sigmoid scalar gate
        self.shared = None
        self.shared_gate = None
        if cfg.shared_expert_intermediate_size:
            import dataclasses

            dense_cfg = dataclasses.replace(
                cfg, intermediate_size=cfg.shared_expert_intermediate_size
            )
            self.shared = MLP(dense_cfg, dtype)
            self.shared_gate = nn.Parameter(
                torch.zeros(1, cfg.hidden_size, dtype=dtype),
                requires_grad=False,
            )

    # Below this many tokens the dense path wins: every expert's weights
    # are streamed from HBM regardless (tokens scatter over all experts),
    # so decode is weight-bandwidth-bound and the extra MFMA work of
    # computing all tokens per expert stays at or under the HBM floor up
    # to ~256 tokens (E*T*3*I*H*2 flops vs E*3*I*H*2 bytes: the crossover
    # at ~1.5 PF effective and ~6 TB/s is T ~ 250). Data-independent
    # shapes make every decode batch size hipGraph-capturable.
    DENSE_TOKENS = 256

    def forward(self, x):
        from ..parallel.comm import tp_all_reduce

        T = x.shape[0]
        router_logits = torch.nn.functional.linear(x, self.gate).float()
        # fused softmax+topk+renorm routing kernel (ops.moe_topk) replaces
        # a 5-kernel torch chain per layer
        weights, selected = ops.moe_topk(router_logits, self.top_k,
                                         self.norm_topk)
        out = torch.zeros_like(x)
        if T <= self.DENSE_TOKENS:
            # Batched over experts: two strided-batch GEMMs (bmm) instead of
            # a per-expert loop — one launch pair regardless of E (Qwen3-MoE
            # has 128 experts), and every expert's weights stream once. The
            # weighted mix over experts is one gather kernel (ops.moe_mix).
            E = self.local_experts
            xb = x.unsqueeze(0).expand(E, T, self.hidden)
            gu = torch.bmm(xb, self.w13)  # [E, T, 2I]
            h = ops.silu_mul(gu.reshape(E * T, 2 * self.inter))
            y = torch.bmm(h.view(E, T, self.inter), self.w2)  # [E, T, H]
            out = ops.moe_mix(y, weights, selected, self.expert_base)
            out = tp_all_reduce(out)
            if self.shared is not None:
                gate = torch.sigmoid(
                    torch.nn.functional.linear(x, self.shared_gate).float()
                ).to(x.dtype)
                out = out + gate * self.shared(x)
            return out
        # Sparse path (prefill-sized T): sort token-expert pairs once so
        # each expert sees a contiguous segment (one host sync per layer
        # for the segment table), run the expert FFNs into ONE flat output
        # buffer — a capacity-clamped grouped bmm plus large-M overflow
        # GEMMs for heavily-routed experts — and combine with the
        # moe_mix_rows gather kernel. No index_add_ scatters anywhere:
        # torch's indexFuncLargeIndex was 36% of Qwen3-30B prefill GPU
        # time (profiles/r02_final.md).
        k = self.top_k
        flat_sel = selected.reshape(-1).long()
        flat_tok = _arange_interleave(T, k, x.device)
        order = torch.argsort(flat_sel, stable=True)
        tok_sorted = flat_tok[order]
        counts = torch.bincount(flat_sel, minlength=self.num_experts)
        offs = torch.cumsum(counts, 0)
        seg = torch.stack((counts, offs)).cpu()  # ONE host sync per layer
        counts_h = seg[0].tolist()
        offs_h = seg[1].tolist()
        El = self.local_experts
        base = self.expert_base
        local_counts_h = counts_h[base:base + El]
        cap = max(local_counts_h or [0])
        cap_lim = max(1, (512 * 1024 * 1024) // (El * 2 * self.inter * 2))
        # clamp the grouped capacity near 2x the mean load: under skewed
        # routing, padding the whole block to the heaviest expert wastes
        # multiples of the useful FLOPs, while spilled experts finish in
        # efficient large-M overflow GEMMs below
        mean2 = 2 * ((sum(local_counts_h) + El - 1) // El) if El else 0
        cap_eff = min(cap, cap_lim, max(128, mean2))
        ovf = [(le, c - cap_eff) for le, c in enumerate(local_counts_h)
               if c > cap_eff]
        n_of = sum(o for _, o in ovf)
        if cap_eff == 0:
            out = tp_all_reduce(out)  # no pairs routed to this rank
            if self.shared is not None:
                gate = torch.sigmoid(
                    torch.nn.functional.linear(x, self.shared_gate).float()
                ).to(x.dtype)
                out = out + gate * self.shared(x)
            return out

        y_all = x.new_empty(El * cap_eff + n_of, self.hidden)
        # grouped part: pad each local expert's segment to cap_eff rows
        # (two strided-batch GEMMs for the whole block; padding rows index
        # row 0 and are never gathered)
        local_counts = counts[base:base + El, None]        # [El, 1]
        local_starts = (offs[base:base + El] -
                        counts[base:base + El])[:, None]   # [El, 1]
        capped = torch.clamp(local_counts, max=cap_eff)
        ar = torch.arange(cap_eff, device=x.device)[None, :]  # [1, cap]
        flat = torch.where(ar < capped, local_starts + ar,
                           torch.zeros_like(ar)).reshape(-1)
        # single fused gather x[tok_sorted[flat]] — materializing an
        # intermediate x_g[T*k, H] costs a full extra round trip per layer
        xp = x[tok_sorted[flat]].view(El, cap_eff, self.hidden)
        gu = torch.bmm(xp, self.w13)
        h = ops.silu_mul(gu.reshape(El * cap_eff, 2 * self.inter))
        torch.bmm(h.view(El, cap_eff, self.inter), self.w2,
                  out=y_all[:El * cap_eff].view(El, cap_eff, self.hidden))
        # overflow part: experts past cap_eff finish with plain large-M
        # GEMMs appended to the same buffer
        p0 = El * cap_eff
        of_base_h = []
        acc = 0
        for le, o in ovf:
            of_base_h.append((le, acc))
            ge = base + le
            sl = slice(offs_h[ge] - counts_h[ge] + cap_eff, offs_h[ge])
            h2 = ops.silu_mul(x[tok_sorted[sl]] @ self.w13[le])
            torch.mm(h2, self.w2[le], out=y_all[p0 + acc:p0 + acc + o])
            acc += o

        # row map [T, k]: each pair's position in y_all
        inv = torch.empty_like(order)
        inv[order] = torch.arange(T * k, device=x.device)
        inv = inv.view(T, k)
        sel = selected.long().view(T, k)
        sel_local = sel - base
        mine = (sel_local >= 0) & (sel_local < El)
        seg_start = offs - counts  # [E] global segment starts
        local_idx = inv - seg_start[sel]
        in_cap = local_idx < cap_eff
        sl_safe = sel_local.clamp(0, El - 1)
        if ovf:
            of_base_t = torch.zeros(El, dtype=torch.long, device=x.device)
            of_base_t[torch.tensor([le for le, _ in of_base_h],
                                   device=x.device)] = torch.tensor(
                [a for _, a in of_base_h], device=x.device)
            rows_of = p0 + of_base_t[sl_safe] + (local_idx - cap_eff)
        else:
            rows_of = torch.zeros_like(local_idx)
        rows = torch.where(in_cap, sl_safe * cap_eff + local_idx, rows_of)
        rows = torch.where(mine, rows, torch.zeros_like(rows))
        w_masked = torch.where(mine, weights,
                               torch.zeros_like(weights))
        out = ops.moe_mix_rows(y_all, w_masked, rows.to(torch.int32))
        out = tp_all_reduce(out)
        if self.shared is not None:
            gate = torch.sigmoid(
                torch.nn.functional.linear(x, self.shared_gate).float()
            ).to(x.dtype)
            out = out + gate * self.shared(x)
        return out


class DecoderLayer(nn.Module):
    def __init__(self, cfg: ModelConfig, dtype=torch.bfloat16,
                 layer_idx: int = 0):
        super().__init__()
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, dtype)
        self.self_attn = Attention(cfg, dtype, layer_idx=layer_idx)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, dtype)
        self.mlp = (
            MoEMLP(cfg, dtype) if cfg.num_local_experts > 0 else MLP(cfg, dtype)
        )

    def forward(self, x, residual, batch: ForwardBatch, kv_cache):
        if residual is None:
            residual = x
            x = self.input_layernorm(x)
        else:
            x, residual = self.input_layernorm(x, residual)
        x = self.self_attn(x, batch, kv_cache)
        x, residual = self.post_attention_layernorm(x, residual)
        x = self.mlp(x)
        return x, residual


class LlamaFamilyForCausalLM(nn.Module):
    """Covers Qwen2ForCausalLM and LlamaForCausalLM."""

    def __init__(self, cfg: ModelConfig, dtype=torch.bfloat16):
        super().__init__()
        self.cfg = cfg
        self.dtype = dtype
        self.embed_tokens = nn.Embedding(
            cfg.vocab_size, cfg.hidden_size, dtype=dtype
        )
        self.embed_tokens.weight.requires_grad_(False)
        self.layers = nn.ModuleList(
            DecoderLayer(cfg, dtype, layer_idx=i)
            for i in range(cfg.num_hidden_layers)
        )
        self.norm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, dtype)
        self.lm_head = ParallelLMHead(cfg.hidden_size, cfg.vocab_size, dtype)
        cos_sin = self._build_cos_sin()
        self.register_buffer("rope_cos_sin", cos_sin, persistent=False)
        for layer in self.layers:
            layer.self_attn._cos_sin = cos_sin

    def quantize_fp8(self) -> None:
        """W8A8 fp8 for the weight-bandwidth-bound GEMMs (profiles/r01_p2:
        gate_up/down/lm_head dominate the decode step; o_proj is left bf16 —
        no gain at its size and its input is the attention output, which
        has no producing kernel to fuse a quant into)."""
        from ..parallel.layers import quantize_module_fp8

        for layer in self.layers:
            quantize_module_fp8(layer.self_attn.qkv_proj)
            layer.input_layernorm.fp8_out = True
            if isinstance(layer.mlp, MoEMLP):
                continue  # MoE experts stay bf16 (v1)
            quantize_module_fp8(layer.mlp.gate_up_proj)
            quantize_module_fp8(layer.mlp.down_proj)
            # producing kernels emit fp8 directly (fused activation quant)
            layer.post_attention_layernorm.fp8_out = True
            layer.mlp.fp8 = True
        quantize_module_fp8(self.lm_head)
        self.norm.fp8_out = True

    def _build_cos_sin(self, min_len: int | None = None):
        from ..ops import ref

        return ref.rope_cos_sin_cache(
            self.cfg.head_dim,
            max(self.cfg.max_position_embeddings, min_len or 0),
            self.cfg.rope_theta, rope_scaling=self.cfg.rope_scaling,
        )

    def extend_rope_table(self, min_len: int) -> None:
        """Grow the RoPE cos/sin table to at least `min_len` positions
        (serving past max_position_embeddings would otherwise read out of
        bounds in the rope kernel; plain RoPE extrapolates)."""
        if self.rope_cos_sin.shape[0] >= min_len:
            return
        cache = self._build_cos_sin(min_len).to(
            device=self.rope_cos_sin.device, dtype=self.rope_cos_sin.dtype
        )
        self.register_buffer("rope_cos_sin", cache, persistent=False)
        for layer in self.layers:
            layer.self_attn._cos_sin = cache

    def _apply(self, fn, recurse=True):  # keep the shared cos_sin in sync
        out = super()._apply(fn, recurse)
        for layer in self.layers:
            layer.self_attn._cos_sin = self.rope_cos_sin
        return out

    def forward(self, batch: ForwardBatch, kv_caches: list) -> torch.Tensor:
        """Returns logits for batch.logits_indices rows."""
        x = self.embed_tokens(batch.input_ids)
        residual = None
        for i, layer in enumerate(self.layers):
            x, residual = layer(x, residual, batch, kv_caches[i])
        x, _ = self.norm(x, residual)
        if isinstance(x, tuple):  # fp8: (values, per-token scales)
            q8, sc = x
            if batch.logits_indices is not None:
                q8 = q8[batch.logits_indices]
                sc = sc[batch.logits_indices]
            return self.lm_head((q8, sc))
        if batch.logits_indices is not None:
            x = x[batch.logits_indices]
        return self.lm_head(x)

    # ------------------------- weight loading -------------------------
    def load_hf_state_dict(self, tensors: dict[str, torch.Tensor]) -> None:
        """Load an HF-layout state dict (full tensors; sharded here)."""
        pending_qkv: dict[int, dict] = {}
        pending_mlp: dict[int, dict] = {}
        own = dict(self.named_parameters())

        def put(name, value):
            p = own[name]
            assert p.shape == value.shape, (name, p.shape, value.shape)
            p.data.copy_(value.to(p.dtype))

        def put_cat(name, parts):
            # fused params load piecewise: each (pinned) slice goes straight
            # into its span of the parameter — no host-side torch.cat copy
            p = own[name]
            off = 0
            for part in parts:
                n = part.shape[0]
                p.data[off:off + n].copy_(part.to(p.dtype))
                off += n
            assert off == p.shape[0], (name, off, p.shape)

        for name, w in tensors.items():
            name = name.removeprefix("model.")
            if name == "embed_tokens.weight":
                put("embed_tokens.weight", w)
                if self.cfg.tie_word_embeddings:
                    put("lm_head.weight", self.lm_head.shard(w))
            elif name in ("lm_head.weight",):
                put("lm_head.weight", self.lm_head.shard(w))
            elif name == "norm.weight":
                put("norm.weight", w)
            elif name.startswith("layers."):
                parts = name.split(".")
                li = int(parts[1])
                sub = ".".join(parts[2:])
                layer = self.layers[li]
                if sub in ("input_layernorm.weight", "post_attention_layernorm.weight"):
                    put(f"layers.{li}.{sub}", w)
                elif sub in ("self_attn.q_norm.weight", "self_attn.k_norm.weight"):
                    put(f"layers.{li}.{sub}", w)
                elif sub.startswith("self_attn.") and sub.split(".")[1] in (
                    "q_proj", "k_proj", "v_proj",
                ):
                    proj, param = sub.split(".")[1], sub.split(".")[2]
                    pending_qkv.setdefault(li, {})[f"{proj}.{param}"] = w
                    d = pending_qkv[li]
                    keys_w = {"q_proj.weight", "k_proj.weight", "v_proj.weight"}
                    keys_b = {"q_proj.bias", "k_proj.bias", "v_proj.bias"}
                    if keys_w <= d.keys():
                        parts = layer.self_attn.qkv_proj.shard_qkv_parts(
                            d["q_proj.weight"], d["k_proj.weight"], d["v_proj.weight"]
                        )
                        put_cat(f"layers.{li}.self_attn.qkv_proj.weight", parts)
                        for kk in keys_w:
                            del d[kk]
                    if layer.self_attn.qkv_proj.bias is not None and keys_b <= d.keys():
                        parts = layer.self_attn.qkv_proj.shard_qkv_parts(
                            d["q_proj.bias"], d["k_proj.bias"], d["v_proj.bias"]
                        )
                        put_cat(f"layers.{li}.self_attn.qkv_proj.bias", parts)
                        for kk in keys_b:
                            del d[kk]
                elif sub == "self_attn.o_proj.weight":
                    put(
                        f"layers.{li}.self_attn.o_proj.weight",
                        layer.self_attn.o_proj.shard(w),
                    )
                elif sub in ("mlp.gate_proj.weight", "mlp.up_proj.weight"):
                    pending_mlp.setdefault(li, {})[sub] = w
                    d = pending_mlp[li]
                    if {"mlp.gate_proj.weight", "mlp.up_proj.weight"} <= d.keys():
                        parts = layer.mlp.gate_up_proj.shard_merged_parts(
                            d["mlp.gate_proj.weight"], d["mlp.up_proj.weight"]
                        )
                        put_cat(f"layers.{li}.mlp.gate_up_proj.weight", parts)
                        d.clear()
                elif sub == "mlp.down_proj.weight":
                    put(f"layers.{li}.mlp.down_proj.weight", layer.mlp.down_proj.shard(w))
                elif sub == "mlp.shared_expert_gate.weight":
                    layer.mlp.shared_gate.data.copy_(
                        w.to(layer.mlp.shared_gate.dtype))
                elif sub in ("mlp.shared_expert.gate_proj.weight",
                             "mlp.shared_expert.up_proj.weight"):
                    pending_mlp.setdefault((li, "se"), {})[sub] = w
                    d = pending_mlp[(li, "se")]
                    if len(d) == 2:
                        fused = layer.mlp.shared.gate_up_proj.shard_merged(
                            d["mlp.shared_expert.gate_proj.weight"],
                            d["mlp.shared_expert.up_proj.weight"],
                        )
                        layer.mlp.shared.gate_up_proj.weight.data.copy_(
                            fused.to(layer.mlp.shared.gate_up_proj.weight.dtype))
                        d.clear()
                elif sub == "mlp.shared_expert.down_proj.weight":
                    layer.mlp.shared.down_proj.weight.data.copy_(
                        layer.mlp.shared.down_proj.shard(w).to(
                            layer.mlp.shared.down_proj.weight.dtype))
                elif sub in ("block_sparse_moe.gate.weight", "mlp.gate.weight"):
                    layer.mlp.gate.data.copy_(w.to(layer.mlp.gate.dtype))
                elif sub.startswith("mlp.experts."):
                    # Qwen3-MoE naming: experts.N.{gate,up,down}_proj.weight
                    moe = layer.mlp
                    e = int(sub.split(".")[2])
                    which = sub.split(".")[3]
                    le = e - moe.expert_base
                    if 0 <= le < moe.local_experts:
                        # HF [out, in] -> our pre-transposed [in, out]
                        I = moe.inter
                        if which == "gate_proj":
                            moe.w13.data[le, :, :I].copy_(w.t().to(moe.w13.dtype))
                        elif which == "up_proj":
                            moe.w13.data[le, :, I:].copy_(w.t().to(moe.w13.dtype))
                        elif which == "down_proj":
                            moe.w2.data[le].copy_(w.t().to(moe.w2.dtype))
                elif sub.startswith("block_sparse_moe.experts."):
                    # Mixtral expert naming: w1=gate, w3=up, w2=down.
                    # EP-sharded: only this rank's experts are kept.
                    moe = layer.mlp
                    e = int(sub.split(".")[2])
                    which = sub.split(".")[3]
                    le = e - moe.expert_base
                    if 0 <= le < moe.local_experts:
                        # HF [out, in] -> our pre-transposed [in, out]
                        I = moe.inter
                        if which == "w1":
                            moe.w13.data[le, :, :I].copy_(w.t().to(moe.w13.dtype))
                        elif which == "w3":
                            moe.w13.data[le, :, I:].copy_(w.t().to(moe.w13.dtype))
                        elif which == "w2":
                            moe.w2.data[le].copy_(w.t().to(moe.w2.dtype))
                # rotary_emb.inv_freq etc. are ignored (recomputed)

    @torch.no_grad()
    def random_init(self, seed: int = 0) -> None:
        """Random-init with TP-consistent semantics: full HF-layout tensors
        are generated deterministically per name (so every TP rank sees
        slices of the SAME logical weights) and routed through the normal
        shard-aware loader. Generates on the model's device when possible."""
        import zlib

        cfg = self.cfg
        dev = self.embed_tokens.weight.device

        def gen(name: str, *shape, std: float = 0.02) -> torch.Tensor:
            s = (seed * 1000003 + zlib.crc32(name.encode())) % (2**63 - 1)
            if dev.type == "cuda":
                g = torch.Generator(device=dev).manual_seed(s)
                t = torch.randn(shape, generator=g, dtype=torch.float32, device=dev)
            else:
                g = torch.Generator().manual_seed(s)
                t = torch.randn(shape, generator=g, dtype=torch.float32)
            return t.mul_(std)

        H, I, hd = cfg.hidden_size, cfg.intermediate_size, cfg.head_dim
        nq, nkv = cfg.num_attention_heads, cfg.num_key_value_heads
        glob: dict[str, torch.Tensor] = {
            "model.embed_tokens.weight": gen("embed", cfg.vocab_size, H),
            "model.norm.weight": torch.ones(H),
        }
        if not cfg.tie_word_embeddings:
            glob["lm_head.weight"] = gen("lm_head", cfg.vocab_size, H)
        self.load_hf_state_dict(glob)
        del glob
        for i in range(cfg.num_hidden_layers):
            pre = f"model.layers.{i}"
            tensors = {
                f"{pre}.self_attn.q_proj.weight": gen(f"{pre}.q", nq * hd, H),
                f"{pre}.self_attn.k_proj.weight": gen(f"{pre}.k", nkv * hd, H),
                f"{pre}.self_attn.v_proj.weight": gen(f"{pre}.v", nkv * hd, H),
                f"{pre}.self_attn.o_proj.weight": gen(f"{pre}.o", H, nq * hd),
                f"{pre}.input_layernorm.weight": torch.ones(H),
                f"{pre}.post_attention_layernorm.weight": torch.ones(H),
            }
            if cfg.num_local_experts > 0:
                Ie = cfg.moe_intermediate_size or I
                if cfg.shared_expert_intermediate_size:
                    Is = cfg.shared_expert_intermediate_size
                    sp = f"{pre}.mlp.shared_expert"
                    tensors[f"{sp}.gate_proj.weight"] = gen(f"{sp}.g", Is, H)
                    tensors[f"{sp}.up_proj.weight"] = gen(f"{sp}.u", Is, H)
                    tensors[f"{sp}.down_proj.weight"] = gen(f"{sp}.d", H, Is)
                    tensors[f"{pre}.mlp.shared_expert_gate.weight"] = gen(
                        f"{sp}.sg", 1, H
                    )
                qstyle = not cfg.architecture.startswith("MixtralFor")
                gname = "mlp.gate" if qstyle else "block_sparse_moe.gate"
                tensors[f"{pre}.{gname}.weight"] = gen(
                    f"{pre}.moe.gate", cfg.num_local_experts, H
                )
                for e in range(cfg.num_local_experts):
                    if qstyle:
                        ep = f"{pre}.mlp.experts.{e}"
                        tensors[f"{ep}.gate_proj.weight"] = gen(f"{ep}.w1", Ie, H)
                        tensors[f"{ep}.up_proj.weight"] = gen(f"{ep}.w3", Ie, H)
                        tensors[f"{ep}.down_proj.weight"] = gen(f"{ep}.w2", H, Ie)
                    else:
                        ep = f"{pre}.block_sparse_moe.experts.{e}"
                        tensors[f"{ep}.w1.weight"] = gen(f"{ep}.w1", Ie, H)
                        tensors[f"{ep}.w3.weight"] = gen(f"{ep}.w3", Ie, H)
                        tensors[f"{ep}.w2.weight"] = gen(f"{ep}.w2", H, Ie)
            else:
                tensors[f"{pre}.mlp.gate_proj.weight"] = gen(f"{pre}.gate", I, H)
                tensors[f"{pre}.mlp.up_proj.weight"] = gen(f"{pre}.up", I, H)
                tensors[f"{pre}.mlp.down_proj.weight"] = gen(f"{pre}.down", H, I)
            if cfg.qk_norm:
                tensors[f"{pre}.self_attn.q_norm.weight"] = torch.ones(hd)
                tensors[f"{pre}.self_attn.k_norm.weight"] = torch.ones(hd)
            if cfg.attention_bias:
                tensors[f"{pre}.self_attn.q_proj.bias"] = torch.zeros(nq * hd)
                tensors[f"{pre}.self_attn.k_proj.bias"] = torch.zeros(nkv * hd)
                tensors[f"{pre}.self_attn.v_proj.bias"] = torch.zeros(nkv * hd)
            # load layer-by-layer to bound peak memory
            self.load_hf_state_dict(tensors)
