"""Op dispatch for arks_amd.

On GPU tensors every op runs the in-tree gfx950 HIP extension
(``arks_amd._C``) — there is no eager fallback: if the extension is missing
on a CUDA/ROCm device we raise immediately so a silently-slow PyTorch path
can never masquerade as the native one. On CPU tensors ops run the plain
PyTorch reference (arks_amd/ops/ref.py) so the engine/scheduler/model layers
are unit-testable without a GPU.
"""

from __future__ import annotations

import os

import torch

from . import ref

_C = None
_C_ERR: str | None = None


def _load_extension():
    global _C, _C_ERR
    if _C is not None or _C_ERR is not None:
        return _C
    try:
        from . import _load  # noqa: F401  (imports the built .so in-tree)

        _C = _load.C
    except Exception as e:  # pragma: no cover - exercised only when unbuilt
        _C_ERR = f"{type(e).__name__}: {e}"
    return _C


def native_available() -> bool:
    return _load_extension() is not None


def _native():
    c = _load_extension()
    if c is None:
        raise RuntimeError(
            "arks_amd HIP extension (arks_amd._C) is not built but a GPU op was "
            "requested. Build it with `python -m arks_amd.ops.build` "
            f"(import error: {_C_ERR})"
        )
    return c


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    if x.is_cuda:
        out = torch.empty_like(x)
        _native().rmsnorm(out, x, weight, eps)
        return out
    return ref.rmsnorm(x, weight, eps)


def fused_add_rmsnorm(x, residual, weight, eps: float = 1e-6):
    """In-place on GPU: residual += x (written to residual), out = norm(residual)."""
    if x.is_cuda:
        out = torch.empty_like(x)
        _native().fused_add_rmsnorm(out, x, residual, weight, eps)
        return out, residual
    return ref.fused_add_rmsnorm(x, residual, weight, eps)


def silu_mul(gate_up: torch.Tensor) -> torch.Tensor:
    if gate_up.is_cuda:
        shape = list(gate_up.shape)
        shape[-1] //= 2
        out = gate_up.new_empty(shape)
        _native().silu_mul(out, gate_up)
        return out
    return ref.silu_mul(gate_up)


def rope_apply_inplace(positions, q, k, cos_sin, head_dim: int):
    """Rotates q and k in place (GPU). CPU path returns new tensors."""
    if q.is_cuda:
        _native().rope_inplace(positions, q, k, cos_sin, head_dim)
        return q, k
    return ref.rope_apply(positions, q, k, cos_sin, head_dim)


def rope_and_cache(positions, q, k, v, k_cache, v_cache, slot_mapping,
                   cos_sin, head_dim: int):
    """Fused RoPE + paged-cache write (GPU): rotates q/k in place and
    scatters rotated k + v into the cache in one launch. CPU path runs the
    two reference steps."""
    if q.is_cuda:
        _native().rope_and_cache(positions, q, k, v, k_cache, v_cache,
                                 slot_mapping, cos_sin, head_dim)
        return q, k
    q2, k2 = ref.rope_apply(positions, q, k, cos_sin, head_dim)
    T = q.shape[0]
    nkv = k.numel() // max(T, 1) // head_dim
    ref.reshape_and_cache(
        k2.reshape(T, nkv, head_dim), v.reshape(T, nkv, head_dim),
        k_cache, v_cache, slot_mapping,
    )
    return q2, k2


def reshape_and_cache(k, v, k_cache, v_cache, slot_mapping):
    if k.is_cuda:
        _native().reshape_and_cache(k, v, k_cache, v_cache, slot_mapping)
        return
    ref.reshape_and_cache(k, v, k_cache, v_cache, slot_mapping)


PREFILL_QTILE = 64  # q rows per workgroup in the prefill kernel


def build_prefill_tiles(seq_lens: list[int], device) -> torch.Tensor:
    """int32 [ntiles, 2] of (seq_idx, q0) covering every sequence in 64-row
    tiles — the prefill kernel's grid."""
    tiles = []
    for i, n in enumerate(seq_lens):
        for q0 in range(0, n, PREFILL_QTILE):
            tiles.append((i, q0))
    return torch.tensor(tiles, dtype=torch.int32, device=device).reshape(-1, 2)


def attention_prefill_varlen(q, k, v, cu_seqlens, seq_lens: list[int],
                             scale: float, window: int = 0):
    """Causal varlen prefill attention. seq_lens is the host-side list of
    per-sequence lengths (used to build the q-tile grid without a D2H sync).
    window > 0 = sliding-window attention."""
    if q.is_cuda:
        out = torch.empty_like(q)
        tile_info = build_prefill_tiles(seq_lens, q.device)
        _native().attention_prefill_varlen(out, q, k, v, cu_seqlens, tile_info,
                                           scale, window)
        return out
    return ref.attention_prefill_varlen(q, k, v, cu_seqlens, scale, window)


EXTEND2_QTILE = 256  # q rows per workgroup in the 8-wave ladder kernel
E2_KVBLK = 64        # keys per LDS tile in the ladder kernel
# Sequences with at least this many new q tokens go to the 256-row ladder
# kernel; shorter ones (decode-adjacent chunks, prefix-cache hits) stay on
# the 64-row kernel where a mostly-empty 256-row tile would waste waves.
EXTEND2_MIN_QLEN = 97


def build_extend_tiles(q_lens: list[int], kv_lens: list[int] | None,
                       kv_fp8: bool, device, num_q_heads: int = 32):
    """Per-seq routing for the extend dispatcher:
    (tiles64, tiles256 [n,4], decode_rows, combine_table, ws_rows).

    q_len == 1 sequences (decode steps mixed into a prefill batch) go to
    the flash-decode kernel — an extend q-tile for one query scans the
    full KV with 255 masked rows (~6x slower than the decode kernel's
    partitioned scan). Remaining sequences split between the 64-row and
    256-row extend kernels; tiles are sorted by DESCENDING key depth
    (causal q-tiles differ up to 4x in KV work and the dispatcher issues
    blocks in order — deep tiles first removes the straggler tail). When
    the batch is too small to fill the chip, deep 256-row tiles are
    additionally SPLIT over their KV range flash-decode-style (part rows
    + a combine pass over f32 partials); split parts sort first so the
    partial workspace only covers them. Build once per batch and reuse
    across all layers."""
    t64, t256, dec = [], [], []
    kvl = kv_lens if kv_lens is not None else q_lens
    for i, n in enumerate(q_lens):
        off = kvl[i] - n
        if n == 1:
            dec.append(i)
        elif kv_fp8 or n < EXTEND2_MIN_QLEN:
            for q0 in range(0, n, PREFILL_QTILE):
                t64.append((i, q0, min(kvl[i], off + q0 + PREFILL_QTILE)))
        else:
            for q0 in range(0, n, EXTEND2_QTILE):
                t256.append((i, q0, min(kvl[i], off + q0 + EXTEND2_QTILE)))
    t64.sort(key=lambda t: -t[2])
    t256.sort(key=lambda t: -t[2])
    mk64 = lambda t: torch.tensor(
        [x[:2] for x in t], dtype=torch.int32, device=device
    ).reshape(-1, 2)

    # KV-split policy for the 256-row kernel: when tiles x heads cannot
    # fill ~2 workgroups per CU, split the deepest tiles' key ranges.
    base_wgs = len(t256) * max(num_q_heads, 1)
    factor = 1
    if t256 and base_wgs < 256:
        # truly starved launches only (256 CUs, 1 block/CU kernel): the
        # partial slabs + combine cost ~2x slab traffic, so marginal
        # shortfalls are better left unsplit
        factor = min(8, -(-512 // base_wgs))
    rows, combine = [], []
    ws_rows = 0
    if factor > 1:
        split, plain = [], []
        for (i, q0, kmax) in t256:
            span = (kmax + E2_KVBLK - 1) // E2_KVBLK
            p = min(factor, max(1, span // 4))
            (split if p > 1 else plain).append((i, q0, p))
        # cap the f32 partial workspace at ~512 MB
        slab = num_q_heads * EXTEND2_QTILE * 130 * 4
        if sum(p for _, _, p in split) * slab > 512 * 1024 * 1024:
            split, plain = [], split + plain
        for (i, q0, p) in split:
            combine.append((len(rows), p, i, q0))
            for part in range(p):
                rows.append((i, q0, part, p))
        ws_rows = len(rows)
        for (i, q0, p) in plain:
            rows.append((i, q0, 0, 1))
    else:
        rows = [(i, q0, 0, 1) for (i, q0, _) in t256]
    t256_t = torch.tensor(rows, dtype=torch.int32,
                          device=device).reshape(-1, 4)
    ctab = torch.tensor(combine, dtype=torch.int32,
                        device=device).reshape(-1, 4)

    if dec:
        # gather indices: the single q row of each decode seq sits at
        # cu_seqlens_q[i]; seq-level metadata indexes with `dec` itself
        cu = [0]
        for n in q_lens:
            cu.append(cu[-1] + n)
        drows = torch.tensor([cu[i] for i in dec], dtype=torch.long,
                             device=device)
        dseqs = torch.tensor(dec, dtype=torch.long, device=device)
    else:
        drows = dseqs = None
    return mk64(t64), t256_t, (drows, dseqs), ctab, ws_rows


def attention_extend_paged(
    q, k_cache, v_cache, block_tables, kv_lens, cu_seqlens_q,
    q_lens: list[int], scale: float, window: int = 0, tiles=None,
):
    """Causal attention of packed NEW tokens over each sequence's full paged
    KV history (cached prefix + new tokens already written by
    reshape_and_cache). The prefix-caching / chunked-prefill attention path.
    q_lens is the host-side per-seq new-token count (tile grid without a
    D2H sync); kv_lens is the device int32 total kv length per seq.
    window > 0 = sliding-window attention.

    Dispatch: big chunks run the 8-wave 32x32-MFMA ladder kernel
    (attn_extend2.hip, 256-row q tiles); small chunks and fp8-KV runs use
    the 4-wave 64-row kernel."""
    if q.is_cuda:
        out = torch.empty(
            (q.shape[0], q.shape[1], q.shape[2]), dtype=q.dtype, device=q.device
        )
        kv_fp8 = k_cache.dtype != torch.bfloat16
        t64, t256, (drows, dseqs), ctab, ws_rows = (
            tiles if tiles is not None
            else build_extend_tiles(q_lens, None, kv_fp8, q.device,
                                    num_q_heads=q.shape[1])
        )
        if t256.numel():
            if ws_rows:
                ws = _extend_ws(
                    ws_rows * q.shape[1] * EXTEND2_QTILE * (q.shape[2] + 2),
                    q.device,
                )
            else:
                ws = q.new_empty(0, dtype=torch.float32)
            _native().attention_extend_paged2(
                out, q, k_cache, v_cache, block_tables, kv_lens, cu_seqlens_q,
                t256, ws, scale, window,
            )
            if ctab.numel():
                _native().attention_extend2_combine(
                    out, ws, ctab, cu_seqlens_q, scale,
                )
        if t64.numel():
            _native().attention_extend_paged(
                out, q, k_cache, v_cache, block_tables, kv_lens, cu_seqlens_q,
                t64, scale, window,
            )
        if drows is not None:
            dq = q[drows].contiguous()
            dout = attention_decode_paged(
                dq, k_cache, v_cache,
                block_tables[dseqs].contiguous(),
                kv_lens[dseqs].contiguous(), scale, window=window,
            )
            out[drows] = dout
        return out
    return ref.attention_extend_paged(
        q, k_cache, v_cache, block_tables, kv_lens, cu_seqlens_q, scale,
        window,
    )


def decode_num_partitions(num_seqs: int, num_kv_heads: int, max_blocks: int) -> int:
    """Flash-decode split factor: fill the 256 CUs (target ~2 workgroups/CU)
    when batch x kv_heads alone cannot. Each partition keeps >= 8 pages so
    all 4 waves of a workgroup get a 2-page chunk and the K-prefetch
    software pipeline has depth (bench_decode2: starved waves at <8
    pages/partition cap short-seq shapes)."""
    base = num_seqs * num_kv_heads
    target = 512
    nparts = max(1, -(-target // base))
    nparts = min(nparts, max(1, max_blocks // 8), 64)
    return nparts


def attention_decode_paged(
    q, k_cache, v_cache, block_tables, seq_lens, scale: float,
    num_partitions: int | None = None, part_out=None, window: int = 0,
):
    if q.is_cuda:
        out = torch.empty(
            (q.shape[0], q.shape[1], q.shape[2]), dtype=q.dtype, device=q.device
        )
        num_kv_heads = k_cache.shape[1]
        if num_partitions is None:
            num_partitions = decode_num_partitions(
                q.shape[0], num_kv_heads, block_tables.shape[1]
            )
        if num_partitions > 1 and part_out is None:
            gq = q.shape[1] // num_kv_heads
            part_out = torch.empty(
                q.shape[0] * num_kv_heads * num_partitions * gq * (q.shape[2] + 2),
                dtype=torch.float32, device=q.device,
            )
        elif part_out is None:
            part_out = q.new_empty(0, dtype=torch.float32)
        _native().attention_decode_paged(
            out, q, k_cache, v_cache, block_tables, seq_lens, scale,
            part_out, num_partitions, window,
        )
        return out
    return ref.attention_decode_paged(
        q, k_cache, v_cache, block_tables, seq_lens, scale, window
    )


_EXTEND_WS: dict = {}


def _extend_ws(need: int, device):
    key = (device.index,)
    ws = _EXTEND_WS.get(key)
    if ws is None or ws.numel() < need:
        ws = torch.empty(need, dtype=torch.float32, device=device)
        _EXTEND_WS[key] = ws
    return ws


_SKINNY_WS: dict = {}


def skinny_gemm(x, weight, bias=None):
    """C = x @ weight^T for decode-sized M (<=64) via the split-K streaming
    MFMA kernel. Beats hipBLASLt's ~19 us latency floor on the small
    qkv/o_proj shapes and its ~46 us in-graph down-proj algo (bench_skinny /
    bench_down logs, r2); the dispatcher keeps the wide streaming shapes
    (gate_up, lm_head: N >> K) on the tuned library algos. Deterministic
    split-K (fixed-order combine, no atomics)."""
    M, K = x.shape
    N = weight.shape[0]
    out = torch.empty((M, N), dtype=x.dtype, device=x.device)
    ntiles = N // 64
    if K > N and ntiles < 128 and M > 48:
        # tall-K (down-proj shape): few N tiles, parallelism comes from
        # K-splits; the deeper-staged AB=3 variant at nsplits ~= 448/ntiles
        # measured fastest (34.6 us vs hipBLASLt's 46.4 in-graph, r2).
        # M-gated: the variant kernel is fixed at MT=4 (full 64-row cost)
        # while hipBLASLt switches to a faster algo below M~48 — routing
        # small-M down-proj here regressed b16x8k decode by 4% (r2)
        nsplits = min(-(-448 // ntiles), -(-K // 256))
        k_per_split = -(-(-(-K // nsplits)) // 32) * 32
        nsplits = -(-K // k_per_split)
        part = (_skinny_ws(nsplits, N, 64, x.device) if nsplits > 1
                else x.new_empty(0, dtype=torch.float32))
        _native().skinny_gemm_v(out, part, x, weight, bias, k_per_split,
                                nsplits, _SKINNY_TALLK_VARIANT, False)
        return out
    if ntiles >= 384:
        nsplits = 1
    else:
        nsplits = min(-(-512 // ntiles), -(-K // 256))
    k_ceil = -(-K // nsplits)
    k_per_split = -(-k_ceil // 32) * 32
    nsplits = -(-K // k_per_split)
    part = _skinny_ws(nsplits, N, M, x.device) if nsplits > 1 else x.new_empty(
        0, dtype=torch.float32
    )
    _native().skinny_gemm(out, part, x, weight, bias, k_per_split, nsplits)
    return out


def _skinny_ws(nsplits: int, N: int, M: int, device):
    mrows = ((M + 15) // 16) * 16
    key = (device.index,)
    need = nsplits * N * mrows
    ws = _SKINNY_WS.get(key)
    if ws is None or ws.numel() < need:
        ws = torch.empty(need, dtype=torch.float32, device=device)
        _SKINNY_WS[key] = ws
    return ws


_USE_SKINNY = os.environ.get("ARKS_SKINNY_GEMM", "1") == "1"
# Weight-elements threshold: the split-K streaming kernel beats hipBLASLt's
# ~19 us latency floor on the small decode shapes (qkv/o: 1.0-1.6x,
# bench_skinny v3); the big streaming shapes (gate_up/down/lm_head) stay on
# the tuned hipBLASLt algos which reach 5.4-6.4 TB/s there.
_SKINNY_MAX_ELEMS = 34_000_000
# tall-K decode shapes (down-proj: K > N) stream W near-HBM-rate through the
# split-K kernel and beat the library's in-graph algo (r2); wide-N shapes
# (gate_up/lm_head) stay on hipBLASLt which reaches ~6 TB/s there.
_SKINNY_TALLK_MAX_ELEMS = 80_000_000
# 7 = AB=3 deep-staged; 8 = same + non-temporal W loads
_SKINNY_TALLK_VARIANT = int(os.environ.get("ARKS_SKINNY_TALLK_V", "7"))


def linear_bf16(x, weight, bias=None):
    """Linear dispatch: the skinny split-K kernel for small decode-shaped
    GEMMs on GPU, hipBLASLt (F.linear) otherwise."""
    if (
        _USE_SKINNY
        and x.is_cuda
        and x.dim() == 2
        and 1 <= x.shape[0] <= 64
        and x.dtype == torch.bfloat16
        and weight.shape[0] % 64 == 0
        and weight.shape[1] % 32 == 0
        and (weight.numel() <= _SKINNY_MAX_ELEMS
             or (weight.shape[1] > weight.shape[0]
                 and x.shape[0] > 48
                 and weight.numel() <= _SKINNY_TALLK_MAX_ELEMS))
        and weight.is_contiguous()
        and native_available()
    ):
        return skinny_gemm(x, weight, bias)
    import torch.nn.functional as F

    return F.linear(x, weight, bias)


def rmsnorm_fp8(x, weight, eps: float = 1e-6):
    """Fused RMSNorm -> per-token fp8 e4m3: (out8, inv_scale). GPU-only
    fast path (hidden <= 8192); otherwise compose the unfused ops."""
    if x.is_cuda and x.shape[-1] <= 8192:
        out = torch.empty(x.shape, dtype=torch.float8_e4m3fn, device=x.device)
        inv_scale = torch.empty(x.shape[0], dtype=torch.float32, device=x.device)
        _native().rmsnorm_fp8(out, inv_scale, x, None, weight, eps)
        return out, inv_scale
    return quant_fp8_rows(rmsnorm(x, weight, eps))


def fused_add_rmsnorm_fp8(x, residual, weight, eps: float = 1e-6):
    """Fused residual-add + RMSNorm -> fp8: (out8, inv_scale, residual)."""
    if x.is_cuda and x.shape[-1] <= 8192:
        out = torch.empty(x.shape, dtype=torch.float8_e4m3fn, device=x.device)
        inv_scale = torch.empty(x.shape[0], dtype=torch.float32, device=x.device)
        _native().rmsnorm_fp8(out, inv_scale, x, residual, weight, eps)
        return out, inv_scale, residual
    y, res = fused_add_rmsnorm(x, residual, weight, eps)
    q, s = quant_fp8_rows(y)
    return q, s, res


def silu_mul_fp8(gate_up):
    """Fused SwiGLU -> per-token fp8: (out8, inv_scale)."""
    d = gate_up.shape[-1] // 2
    # dynamic-LDS default cap is 64 KiB; stay under it (d*2 bytes staged)
    if gate_up.is_cuda and d % 8 == 0 and d * 2 <= 60 * 1024:
        out = torch.empty((gate_up.shape[0], d), dtype=torch.float8_e4m3fn,
                          device=gate_up.device)
        inv_scale = torch.empty(gate_up.shape[0], dtype=torch.float32,
                                device=gate_up.device)
        _native().silu_mul_fp8(out, inv_scale, gate_up.contiguous())
        return out, inv_scale
    return quant_fp8_rows(silu_mul(gate_up))


def quant_fp8_rows(x):
    """Dynamic per-token fp8 e4m3 quantization: [M, K] bf16 ->
    ([M, K] float8_e4m3fn, [M] fp32 dequant scales) for W8A8 GEMMs
    (torch._scaled_mm rowwise)."""
    if x.is_cuda:
        out = torch.empty(x.shape, dtype=torch.float8_e4m3fn, device=x.device)
        inv_scale = torch.empty(x.shape[0], dtype=torch.float32, device=x.device)
        _native().quant_fp8_rows(out, inv_scale, x.contiguous())
        return out, inv_scale
    return ref.quant_fp8_rows(x)


def moe_topk(router_logits, k: int, renorm: bool):
    """softmax + top-k + optional renorm in one kernel:
    (weights [T,k] f32, ids [T,k] i32). torch.topk tie-breaking (lower
    expert id wins)."""
    if router_logits.is_cuda:
        T = router_logits.shape[0]
        weights = torch.empty(T, k, dtype=torch.float32,
                              device=router_logits.device)
        ids = torch.empty(T, k, dtype=torch.int32,
                          device=router_logits.device)
        _native().moe_topk(weights, ids, router_logits.contiguous(), k,
                           1 if renorm else 0)
        return weights, ids
    return ref.moe_topk(router_logits, k, renorm)


def moe_mix(y, weights, ids, expert_base: int = 0):
    """out[t] = sum_i weights[t,i] * y[ids[t,i]-expert_base, t] with
    out-of-slice experts contributing 0 (TP expert parallelism)."""
    if y.is_cuda:
        El, T, H = y.shape
        out = torch.empty(T, H, dtype=y.dtype, device=y.device)
        _native().moe_mix(out, y.contiguous(), weights, ids, expert_base, El)
        return out
    return ref.moe_mix(y, weights, ids, expert_base)


def moe_mix_rows(y, weights, rows):
    """out[t] = sum_j weights[t,j] * y[rows[t,j]] — the gather-side mix
    over a flat per-pair expert-output buffer (zero-weight slots point at
    row 0)."""
    if y.is_cuda:
        T, k = weights.shape
        out = torch.empty(T, y.shape[1], dtype=y.dtype, device=y.device)
        _native().moe_mix_rows(out, y.contiguous(), weights, rows)
        return out
    return ref.moe_mix_rows(y, weights, rows)


def sample_tokens(logits, temperatures, uniform):
    """Gumbel-max categorical sampling; rows with temperature 0 are greedy."""
    if logits.is_cuda:
        out = torch.empty(logits.shape[0], dtype=torch.int64, device=logits.device)
        _native().gumbel_sample(out, logits, temperatures, uniform)
        return out
    return ref.gumbel_sample(logits, temperatures, uniform)


def greedy_sample(logits):
    if logits.is_cuda:
        out = torch.empty(logits.shape[0], dtype=torch.int64, device=logits.device)
        _native().greedy_sample(out, logits)
        return out
    return ref.greedy_sample(logits)
