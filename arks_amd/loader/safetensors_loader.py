"""Safetensors weight loading with pinned-host staging.

Cold-start time is a headline metric: weights stream PVC ->
page cache -> pinned staging buffer -> HBM with hipMemcpyAsync on a side
stream, double-buffered so disk reads overlap H2D copies. TP sharding happens
on the host slice before upload, so only this rank's bytes cross PCIe.
"""

from __future__ import annotations

import glob
import json
import os

import torch


def _iter_safetensors(model_path: str):
    files = sorted(glob.glob(os.path.join(model_path, "*.safetensors")))
    if not files:
        raise FileNotFoundError(f"no *.safetensors under {model_path}")
    from safetensors import safe_open

    for f in files:
        with safe_open(f, framework="pt", device="cpu") as sf:
            for name in sf.keys():
                yield name, sf.get_tensor(name)


def _iter_chunks(model_path: str, limit_bytes: int, pin: bool):
    """Group tensors into >= limit_bytes chunks with fused-weight partners
    co-resident (load_hf_state_dict needs q/k/v and gate/up together).
    With pin=True each tensor is copied into pinned memory here — done on
    the reader thread so disk reads + pinning overlap the H2D uploads."""
    chunk: dict[str, torch.Tensor] = {}
    chunk_bytes = 0
    for name, t in _iter_safetensors(model_path):
        chunk[name] = t.pin_memory() if pin else t
        chunk_bytes += t.numel() * t.element_size()
        if chunk_bytes >= limit_bytes and _chunk_complete(chunk):
            yield chunk
            chunk, chunk_bytes = {}, 0
    if chunk:
        yield chunk


_ST_DTYPES = {
    "BF16": torch.bfloat16, "F16": torch.float16, "F32": torch.float32,
    "F64": torch.float64, "I64": torch.int64, "I32": torch.int32,
    "I16": torch.int16, "I8": torch.int8, "U8": torch.uint8,
    "BOOL": torch.bool,
}


def _parse_header(path: str):
    """safetensors layout: u64 LE header length, JSON header with per-tensor
    {dtype, shape, data_offsets}, then the data section. Returns
    (data_start, [(name, dtype, shape, off0, off1) sorted by offset])."""
    import struct

    with open(path, "rb") as f:
        (hlen,) = struct.unpack("<Q", f.read(8))
        header = json.loads(f.read(hlen))
    data_start = 8 + hlen
    ts = [
        (name, _ST_DTYPES[m["dtype"]], m["shape"],
         m["data_offsets"][0], m["data_offsets"][1])
        for name, m in header.items() if name != "__metadata__"
    ]
    ts.sort(key=lambda x: x[3])
    return data_start, ts


def load_model_weights(model, model_path: str, device: torch.device,
                       chunk_bytes: int = 1 << 30) -> None:
    """Stream HF-layout safetensors into the (possibly TP-sharded) model.

    The model's load_hf_state_dict handles name mapping + sharding. The GPU
    path parses the safetensors headers directly and `readinto`s each chunk
    of the data section — tensors grouped in file-offset order — into one
    of TWO recycled pinned staging buffers; tensor views over the pinned
    bytes upload with async H2D on a side stream while the reader fills the
    other buffer. Pinned memory is allocated exactly twice for the whole
    load: the previous per-tensor `pin_memory()` scheme spent most of the
    cold start page-locking fresh allocations (0.59 GB/s loader vs
    3.5 GB/s raw disk on the same box — cold start is a headline metric,
    BASELINE.md).
    """
    model.to(device)
    if device.type != "cuda":
        for chunk in _iter_chunks(model_path, chunk_bytes, pin=False):
            model.load_hf_state_dict(chunk)
        return

    import queue
    import threading

    files = sorted(glob.glob(os.path.join(model_path, "*.safetensors")))
    if not files:
        raise FileNotFoundError(f"no *.safetensors under {model_path}")

    bufs = [torch.empty(chunk_bytes + (64 << 20), dtype=torch.uint8,
                        pin_memory=True) for _ in range(2)]
    free_q: "queue.Queue" = queue.Queue()
    for i in range(2):
        free_q.put((i, None))
    q: "queue.Queue" = queue.Queue(maxsize=2)

    def reader():
        try:
            # fused-partner tensors straddling a file/buffer boundary are
            # CLONED out of the recycled pinned buffer and re-joined with
            # the next chunk (their group applies only when complete)
            carry: dict[str, torch.Tensor] = {}
            for path in files:
                data_start, ts = _parse_header(path)
                with open(path, "rb", buffering=0) as f:
                    i = 0
                    while i < len(ts):
                        buf_id, ev = free_q.get()
                        if ev is not None:
                            ev.synchronize()  # prior H2D from this buffer done
                        buf = bufs[buf_id]
                        cap = buf.numel()
                        # take tensors until chunk_bytes AND fused partners
                        # complete (q/k/v, gate/up must land together)
                        j = i
                        names: dict[str, torch.Tensor] = {}
                        span0 = ts[i][3]
                        while j < len(ts):
                            name, dt, shape, o0, o1 = ts[j]
                            if o1 - span0 > cap:
                                break
                            names[name] = (dt, shape, o0, o1)
                            j += 1
                            if (o1 - span0 >= chunk_bytes
                                    and _chunk_complete(names)):
                                break
                        if j == i:  # single tensor larger than the buffer
                            raise RuntimeError(
                                f"tensor {ts[i][0]} exceeds staging buffer")
                        span1 = ts[j - 1][4]
                        mv = memoryview(buf.numpy())[: span1 - span0]
                        f.seek(data_start + span0)
                        got = f.readinto(mv)
                        assert got == span1 - span0, (got, span1 - span0)
                        chunk: dict[str, torch.Tensor] = dict(carry)
                        carry = {}
                        for name, (dt, shape, o0, o1) in names.items():
                            sl = buf[o0 - span0: o1 - span0]
                            chunk[name] = sl.view(dt).view(shape)
                        for name in _dangling_names(chunk):
                            carry[name] = chunk.pop(name).clone()
                        q.put((chunk, buf_id))
                        i = j
            assert not carry, f"unpaired fused weights: {sorted(carry)}"
            q.put(None)
        except BaseException as exc:  # surface disk errors on the consumer
            q.put(exc)

    threading.Thread(target=reader, daemon=True).start()
    side = torch.cuda.Stream(device)
    while True:
        item = q.get()
        if item is None:
            break
        if isinstance(item, BaseException):
            raise item
        chunk, buf_id = item
        with torch.cuda.stream(side):
            model.load_hf_state_dict(chunk)
            ev = torch.cuda.Event()
            ev.record(side)
        free_q.put((buf_id, ev))
    torch.cuda.current_stream(device).wait_stream(side)
    torch.cuda.synchronize(device)


def _dangling_names(chunk) -> set:
    """Names whose fused-weight partner group is incomplete in `chunk`
    (q/k/v and gate/up must be applied together — a group split across a
    file or staging-buffer boundary must carry over, or its weights would
    silently never load)."""
    names = set(chunk)
    out = set()
    for n in names:
        if "q_proj" in n or "k_proj" in n or "v_proj" in n:
            stem = n.rsplit(".", 2)[0]
            kind = n.rsplit(".", 1)[1]
            if any(f"{stem}.{p}.{kind}" not in names
                   for p in ("q_proj", "k_proj", "v_proj")):
                out.add(n)
        if "gate_proj" in n or "up_proj" in n:
            stem = n.rsplit(".", 2)[0]
            if any(f"{stem}.{p}.weight" not in names
                   for p in ("gate_proj", "up_proj")):
                out.add(n)
    return out


def _chunk_complete(chunk: dict[str, torch.Tensor]) -> bool:
    """True when no fused-weight partner is missing (q/k/v and gate/up pairs
    must land in the same chunk)."""
    names = set(chunk)
    for n in names:
        if "q_proj" in n or "k_proj" in n or "v_proj" in n:
            stem = n.rsplit(".", 2)[0]
            kind = n.rsplit(".", 1)[1]
            for p in ("q_proj", "k_proj", "v_proj"):
                if f"{stem}.{p}.{kind}" not in names:
                    return False
        if "gate_proj" in n or "up_proj" in n:
            stem = n.rsplit(".", 2)[0]
            for p in ("gate_proj", "up_proj"):
                if f"{stem}.{p}.weight" not in names:
                    return False
    return True


def save_random_checkpoint(cfg, out_dir: str, seed: int = 0) -> None:
    """Write a random-init HF-layout checkpoint (config.json + safetensors)
    for tests of the loading path (no network for real checkpoints)."""
    from safetensors.torch import save_file

    from ..models import create_model
    from ..parallel.comm import get_tp_world_size

    assert get_tp_world_size() == 1, "save from a TP=1 process"
    model = create_model(cfg)
    model.random_init(seed)
    os.makedirs(out_dir, exist_ok=True)
    # Emit HF-layout names (split fused weights back apart).
    out: dict[str, torch.Tensor] = {
        "model.embed_tokens.weight": model.embed_tokens.weight.data,
        "model.norm.weight": model.norm.weight.data,
    }
    if not cfg.tie_word_embeddings:
        out["lm_head.weight"] = model.lm_head.weight.data[: cfg.vocab_size]
    hd = cfg.head_dim
    nq, nkv = cfg.num_attention_heads, cfg.num_key_value_heads
    for i, layer in enumerate(model.layers):
        pre = f"model.layers.{i}"
        qkv = layer.self_attn.qkv_proj.weight.data
        q, k, v = qkv.split([nq * hd, nkv * hd, nkv * hd], dim=0)
        out[f"{pre}.self_attn.q_proj.weight"] = q
        out[f"{pre}.self_attn.k_proj.weight"] = k
        out[f"{pre}.self_attn.v_proj.weight"] = v
        if layer.self_attn.qkv_proj.bias is not None:
            qb, kb, vb = layer.self_attn.qkv_proj.bias.data.split(
                [nq * hd, nkv * hd, nkv * hd], dim=0
            )
            out[f"{pre}.self_attn.q_proj.bias"] = qb
            out[f"{pre}.self_attn.k_proj.bias"] = kb
            out[f"{pre}.self_attn.v_proj.bias"] = vb
        out[f"{pre}.self_attn.o_proj.weight"] = layer.self_attn.o_proj.weight.data
        if layer.self_attn.q_norm is not None:
            out[f"{pre}.self_attn.q_norm.weight"] = layer.self_attn.q_norm.weight.data
            out[f"{pre}.self_attn.k_norm.weight"] = layer.self_attn.k_norm.weight.data
        if hasattr(layer.mlp, "w13"):  # sparse MoE block
            I = layer.mlp.inter
            qstyle = not cfg.architecture.startswith("MixtralFor")
            if layer.mlp.shared is not None:
                sp = f"{pre}.mlp.shared_expert"
                sgu = layer.mlp.shared.gate_up_proj.weight.data
                sg, su = sgu.split(sgu.shape[0] // 2, dim=0)
                out[f"{sp}.gate_proj.weight"] = sg
                out[f"{sp}.up_proj.weight"] = su
                out[f"{sp}.down_proj.weight"] = layer.mlp.shared.down_proj.weight.data
                out[f"{pre}.mlp.shared_expert_gate.weight"] = layer.mlp.shared_gate.data
            gname = "mlp.gate" if qstyle else "block_sparse_moe.gate"
            out[f"{pre}.{gname}.weight"] = layer.mlp.gate.data
            # w13/w2 are stored pre-transposed [in, out]; export back to the
            # HF [out, in] convention
            for e in range(layer.mlp.local_experts):
                if qstyle:
                    ep = f"{pre}.mlp.experts.{e}"
                    out[f"{ep}.gate_proj.weight"] = (
                        layer.mlp.w13.data[e, :, :I].t().contiguous())
                    out[f"{ep}.up_proj.weight"] = (
                        layer.mlp.w13.data[e, :, I:].t().contiguous())
                    out[f"{ep}.down_proj.weight"] = (
                        layer.mlp.w2.data[e].t().contiguous())
                else:
                    ep = f"{pre}.block_sparse_moe.experts.{e}"
                    out[f"{ep}.w1.weight"] = (
                        layer.mlp.w13.data[e, :, :I].t().contiguous())
                    out[f"{ep}.w3.weight"] = (
                        layer.mlp.w13.data[e, :, I:].t().contiguous())
                    out[f"{ep}.w2.weight"] = (
                        layer.mlp.w2.data[e].t().contiguous())
        else:
            gu = layer.mlp.gate_up_proj.weight.data
            g, u = gu.split(gu.shape[0] // 2, dim=0)
            out[f"{pre}.mlp.gate_proj.weight"] = g
            out[f"{pre}.mlp.up_proj.weight"] = u
            out[f"{pre}.mlp.down_proj.weight"] = layer.mlp.down_proj.weight.data
        out[f"{pre}.input_layernorm.weight"] = layer.input_layernorm.weight.data
        out[f"{pre}.post_attention_layernorm.weight"] = (
            layer.post_attention_layernorm.weight.data
        )
    save_file({k: v.contiguous() for k, v in out.items()},
              os.path.join(out_dir, "model.safetensors"))
    hf_cfg = {
        "architectures": [cfg.architecture],
        "vocab_size": cfg.vocab_size,
        "hidden_size": cfg.hidden_size,
        "intermediate_size": cfg.intermediate_size,
        "num_hidden_layers": cfg.num_hidden_layers,
        "num_attention_heads": cfg.num_attention_heads,
        "num_key_value_heads": cfg.num_key_value_heads,
        "head_dim": cfg.head_dim,
        "rms_norm_eps": cfg.rms_norm_eps,
        "rope_theta": cfg.rope_theta,
        "max_position_embeddings": cfg.max_position_embeddings,
        "tie_word_embeddings": cfg.tie_word_embeddings,
        "attention_bias": cfg.attention_bias,
        "eos_token_id": cfg.eos_token_id,
        "bos_token_id": cfg.bos_token_id,
        "torch_dtype": "bfloat16",
    }
    with open(os.path.join(out_dir, "config.json"), "w") as f:
        json.dump(hf_cfg, f, indent=1)
