"""Logits parity of the arks_amd model vs HuggingFace transformers on the
tiny config (CPU, fp32) — validates model structure, RoPE, GQA attention and
the HF weight-name mapping end to end."""

import numpy as np
import pytest
import torch

from arks_amd.config import PRESET_CONFIGS, EngineConfig
from arks_amd.engine.forward_batch import ForwardBatch
from arks_amd.engine.kv_cache import BlockAllocator
from arks_amd.models import create_model

transformers = pytest.importorskip("transformers")


def build_pair(tmp_path):
    cfg = PRESET_CONFIGS["tiny"]
    from arks_amd.loader.safetensors_loader import save_random_checkpoint

    save_random_checkpoint(cfg, str(tmp_path), seed=7)

    ours = create_model(cfg, dtype=torch.float32)
    from arks_amd.loader.safetensors_loader import load_model_weights

    load_model_weights(ours, str(tmp_path), torch.device("cpu"))

    hf_cfg = transformers.Qwen2Config(
        vocab_size=cfg.vocab_size,
        hidden_size=cfg.hidden_size,
        intermediate_size=cfg.intermediate_size,
        num_hidden_layers=cfg.num_hidden_layers,
        num_attention_heads=cfg.num_attention_heads,
        num_key_value_heads=cfg.num_key_value_heads,
        rms_norm_eps=cfg.rms_norm_eps,
        rope_theta=cfg.rope_theta,
        max_position_embeddings=cfg.max_position_embeddings,
        tie_word_embeddings=cfg.tie_word_embeddings,
        attention_dropout=0.0,
    )
    hf = transformers.Qwen2ForCausalLM.from_pretrained(
        str(tmp_path), config=hf_cfg, torch_dtype=torch.float32
    )
    hf.eval()
    return cfg, ours, hf


def forward_ours(model, cfg, token_ids):
    n = len(token_ids)
    bs = 16
    nb = BlockAllocator.blocks_needed(n, bs)
    nkv = cfg.num_key_value_heads
    caches = [
        (
            torch.zeros(nb, nkv, bs, cfg.head_dim, dtype=torch.float32),
            torch.zeros(nb, nkv, bs, cfg.head_dim, dtype=torch.float32),
        )
        for _ in range(cfg.num_hidden_layers)
    ]
    fb = ForwardBatch(
        is_prefill=True,
        input_ids=torch.tensor(token_ids, dtype=torch.int64),
        positions=torch.arange(n, dtype=torch.int64),
        slot_mapping=torch.arange(n, dtype=torch.int64),
        cu_seqlens=torch.tensor([0, n], dtype=torch.int32),
        seq_lens_list=[n],
        logits_indices=None,  # all positions
    )
    with torch.no_grad():
        return model(fb, caches)


def test_logits_match_transformers(tmp_path):
    torch.manual_seed(0)
    cfg, ours, hf = build_pair(tmp_path)
    ids = [1, 17, 300, 42, 42, 7, 99, 123, 8, 55, 4]
    logits = forward_ours(ours, cfg, ids)
    with torch.no_grad():
        hf_logits = hf(torch.tensor([ids])).logits[0]
    # fp32 end to end; small numeric drift from op ordering only
    diff = (logits - hf_logits).abs().max().item()
    assert diff < 2e-3, f"max logits diff {diff}"
    # and the next-token argmax ranking agrees everywhere
    assert torch.equal(logits.argmax(-1), hf_logits.argmax(-1))


def test_llama_rope_scaling_logits_match_transformers(tmp_path):
    """Llama arch (no qkv bias) + llama3 rope_scaling vs transformers."""
    import dataclasses

    from arks_amd.loader.safetensors_loader import (
        load_model_weights,
        save_random_checkpoint,
    )

    base = PRESET_CONFIGS["tiny"]
    cfg = dataclasses.replace(
        base,
        architecture="LlamaForCausalLM",
        attention_bias=False,
        rope_theta=500000.0,
        rope_scaling={
            "rope_type": "llama3", "factor": 8.0, "low_freq_factor": 1.0,
            "high_freq_factor": 4.0, "original_max_position_embeddings": 256,
        },
    )
    save_random_checkpoint(cfg, str(tmp_path), seed=11)
    ours = create_model(cfg, dtype=torch.float32)
    load_model_weights(ours, str(tmp_path), torch.device("cpu"))

    hf_cfg = transformers.LlamaConfig(
        vocab_size=cfg.vocab_size,
        hidden_size=cfg.hidden_size,
        intermediate_size=cfg.intermediate_size,
        num_hidden_layers=cfg.num_hidden_layers,
        num_attention_heads=cfg.num_attention_heads,
        num_key_value_heads=cfg.num_key_value_heads,
        rms_norm_eps=cfg.rms_norm_eps,
        rope_theta=cfg.rope_theta,
        max_position_embeddings=cfg.max_position_embeddings,
        rope_scaling=dict(cfg.rope_scaling),
        tie_word_embeddings=cfg.tie_word_embeddings,
        attention_bias=False,
        attention_dropout=0.0,
    )
    hf = transformers.LlamaForCausalLM.from_pretrained(
        str(tmp_path), config=hf_cfg, torch_dtype=torch.float32
    )
    hf.eval()
    ids = [3, 1, 4, 1, 5, 9, 2, 6, 5, 3]
    logits = forward_ours(ours, cfg, ids)
    with torch.no_grad():
        hf_logits = hf(torch.tensor([ids])).logits[0]
    diff = (logits - hf_logits).abs().max().item()
    assert diff < 2e-3, f"max logits diff {diff}"
    assert torch.equal(logits.argmax(-1), hf_logits.argmax(-1))


def test_mixtral_moe_logits_match_transformers(tmp_path):
    """Sparse-MoE block (router + top-2 experts) vs transformers Mixtral."""
    from arks_amd.loader.safetensors_loader import (
        load_model_weights,
        save_random_checkpoint,
    )

    cfg = PRESET_CONFIGS["tiny-moe"]
    save_random_checkpoint(cfg, str(tmp_path), seed=13)
    ours = create_model(cfg, dtype=torch.float32)
    load_model_weights(ours, str(tmp_path), torch.device("cpu"))

    hf_cfg = transformers.MixtralConfig(
        vocab_size=cfg.vocab_size,
        hidden_size=cfg.hidden_size,
        intermediate_size=cfg.intermediate_size,
        num_hidden_layers=cfg.num_hidden_layers,
        num_attention_heads=cfg.num_attention_heads,
        num_key_value_heads=cfg.num_key_value_heads,
        rms_norm_eps=cfg.rms_norm_eps,
        rope_theta=cfg.rope_theta,
        max_position_embeddings=cfg.max_position_embeddings,
        num_local_experts=cfg.num_local_experts,
        num_experts_per_tok=cfg.num_experts_per_tok,
        tie_word_embeddings=cfg.tie_word_embeddings,
        attention_dropout=0.0,
    )
    hf = transformers.MixtralForCausalLM.from_pretrained(
        str(tmp_path), config=hf_cfg, torch_dtype=torch.float32
    )
    hf.eval()
    # > MoEMLP.DENSE_TOKENS so this exercises the sorted sparse dispatch
    ids = [(13 * i + 7) % 500 for i in range(80)]
    logits = forward_ours(ours, cfg, ids)
    with torch.no_grad():
        hf_logits = hf(torch.tensor([ids])).logits[0]
    diff = (logits - hf_logits).abs().max().item()
    assert diff < 2e-3, f"max logits diff {diff}"
    assert torch.equal(logits.argmax(-1), hf_logits.argmax(-1))


def test_qwen3_qk_norm_logits_match_transformers(tmp_path):
    """Qwen3 arch (per-head q/k RMSNorm before RoPE) vs transformers."""
    from arks_amd.loader.safetensors_loader import (
        load_model_weights,
        save_random_checkpoint,
    )

    cfg = PRESET_CONFIGS["tiny-qwen3"]
    save_random_checkpoint(cfg, str(tmp_path), seed=17)
    ours = create_model(cfg, dtype=torch.float32)
    load_model_weights(ours, str(tmp_path), torch.device("cpu"))

    hf_cfg = transformers.Qwen3Config(
        vocab_size=cfg.vocab_size,
        hidden_size=cfg.hidden_size,
        intermediate_size=cfg.intermediate_size,
        num_hidden_layers=cfg.num_hidden_layers,
        num_attention_heads=cfg.num_attention_heads,
        num_key_value_heads=cfg.num_key_value_heads,
        head_dim=cfg.head_dim,
        rms_norm_eps=cfg.rms_norm_eps,
        rope_theta=cfg.rope_theta,
        max_position_embeddings=cfg.max_position_embeddings,
        tie_word_embeddings=cfg.tie_word_embeddings,
        attention_bias=False,
        attention_dropout=0.0,
    )
    hf = transformers.Qwen3ForCausalLM.from_pretrained(
        str(tmp_path), config=hf_cfg, torch_dtype=torch.float32
    )
    hf.eval()
    ids = [9, 1, 44, 8, 250, 3, 77]
    logits = forward_ours(ours, cfg, ids)
    with torch.no_grad():
        hf_logits = hf(torch.tensor([ids])).logits[0]
    diff = (logits - hf_logits).abs().max().item()
    assert diff < 2e-3, f"max logits diff {diff}"
    assert torch.equal(logits.argmax(-1), hf_logits.argmax(-1))


def test_qwen3_moe_logits_match_transformers(tmp_path):
    """Qwen3-MoE (qk-norm + sparse MoE, norm_topk_prob=False) vs transformers."""
    from arks_amd.loader.safetensors_loader import (
        load_model_weights,
        save_random_checkpoint,
    )

    cfg = PRESET_CONFIGS["tiny-qwen3moe"]
    save_random_checkpoint(cfg, str(tmp_path), seed=19)
    ours = create_model(cfg, dtype=torch.float32)
    load_model_weights(ours, str(tmp_path), torch.device("cpu"))

    hf_cfg = transformers.Qwen3MoeConfig(
        vocab_size=cfg.vocab_size,
        hidden_size=cfg.hidden_size,
        intermediate_size=cfg.intermediate_size,
        moe_intermediate_size=cfg.moe_intermediate_size,
        num_hidden_layers=cfg.num_hidden_layers,
        num_attention_heads=cfg.num_attention_heads,
        num_key_value_heads=cfg.num_key_value_heads,
        head_dim=cfg.head_dim,
        rms_norm_eps=cfg.rms_norm_eps,
        rope_theta=cfg.rope_theta,
        max_position_embeddings=cfg.max_position_embeddings,
        num_experts=cfg.num_local_experts,
        num_experts_per_tok=cfg.num_experts_per_tok,
        norm_topk_prob=cfg.norm_topk_prob,
        decoder_sparse_step=1,
        mlp_only_layers=[],
        tie_word_embeddings=cfg.tie_word_embeddings,
        attention_bias=False,
        attention_dropout=0.0,
    )
    hf = transformers.Qwen3MoeForCausalLM.from_pretrained(
        str(tmp_path), config=hf_cfg, torch_dtype=torch.float32
    )
    hf.eval()
    ids = [11, 6, 42, 9, 127, 8]
    logits = forward_ours(ours, cfg, ids)
    with torch.no_grad():
        hf_logits = hf(torch.tensor([ids])).logits[0]
    diff = (logits - hf_logits).abs().max().item()
    assert diff < 2e-3, f"max logits diff {diff}"
    assert torch.equal(logits.argmax(-1), hf_logits.argmax(-1))


def test_qwen2_moe_shared_expert_logits_match_transformers(tmp_path):
    """Qwen2-MoE: sparse experts + sigmoid-gated shared expert."""
    from arks_amd.loader.safetensors_loader import (
        load_model_weights,
        save_random_checkpoint,
    )

    cfg = PRESET_CONFIGS["tiny-qwen2moe"]
    save_random_checkpoint(cfg, str(tmp_path), seed=23)
    ours = create_model(cfg, dtype=torch.float32)
    load_model_weights(ours, str(tmp_path), torch.device("cpu"))

    hf_cfg = transformers.Qwen2MoeConfig(
        vocab_size=cfg.vocab_size,
        hidden_size=cfg.hidden_size,
        intermediate_size=cfg.intermediate_size,
        moe_intermediate_size=cfg.moe_intermediate_size,
        shared_expert_intermediate_size=cfg.shared_expert_intermediate_size,
        num_hidden_layers=cfg.num_hidden_layers,
        num_attention_heads=cfg.num_attention_heads,
        num_key_value_heads=cfg.num_key_value_heads,
        rms_norm_eps=cfg.rms_norm_eps,
        rope_theta=cfg.rope_theta,
        max_position_embeddings=cfg.max_position_embeddings,
        num_experts=cfg.num_local_experts,
        num_experts_per_tok=cfg.num_experts_per_tok,
        norm_topk_prob=cfg.norm_topk_prob,
        decoder_sparse_step=1,
        mlp_only_layers=[],
        tie_word_embeddings=cfg.tie_word_embeddings,
        attention_dropout=0.0,
    )
    hf = transformers.Qwen2MoeForCausalLM.from_pretrained(
        str(tmp_path), config=hf_cfg, torch_dtype=torch.float32
    )
    hf.eval()
    # both dense (<=64) and sparse (>64) dispatch paths
    for ids in ([5, 77, 2, 30, 9], [(11 * i + 3) % 500 for i in range(80)]):
        logits = forward_ours(ours, cfg, ids)
        with torch.no_grad():
            hf_logits = hf(torch.tensor([ids])).logits[0]
        diff = (logits - hf_logits).abs().max().item()
        assert diff < 2e-3, f"max logits diff {diff}"
        assert torch.equal(logits.argmax(-1), hf_logits.argmax(-1))


def test_wide_expert_grouped_dispatch_matches_naive():
    """E=40 (>32) sparse grouped dispatch — the path that memory-faulted on
    HW with transposed-view bmm operands (scripts/probe_bmm_fault.py) — must
    match a naive per-token fp32 loop."""
    import dataclasses

    from arks_amd.config import PRESET_CONFIGS
    from arks_amd.models.llama_family import MoEMLP

    cfg = dataclasses.replace(
        PRESET_CONFIGS["tiny-moe"], num_local_experts=40, num_experts_per_tok=4
    )
    torch.manual_seed(3)
    m = MoEMLP(cfg)
    for p in m.parameters():
        p.data.normal_(0, 0.05)
    T = 100  # > DENSE_TOKENS -> sparse path; 40 experts -> grouped bmm
    x = torch.randn(T, cfg.hidden_size, dtype=torch.bfloat16)
    y = m(x)

    # naive reference: per-token, per-selected-expert in fp32
    logits = (x.float() @ m.gate.float().t())
    probs = torch.softmax(logits, dim=-1)
    w, sel = probs.topk(cfg.num_experts_per_tok, dim=-1)
    if cfg.norm_topk_prob:
        w = w / w.sum(-1, keepdim=True)
    ref = torch.zeros(T, cfg.hidden_size)
    for t in range(T):
        for j in range(cfg.num_experts_per_tok):
            e = int(sel[t, j])
            # w13/w2 are stored pre-transposed [in, out]
            gu = x[t].float() @ m.w13[e].float()
            g, u = gu.chunk(2)
            h = torch.nn.functional.silu(g) * u
            ref[t] += w[t, j].float() * (h @ m.w2[e].float())
    rel = (y.float() - ref).abs().mean() / ref.abs().mean()
    assert rel < 0.05, rel


def test_loader_chunked_matches_whole(tmp_path):
    """Chunked streaming (tiny chunk limit -> many chunks with fused-weight
    partner grouping) must load identically to one big chunk."""
    import dataclasses

    from arks_amd.config import PRESET_CONFIGS
    from arks_amd.loader.safetensors_loader import (
        load_model_weights,
        save_random_checkpoint,
    )
    from arks_amd.models import create_model

    cfg = PRESET_CONFIGS["tiny"]
    save_random_checkpoint(cfg, str(tmp_path), seed=13)
    whole = create_model(cfg)
    load_model_weights(whole, str(tmp_path), torch.device("cpu"),
                       chunk_bytes=1 << 40)
    chunked = create_model(cfg)
    load_model_weights(chunked, str(tmp_path), torch.device("cpu"),
                       chunk_bytes=1024)  # forces a chunk per few tensors
    sw = dict(whole.named_parameters())
    for name, p in chunked.named_parameters():
        assert torch.equal(p, sw[name]), name


def test_parse_header_matches_safetensors_lib(tmp_path):
    """The direct header parser (GPU fast path, loader/safetensors_loader.py
    _parse_header) agrees with the official safetensors library on names,
    dtypes, shapes, and byte ranges — including a __metadata__ entry and
    0-d / empty tensors."""
    import torch
    from safetensors import safe_open
    from safetensors.torch import save_file

    from arks_amd.loader.safetensors_loader import _parse_header

    tensors = {
        "a.weight": torch.randn(3, 5, dtype=torch.bfloat16),
        "b.bias": torch.randn(7, dtype=torch.float32),
        "c.scalar": torch.tensor(2.5, dtype=torch.float16),
        "d.empty": torch.empty(0, 4, dtype=torch.bfloat16),
        "e.int": torch.arange(6, dtype=torch.int32).reshape(2, 3),
    }
    path = str(tmp_path / "model.safetensors")
    save_file(tensors, path, metadata={"format": "pt"})

    data_start, ts = _parse_header(path)
    assert sorted(n for n, *_ in ts) == sorted(tensors)
    by_name = {n: (dt, shape, o0, o1) for n, dt, shape, o0, o1 in ts}
    with safe_open(path, framework="pt") as f:
        for name, ref in tensors.items():
            dt, shape, o0, o1 = by_name[name]
            assert list(shape) == list(ref.shape)
            assert dt == ref.dtype
            assert o1 - o0 == ref.numel() * ref.element_size()
            if ref.numel() == 0:
                continue  # frombuffer rejects empty buffers
            # byte range reproduces the tensor exactly
            with open(path, "rb") as raw:
                raw.seek(data_start + o0)
                buf = raw.read(o1 - o0)
            got = torch.frombuffer(bytearray(buf), dtype=ref.dtype)
            assert torch.equal(got.reshape(ref.shape), f.get_tensor(name))
    # offsets are sorted and non-overlapping (the chunked reader relies
    # on this to stream spans sequentially)
    offs = [(o0, o1) for *_, o0, o1 in ts]
    assert offs == sorted(offs) and all(
        offs[i][1] <= offs[i + 1][0] for i in range(len(offs) - 1))


def test_smollm3_nope_logits_match_transformers(tmp_path):
    """SmolLM3: llama arch with NoPE layers (no_rope_layers[i]==0 attends
    without positional encoding — HF modeling_smollm3.py:200-226) and tied
    embeddings, vs transformers."""
    import dataclasses

    from arks_amd.loader.safetensors_loader import (
        load_model_weights,
        save_random_checkpoint,
    )

    base = PRESET_CONFIGS["tiny"]
    cfg = dataclasses.replace(
        base,
        architecture="SmolLM3ForCausalLM",
        attention_bias=False,
        tie_word_embeddings=True,
        no_rope_layers=[1, 0],  # layer 1 is NoPE
    )
    save_random_checkpoint(cfg, str(tmp_path), seed=23)
    ours = create_model(cfg, dtype=torch.float32)
    load_model_weights(ours, str(tmp_path), torch.device("cpu"))

    hf_cfg = transformers.SmolLM3Config(
        vocab_size=cfg.vocab_size,
        hidden_size=cfg.hidden_size,
        intermediate_size=cfg.intermediate_size,
        num_hidden_layers=cfg.num_hidden_layers,
        num_attention_heads=cfg.num_attention_heads,
        num_key_value_heads=cfg.num_key_value_heads,
        rms_norm_eps=cfg.rms_norm_eps,
        rope_theta=cfg.rope_theta,
        max_position_embeddings=cfg.max_position_embeddings,
        tie_word_embeddings=True,
        use_sliding_window=False,
        no_rope_layers=[1, 0],
        attention_dropout=0.0,
        pad_token_id=0,
        bos_token_id=1,
        eos_token_id=2,
    )
    hf = transformers.SmolLM3ForCausalLM.from_pretrained(
        str(tmp_path), config=hf_cfg, torch_dtype=torch.float32
    )
    hf.eval()
    ids = [3, 1, 4, 1, 5, 9, 2, 6, 5, 3]
    logits = forward_ours(ours, cfg, ids)
    with torch.no_grad():
        hf_logits = hf(torch.tensor([ids])).logits[0]
    diff = (logits - hf_logits).abs().max().item()
    assert diff < 2e-3, f"max logits diff {diff}"
    assert torch.equal(logits.argmax(-1), hf_logits.argmax(-1))
