"""Model and engine configuration.

ModelConfig is HF-config-driven: it reads the same config.json fields the
reference's delegated runtimes (vLLM/SGLang at reference
arksapplication_controller.go:941-1014) consume, so any Qwen2/Llama-family
checkpoint directory works unchanged.
"""

from __future__ import annotations

import json
import os
from dataclasses import dataclass


@dataclass
class ModelConfig:
    architecture: str = "Qwen2ForCausalLM"
    vocab_size: int = 151936
    hidden_size: int = 3584
    intermediate_size: int = 18944
    num_hidden_layers: int = 28
    num_attention_heads: int = 28
    num_key_value_heads: int = 4
    head_dim: int = 128
    rms_norm_eps: float = 1e-6
    rope_theta: float = 1000000.0
    max_position_embeddings: int = 32768
    # HF rope_scaling dict: {"rope_type": "llama3"|"yarn"|"linear", ...}
    rope_scaling: dict | None = None
    # Mixtral-style sparse MoE (0 experts = dense MLP)
    num_local_experts: int = 0
    num_experts_per_tok: int = 2
    moe_intermediate_size: int | None = None  # per-expert FFN width (MoE)
    shared_expert_intermediate_size: int | None = None  # Qwen2-MoE
    norm_topk_prob: bool = True  # renormalize top-k routing weights
    sliding_window: int | None = None  # sliding-window attention size
    # Qwen2-style per-layer mixing: layers >= max_window_layers slide,
    # lower layers use full attention (HF layer_types overrides when set)
    max_window_layers: int | None = None
    layer_types: list | None = None
    qk_norm: bool = False  # Qwen3: per-head RMSNorm on q/k before RoPE
    # SmolLM3 NoPE: no_rope_layers[i] == 1 -> layer i USES RoPE, 0 -> the
    # layer attends without positional encoding (HF configuration_smollm3
    # semantics; default = one NoPE layer every no_rope_layer_interval)
    no_rope_layers: list | None = None
    tie_word_embeddings: bool = False
    attention_bias: bool = True  # qwen2 has qkv bias; llama does not
    eos_token_id: int = 151645
    bos_token_id: int = 151643
    torch_dtype: str = "bfloat16"

    def __post_init__(self):
        # Sliding-window attention is implemented end to end: the
        # attention kernels mask keys outside each query's per-layer
        # window (layer_window: Mistral = all layers; Qwen2
        # max_window_layers / layer_types mixing honored) and the engine
        # drops out-of-window KV pages when EVERY layer slides
        # (uniform_window).
        if self.sliding_window is not None and self.sliding_window <= 0:
            raise ValueError(
                f"invalid sliding_window={self.sliding_window}"
            )

    def layer_uses_rope(self, layer_idx: int) -> bool:
        if self.no_rope_layers is None:
            return True
        return bool(self.no_rope_layers[layer_idx])

    def layer_window(self, layer_idx: int) -> int:
        """Effective attention window for one layer (0 = full attention).
        HF Qwen2 semantics: layer_types wins when present; otherwise
        layers >= max_window_layers slide and lower layers are full."""
        if self.sliding_window is None:
            return 0
        if self.layer_types is not None:
            return (self.sliding_window
                    if self.layer_types[layer_idx] == "sliding_attention"
                    else 0)
        if self.max_window_layers is not None:
            return self.sliding_window if layer_idx >= self.max_window_layers else 0
        return self.sliding_window

    def uniform_window(self) -> int:
        """The window when EVERY layer slides (else 0): KV page dropping
        and the prefix-cache gate need all layers windowed — a single
        full-attention layer keeps the whole history live."""
        ws = [self.layer_window(i) for i in range(self.num_hidden_layers)]
        return ws[0] if ws and all(w == ws[0] and w > 0 for w in ws) else 0

    @classmethod
    def from_hf_config(cls, cfg: dict) -> "ModelConfig":
        arch = (cfg.get("architectures") or ["Qwen2ForCausalLM"])[0]
        num_heads = cfg.get("num_attention_heads", 28)
        hidden = cfg.get("hidden_size", 3584)
        eos = cfg.get("eos_token_id", 151645)
        if isinstance(eos, list):
            eos = eos[0]
        return cls(
            architecture=arch,
            vocab_size=cfg.get("vocab_size", 151936),
            hidden_size=hidden,
            intermediate_size=cfg.get("intermediate_size", 18944),
            num_hidden_layers=cfg.get("num_hidden_layers", 28),
            num_attention_heads=num_heads,
            num_key_value_heads=cfg.get("num_key_value_heads", num_heads),
            head_dim=cfg.get("head_dim", hidden // num_heads),
            rms_norm_eps=cfg.get("rms_norm_eps", 1e-6),
            rope_theta=cfg.get("rope_theta", 10000.0),
            max_position_embeddings=cfg.get("max_position_embeddings", 32768),
            rope_scaling=cfg.get("rope_scaling"),
            qk_norm=arch.startswith("Qwen3"),
            no_rope_layers=(
                cfg["no_rope_layers"] if cfg.get("no_rope_layers") is not None
                else [int((i + 1) % cfg.get("no_rope_layer_interval", 4) != 0)
                      for i in range(cfg.get("num_hidden_layers", 28))]
                if arch == "SmolLM3ForCausalLM" else None),
            num_local_experts=cfg.get("num_local_experts",
                                      cfg.get("num_experts", 0)),
            # Qwen2-family configs declare a window but disable it
            sliding_window=(cfg.get("sliding_window")
                            if cfg.get("use_sliding_window", True) else None),
            max_window_layers=cfg.get("max_window_layers"),
            layer_types=cfg.get("layer_types"),
            num_experts_per_tok=cfg.get("num_experts_per_tok", 2),
            moe_intermediate_size=cfg.get("moe_intermediate_size"),
            shared_expert_intermediate_size=cfg.get(
                "shared_expert_intermediate_size"),
            norm_topk_prob=cfg.get("norm_topk_prob", "num_local_experts" in cfg),
            tie_word_embeddings=cfg.get("tie_word_embeddings", False),
            attention_bias=cfg.get("attention_bias", arch == "Qwen2ForCausalLM"
                                   or arch == "Qwen2MoeForCausalLM"),
            eos_token_id=eos,
            bos_token_id=cfg.get("bos_token_id", 1),
            torch_dtype=cfg.get("torch_dtype", "bfloat16"),
        )

    @classmethod
    def from_pretrained(cls, model_path: str) -> "ModelConfig":
        with open(os.path.join(model_path, "config.json")) as f:
            return cls.from_hf_config(json.load(f))

    @property
    def num_qo_heads(self) -> int:
        return self.num_attention_heads


# Named synthetic configs for benchmarks / tests (random-init weights; no
# network access for real checkpoints — BASELINE.json's models by shape).
PRESET_CONFIGS: dict[str, ModelConfig] = {
    "qwen2.5-0.5b": ModelConfig(
        architecture="Qwen2ForCausalLM",
        vocab_size=151936,
        hidden_size=896,
        intermediate_size=4864,
        num_hidden_layers=24,
        num_attention_heads=14,
        num_key_value_heads=2,
        head_dim=64,
        rope_theta=1000000.0,
        tie_word_embeddings=True,
    ),
    "qwen2.5-7b": ModelConfig(
        architecture="Qwen2ForCausalLM",
        vocab_size=152064,
        hidden_size=3584,
        intermediate_size=18944,
        num_hidden_layers=28,
        num_attention_heads=28,
        num_key_value_heads=4,
        head_dim=128,
        rope_theta=1000000.0,
    ),
    "qwen2.5-32b": ModelConfig(
        architecture="Qwen2ForCausalLM",
        vocab_size=152064,
        hidden_size=5120,
        intermediate_size=27648,
        num_hidden_layers=64,
        num_attention_heads=40,
        num_key_value_heads=8,
        head_dim=128,
        rope_theta=1000000.0,
    ),
    "llama-3-8b": ModelConfig(
        architecture="LlamaForCausalLM",
        vocab_size=128256,
        hidden_size=4096,
        intermediate_size=14336,
        num_hidden_layers=32,
        num_attention_heads=32,
        num_key_value_heads=8,
        head_dim=128,
        rms_norm_eps=1e-5,
        rope_theta=500000.0,
        attention_bias=False,
        eos_token_id=128001,
        bos_token_id=128000,
    ),
    "llama-3-70b": ModelConfig(
        architecture="LlamaForCausalLM",
        vocab_size=128256,
        hidden_size=8192,
        intermediate_size=28672,
        num_hidden_layers=80,
        num_attention_heads=64,
        num_key_value_heads=8,
        head_dim=128,
        rms_norm_eps=1e-5,
        rope_theta=500000.0,
        attention_bias=False,
        eos_token_id=128001,
        bos_token_id=128000,
    ),
    "mistral-7b": ModelConfig(  # v0.1: global sliding window 4096
        architecture="MistralForCausalLM",
        vocab_size=32000,
        hidden_size=4096,
        intermediate_size=14336,
        num_hidden_layers=32,
        num_attention_heads=32,
        num_key_value_heads=8,
        head_dim=128,
        rms_norm_eps=1e-5,
        rope_theta=10000.0,
        attention_bias=False,
        sliding_window=4096,
        eos_token_id=2,
        bos_token_id=1,
    ),
    "mixtral-8x7b": ModelConfig(
        architecture="MixtralForCausalLM",
        vocab_size=32000,
        hidden_size=4096,
        intermediate_size=14336,  # per expert
        num_hidden_layers=32,
        num_attention_heads=32,
        num_key_value_heads=8,
        head_dim=128,
        rms_norm_eps=1e-5,
        rope_theta=1000000.0,
        attention_bias=False,
        num_local_experts=8,
        num_experts_per_tok=2,
        eos_token_id=2,
        bos_token_id=1,
    ),
    "tiny-moe": ModelConfig(  # CPU-test-sized sparse MoE
        architecture="MixtralForCausalLM",
        vocab_size=512,
        hidden_size=128,
        intermediate_size=192,
        num_hidden_layers=2,
        num_attention_heads=4,
        num_key_value_heads=2,
        head_dim=32,
        rope_theta=10000.0,
        max_position_embeddings=2048,
        attention_bias=False,
        num_local_experts=4,
        num_experts_per_tok=2,
        eos_token_id=2,
        bos_token_id=1,
    ),
    "qwen3-8b": ModelConfig(
        architecture="Qwen3ForCausalLM",
        vocab_size=151936,
        hidden_size=4096,
        intermediate_size=12288,
        num_hidden_layers=36,
        num_attention_heads=32,
        num_key_value_heads=8,
        head_dim=128,
        rope_theta=1000000.0,
        attention_bias=False,
        qk_norm=True,
    ),
    "tiny-qwen3": ModelConfig(  # CPU-test-sized qk-norm arch
        architecture="Qwen3ForCausalLM",
        vocab_size=512,
        hidden_size=128,
        intermediate_size=256,
        num_hidden_layers=2,
        num_attention_heads=4,
        num_key_value_heads=2,
        head_dim=32,
        rope_theta=10000.0,
        max_position_embeddings=2048,
        attention_bias=False,
        qk_norm=True,
        eos_token_id=2,
        bos_token_id=1,
    ),
    "tiny-qwen2moe": ModelConfig(  # CPU-test-sized Qwen2-MoE (shared expert)
        architecture="Qwen2MoeForCausalLM",
        vocab_size=512,
        hidden_size=128,
        intermediate_size=256,
        moe_intermediate_size=96,
        shared_expert_intermediate_size=160,
        num_hidden_layers=2,
        num_attention_heads=4,
        num_key_value_heads=2,
        head_dim=32,
        rope_theta=10000.0,
        max_position_embeddings=2048,
        attention_bias=True,
        num_local_experts=4,
        num_experts_per_tok=2,
        norm_topk_prob=False,
        eos_token_id=2,
        bos_token_id=1,
    ),
    "tiny-qwen3moe": ModelConfig(  # CPU-test-sized Qwen3-MoE
        architecture="Qwen3MoeForCausalLM",
        vocab_size=512,
        hidden_size=128,
        intermediate_size=256,
        moe_intermediate_size=96,
        num_hidden_layers=2,
        num_attention_heads=4,
        num_key_value_heads=2,
        head_dim=32,
        rope_theta=10000.0,
        max_position_embeddings=2048,
        attention_bias=False,
        qk_norm=True,
        num_local_experts=4,
        num_experts_per_tok=2,
        norm_topk_prob=False,
        eos_token_id=2,
        bos_token_id=1,
    ),
    "qwen3-30b-a3b": ModelConfig(  # Qwen3-30B-A3B MoE shape
        architecture="Qwen3MoeForCausalLM",
        vocab_size=151936,
        hidden_size=2048,
        intermediate_size=6144,
        moe_intermediate_size=768,
        num_hidden_layers=48,
        num_attention_heads=32,
        num_key_value_heads=4,
        head_dim=128,
        rope_theta=1000000.0,
        attention_bias=False,
        qk_norm=True,
        num_local_experts=128,
        num_experts_per_tok=8,
        norm_topk_prob=True,
    ),
    "tiny-moe-gpu": ModelConfig(  # GPU-test-sized sparse MoE (head_dim 128)
        architecture="MixtralForCausalLM",
        vocab_size=2048,
        hidden_size=512,
        intermediate_size=256,
        num_hidden_layers=2,
        num_attention_heads=4,
        num_key_value_heads=2,
        head_dim=128,
        rope_theta=10000.0,
        max_position_embeddings=4096,
        attention_bias=False,
        num_local_experts=4,
        num_experts_per_tok=2,
        eos_token_id=2,
        bos_token_id=1,
    ),
    "smollm3-3b": ModelConfig(
        architecture="SmolLM3ForCausalLM",
        vocab_size=128256,
        hidden_size=2048,
        intermediate_size=11008,
        num_hidden_layers=36,
        num_attention_heads=16,
        num_key_value_heads=4,
        head_dim=128,
        rms_norm_eps=1e-6,
        rope_theta=5000000.0,
        max_position_embeddings=65536,
        tie_word_embeddings=True,
        attention_bias=False,
        no_rope_layers=[int((i + 1) % 4 != 0) for i in range(36)],
        eos_token_id=128012,
        bos_token_id=128000,
    ),
    "tiny-smollm3": ModelConfig(  # CPU-test-sized, NoPE on odd layers
        architecture="SmolLM3ForCausalLM",
        vocab_size=512,
        hidden_size=128,
        intermediate_size=256,
        num_hidden_layers=2,
        num_attention_heads=4,
        num_key_value_heads=2,
        head_dim=32,
        rope_theta=10000.0,
        max_position_embeddings=2048,
        tie_word_embeddings=True,
        attention_bias=False,
        no_rope_layers=[1, 0],
        eos_token_id=2,
        bos_token_id=1,
    ),
    "tiny": ModelConfig(  # CPU-test-sized
        architecture="Qwen2ForCausalLM",
        vocab_size=512,
        hidden_size=128,
        intermediate_size=256,
        num_hidden_layers=2,
        num_attention_heads=4,
        num_key_value_heads=2,
        head_dim=32,
        rope_theta=10000.0,
        max_position_embeddings=2048,
        eos_token_id=2,
        bos_token_id=1,
    ),
    "tiny-gpu": ModelConfig(  # GPU-test-sized (head_dim 128 kernels)
        architecture="Qwen2ForCausalLM",
        vocab_size=2048,
        hidden_size=512,
        intermediate_size=1024,
        num_hidden_layers=2,
        num_attention_heads=4,
        num_key_value_heads=2,
        head_dim=128,
        rope_theta=10000.0,
        max_position_embeddings=4096,
        eos_token_id=2,
        bos_token_id=1,
    ),
}


@dataclass
class EngineConfig:
    model_path: str | None = None  # dir with config.json + *.safetensors
    preset: str | None = None  # or a PRESET_CONFIGS key (random-init)
    served_model_name: str = "model"
    tensor_parallel_size: int = 1
    block_size: int = 16  # KV page size (tokens) — fixed by the decode kernel
    gpu_memory_utilization: float = 0.90
    max_num_seqs: int = 256
    max_num_batched_tokens: int = 8192
    max_model_len: int = 8192
    kv_cache_blocks: int | None = None  # override (else sized from free HBM)
    enforce_eager: bool = False  # disable hipGraph decode capture
    enable_prefix_caching: bool = True  # content-addressed KV block reuse
    # mixed batching: prefill chunks share a step with in-flight decodes so
    # long prompts never stall token streams (vLLM-v1-style scheduling)
    enable_mixed_batching: bool = True
    # None | "fp8": W8A8 OCP-e4m3 for qkv/gate_up/down/lm_head GEMMs
    # (dynamic per-token activation scales; o_proj and KV stay bf16)
    quantization: str | None = None
    # "auto" (bf16) | "fp8": e4m3 KV pages (halves the decode KV stream;
    # scale 1.0, opt-in like vLLM's --kv-cache-dtype fp8)
    kv_cache_dtype: str = "auto"
    # None | "ngram": prompt-lookup speculative decoding (draft tokens from
    # n-gram matches in the sequence's own history, verified in one extend
    # forward; greedy-exact, applied only to temperature-0 requests without
    # penalties/logprobs — same surface as vLLM's ngram speculator)
    speculative: str | None = None
    num_speculative_tokens: int = 4  # max draft length per step
    # speculative="draft": the draft model ("preset:<name>" or a checkpoint
    # path; must share the target's tokenizer/vocab)
    draft_model: str | None = None
    device: str = "cuda"
    dtype: str = "bfloat16"
    seed: int = 0

    def model_config(self) -> ModelConfig:
        if self.model_path:
            return ModelConfig.from_pretrained(self.model_path)
        if self.preset:
            return PRESET_CONFIGS[self.preset]
        raise ValueError("EngineConfig needs model_path or preset")
