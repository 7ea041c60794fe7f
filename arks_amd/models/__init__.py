"""Model registry: HF `architectures[0]` -> implementation class."""

from __future__ import annotations

from ..config import ModelConfig
from .llama_family import LlamaFamilyForCausalLM

_REGISTRY = {
    "Qwen2ForCausalLM": LlamaFamilyForCausalLM,
    "LlamaForCausalLM": LlamaFamilyForCausalLM,
    "MixtralForCausalLM": LlamaFamilyForCausalLM,  # sparse MoE MLP branch
    # Mistral v0.3+ ships sliding_window=null -> plain llama arch; models
    # that DO set a window are rejected at config load (no SWA kernels yet)
    "MistralForCausalLM": LlamaFamilyForCausalLM,
    "Qwen3ForCausalLM": LlamaFamilyForCausalLM,  # + per-head q/k RMSNorm
    "Qwen3MoeForCausalLM": LlamaFamilyForCausalLM,  # qk-norm + sparse MoE
    "Qwen2MoeForCausalLM": LlamaFamilyForCausalLM,  # MoE + shared expert
    # llama arch + NoPE every no_rope_layer_interval-th layer + tied embeds
    "SmolLM3ForCausalLM": LlamaFamilyForCausalLM,
}


def create_model(cfg: ModelConfig, dtype=None):
    import torch

    cls = _REGISTRY.get(cfg.architecture)
    if cls is None:
        raise ValueError(
            f"unsupported architecture {cfg.architecture!r}; known: {sorted(_REGISTRY)}"
        )
    return cls(cfg, dtype=dtype or torch.bfloat16)
