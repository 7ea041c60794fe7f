"""CPU sanity tests for the reference ops (the oracle the HIP kernels are
tested against) — internal consistency checks that run without a GPU."""

import math

import torch

from arks_amd.ops import ref


def test_rmsnorm_matches_manual():
    torch.manual_seed(0)
    x = torch.randn(4, 64)
    w = torch.randn(64)
    out = ref.rmsnorm(x, w, 1e-6)
    manual = x / (x.pow(2).mean(-1, keepdim=True) + 1e-6).sqrt() * w
    torch.testing.assert_close(out, manual, atol=1e-5, rtol=1e-5)


def test_decode_consistent_with_prefill_last_token():
    """Decoding the last token against the cache must equal the last row of
    full prefill attention."""
    torch.manual_seed(1)
    hq, hkv, hd, L, bs = 4, 2, 64, 37, 16
    q = torch.randn(L, hq, hd)
    k = torch.randn(L, hkv, hd)
    v = torch.randn(L, hkv, hd)
    cu = torch.tensor([0, L], dtype=torch.int32)
    scale = 1.0 / math.sqrt(hd)
    full = ref.attention_prefill_varlen(q, k, v, cu, scale)

    nb = (L + bs - 1) // bs
    kc = torch.zeros(nb, hkv, bs, hd)
    vc = torch.zeros(nb, hkv, bs, hd)
    slots = torch.arange(L, dtype=torch.int64)
    ref.reshape_and_cache(k, v, kc, vc, slots)
    bt = torch.arange(nb, dtype=torch.int32).unsqueeze(0)
    sl = torch.tensor([L], dtype=torch.int32)
    dec = ref.attention_decode_paged(q[-1:].clone(), kc, vc, bt, sl, scale)
    torch.testing.assert_close(dec[0], full[-1], atol=1e-4, rtol=1e-4)


def test_rope_preserves_norm():
    torch.manual_seed(2)
    hd = 64
    pos = torch.arange(10)
    q = torch.randn(10, 2 * hd)
    k = torch.randn(10, hd)
    cs = ref.rope_cos_sin_cache(hd, 32)
    q2, k2 = ref.rope_apply(pos, q, k, cs, hd)
    # Rotation preserves the norm of each head.
    torch.testing.assert_close(
        q2.view(10, 2, hd).norm(dim=-1), q.view(10, 2, hd).norm(dim=-1), atol=1e-4, rtol=1e-4
    )
    # position 0 is identity
    torch.testing.assert_close(q2[0], q[0], atol=1e-6, rtol=1e-6)


def test_gumbel_sample_greedy_when_t0():
    torch.manual_seed(3)
    logits = torch.randn(5, 100)
    temps = torch.zeros(5)
    u = torch.rand(5, 100)
    out = ref.gumbel_sample(logits, temps, u)
    torch.testing.assert_close(out, logits.argmax(-1))


def test_rope_scaling_matches_transformers():
    """llama3 + yarn rope_scaling pinned against transformers' rope_utils."""
    import torch
    from transformers import LlamaConfig
    from transformers.modeling_rope_utils import ROPE_INIT_FUNCTIONS

    from arks_amd.ops.ref import _scaled_inv_freq

    base_inv = 1.0 / (500000.0 ** (torch.arange(0, 128, 2, dtype=torch.float64) / 128))
    l3 = {"rope_type": "llama3", "factor": 8.0, "low_freq_factor": 1.0,
          "high_freq_factor": 4.0, "original_max_position_embeddings": 8192}
    cfg = LlamaConfig(rope_theta=500000.0, hidden_size=4096,
                      num_attention_heads=32,
                      max_position_embeddings=131072, rope_scaling=dict(l3))
    inv_hf, _ = ROPE_INIT_FUNCTIONS["llama3"](cfg, "cpu")
    mine, ms = _scaled_inv_freq(base_inv, l3, 500000.0, 128)
    assert ms == 1.0
    torch.testing.assert_close(mine, inv_hf.double(), rtol=1e-5, atol=0)

    yarn = {"rope_type": "yarn", "factor": 4.0,
            "original_max_position_embeddings": 4096}
    cfg2 = LlamaConfig(rope_theta=10000.0, hidden_size=4096,
                       num_attention_heads=32,
                       max_position_embeddings=16384, rope_scaling=dict(yarn))
    inv_hf2, att2 = ROPE_INIT_FUNCTIONS["yarn"](cfg2, "cpu")
    base2 = 1.0 / (10000.0 ** (torch.arange(0, 128, 2, dtype=torch.float64) / 128))
    mine2, ms2 = _scaled_inv_freq(base2, yarn, 10000.0, 128)
    torch.testing.assert_close(mine2, inv_hf2.double(), rtol=1e-5, atol=0)
    assert abs(ms2 - att2) < 1e-9
