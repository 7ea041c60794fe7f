"""fp8 W8A8 quantization path (CPU emulation; GPU numerics in
tests/test_ops_gpu.py + test_engine_gpu.py)."""

import torch

from arks_amd.config import EngineConfig
from arks_amd.engine import LLMEngine, SamplingParams
from arks_amd.ops import ref
from arks_amd.parallel.layers import ColumnParallelLinear, quantize_module_fp8


def test_quant_fp8_rows_ref():
    torch.manual_seed(0)
    x = torch.randn(5, 64, dtype=torch.bfloat16) * 3
    q, inv_s = ref.quant_fp8_rows(x)
    assert q.dtype == torch.float8_e4m3fn and inv_s.shape == (5,)
    xd = q.float() * inv_s[:, None]
    rel = (xd - x.float()).abs().max() / x.float().abs().max()
    assert rel < 0.07  # e4m3 mantissa step
    # zero row must not divide by zero
    x0 = torch.zeros(1, 64, dtype=torch.bfloat16)
    q0, s0 = ref.quant_fp8_rows(x0)
    assert torch.isfinite(s0).all() and (q0.float() == 0).all()


def test_fp8_linear_cpu_close_to_bf16():
    torch.manual_seed(1)
    lin = ColumnParallelLinear(128, 96, bias=True)
    lin.weight.data.normal_(0, 0.05)
    lin.bias.data.normal_(0, 0.05)
    x = torch.randn(7, 128, dtype=torch.bfloat16)
    y_ref = lin(x)
    quantize_module_fp8(lin)
    y_q = lin(x)
    rel = (y_q.float() - y_ref.float()).abs().mean() / y_ref.float().abs().mean()
    assert rel < 0.15, rel


def test_engine_fp8_cpu_runs_deterministic():
    def run():
        eng = LLMEngine(EngineConfig(
            preset="tiny", device="cpu", kv_cache_blocks=128, max_model_len=256,
            quantization="fp8", seed=2,
        ))
        return eng.generate([[3, 1, 4, 1, 5], [9, 2, 6]],
                            SamplingParams(max_tokens=6, ignore_eos=True))

    a, b = run(), run()
    assert a == b and all(len(o) == 6 for o in a)


def test_engine_fp8_kv_cache_cpu():
    """kv_cache_dtype=fp8: e4m3 pages end to end on the CPU ref path."""
    eng = LLMEngine(EngineConfig(
        preset="tiny", device="cpu", kv_cache_blocks=128, max_model_len=256,
        kv_cache_dtype="fp8", seed=12,
    ))
    assert eng.runner.kv_caches[0][0].dtype == torch.float8_e4m3fn
    out = eng.generate([[5, 2, 8, 1] * 5, [7] * 30],
                       SamplingParams(max_tokens=6, ignore_eos=True))
    assert all(len(o) == 6 for o in out)
    out2 = eng.generate([[5, 2, 8, 1] * 5, [7] * 30],
                        SamplingParams(max_tokens=6, ignore_eos=True))
    assert out == out2
