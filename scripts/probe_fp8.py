#!/usr/bin/env python3
"""Probe fp8 (OCP e4m3) GEMM support and speed on gfx950.

Checks which fp8 dtypes torch+hipBLASLt accept on this box, validates
numerics of torch._scaled_mm (tensorwise and rowwise scales) against a bf16
reference, and times the Qwen2.5-7B decode GEMM shapes at small M — the
weight-bandwidth-bound regime where fp8 weights should approach 2x bf16.
"""

import time

import torch

torch.manual_seed(0)
dev = "cuda"


def bench(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / iters * 1e6  # us


def main():
    print("torch", torch.__version__)
    for name in ("float8_e4m3fn", "float8_e4m3fnuz", "float8_e5m2"):
        print(name, hasattr(torch, name))

    dt = torch.float8_e4m3fn
    # correctness: tensorwise
    M, K, N = 64, 3584, 4608
    a = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
    w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.02
    ref = a @ w.t()

    def q_tensorwise(x):
        amax = x.abs().amax().clamp(min=1e-6)
        s = (448.0 / amax).float()
        return (x.float() * s).clamp(-448, 448).to(dt), (1.0 / s)

    aq, sa = q_tensorwise(a)
    wq, sw = q_tensorwise(w)
    try:
        out = torch._scaled_mm(aq, wq.t(), scale_a=sa, scale_b=sw,
                               out_dtype=torch.bfloat16)
        err = (out.float() - ref.float()).abs().mean() / ref.float().abs().mean()
        print(f"tensorwise _scaled_mm OK relerr={err:.4f}")
    except Exception as e:
        print("tensorwise _scaled_mm FAIL:", repr(e))

    # rowwise scales
    try:
        sa_r = (448.0 / a.abs().amax(dim=1, keepdim=True).clamp(min=1e-6)).float()
        sw_r = (448.0 / w.abs().amax(dim=1, keepdim=True).clamp(min=1e-6)).float()
        aq_r = (a.float() * sa_r).clamp(-448, 448).to(dt)
        wq_r = (w.float() * sw_r).clamp(-448, 448).to(dt)
        out = torch._scaled_mm(aq_r, wq_r.t(), scale_a=1.0 / sa_r,
                               scale_b=(1.0 / sw_r).t(), out_dtype=torch.bfloat16)
        err = (out.float() - ref.float()).abs().mean() / ref.float().abs().mean()
        print(f"rowwise _scaled_mm OK relerr={err:.4f}")
    except Exception as e:
        print("rowwise _scaled_mm FAIL:", repr(e))

    # speed on the decode shapes (per-GPU 7B)
    shapes = [("qkv", 3584, 4608), ("o", 4608 - 1024, 3584),
              ("gate_up", 3584, 37888), ("down", 18944, 3584),
              ("lm_head", 3584, 152064)]
    for Mtest in (16, 64, 256):
        print(f"--- M={Mtest}")
        for name, K, N in shapes:
            a = torch.randn(Mtest, K, device=dev, dtype=torch.bfloat16)
            w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.02
            t_bf = bench(lambda: a @ w.t())
            aq, sa = q_tensorwise(a)
            wq, sw = q_tensorwise(w)
            wqt = wq.t()
            try:
                t_f8 = bench(lambda: torch._scaled_mm(
                    aq, wqt, scale_a=sa, scale_b=sw, out_dtype=torch.bfloat16))
            except Exception:
                t_f8 = float("nan")
            # include dynamic act-quant cost in a third variant
            def quant_and_mm():
                amax = a.abs().amax().clamp(min=1e-6)
                s = (448.0 / amax).float()
                aq2 = (a.float() * s).clamp(-448, 448).to(dt)
                return torch._scaled_mm(aq2, wqt, scale_a=1.0 / s, scale_b=sw,
                                        out_dtype=torch.bfloat16)
            try:
                t_f8q = bench(quant_and_mm)
            except Exception:
                t_f8q = float("nan")
            gbps = (N * K) / t_bf / 500.0  # bf16 weight GB/s (2B / 1e3 us->s)
            print(f"{name:8s} K={K:6d} N={N:6d}  bf16 {t_bf:8.1f}us ({gbps:5.0f} GB/s)"
                  f"  fp8 {t_f8:8.1f}us  fp8+quant {t_f8q:8.1f}us")


if __name__ == "__main__":
    main()
