#!/usr/bin/env python3
"""Decode-shape GEMMs: skinny split-K kernel vs hipBLASLt on the big
M=64 shapes (gate_up/down) that currently stay on the library."""
import time

import torch
import torch.nn.functional as F
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import arks_amd.ops as ops


def bench(fn, iters=50):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / iters * 1e6


def main():
    torch.manual_seed(0)
    for name, N, K in (("qkv", 4608, 3584), ("o", 3584, 3584),
                       ("gate_up", 37888, 3584), ("down", 3584, 18944),
                       ("moe30b_gu", 3072 * 2, 2048), ("moe30b_dn", 2048, 768 * 128 // 64)):
        for M in (16, 64, 128, 256):
            x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
            w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
            t_lib = bench(lambda: F.linear(x, w))
            try:
                t_sk = bench(lambda: ops.skinny_gemm(x, w))
                o1 = ops.skinny_gemm(x, w)
                o2 = F.linear(x, w)
                ok = torch.allclose(o1.float(), o2.float(), atol=1e-1, rtol=2e-2)
            except Exception as e:
                t_sk, ok = float("nan"), str(e)[:40]
            bw = (N * K * 2) / 1e9
            print(f"{name:10s} M={M:4d} N={N:6d} K={K:6d}  "
                  f"lib {t_lib:7.1f}us ({bw/t_lib*1e6/1e3:4.1f} TB/s)  "
                  f"skinny {t_sk:7.1f}us ({bw/t_sk*1e6/1e3:4.1f} TB/s)  ok={ok}")


if __name__ == "__main__":
    main()
