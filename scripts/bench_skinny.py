#!/usr/bin/env python3
"""Per-shape microbench: skinny_gemm vs hipBLASLt (tuned table loaded) on the
decode GEMM shapes. Prints achieved weight-streaming TB/s."""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import arks_amd.ops as ops

SHAPES = [("qkv", 3584, 4608), ("o", 3584, 3584), ("gate_up", 3584, 37888),
          ("down", 18944, 3584), ("lm_head", 3584, 152064)]


def bench(fn, iters=100):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / iters * 1e6


def main():
    try:
        import torch.cuda.tunable as tunable

        tunable.enable(True)
        tunable.tuning_enable(False)
        tunable.read_file("arks_amd/data/tunableop_gfx950.csv")
    except Exception:
        pass
    torch.manual_seed(0)
    for M in (1, 8, 16, 32, 64):
        print(f"--- M={M}")
        for name, K, N in SHAPES:
            a = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
            w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.05
            t_lib = bench(lambda: torch.nn.functional.linear(a, w))
            t_sk = bench(lambda: ops.skinny_gemm(a, w))
            bytes_w = N * K * 2
            print(f"{name:8s} K={K:6d} N={N:6d} lib {t_lib:7.1f}us "
                  f"({bytes_w/t_lib/1e6:5.2f} TB/s)  skinny {t_sk:7.1f}us "
                  f"({bytes_w/t_sk/1e6:5.2f} TB/s)  x{t_lib/t_sk:4.2f}")


if __name__ == "__main__":
    main()
