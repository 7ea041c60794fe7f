#!/usr/bin/env python3
"""Prefill-attention microbench: effective TFLOP/s at serving shapes."""

import math
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import arks_amd.ops as ops


def bench(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / iters * 1e6


def main():
    torch.manual_seed(0)
    hq, hkv, hd = 28, 4, 128
    for nseq, L in ((16, 512), (4, 2048), (1, 8192), (1, 16384)):
        T = nseq * L
        q = torch.randn(T, hq, hd, dtype=torch.bfloat16, device="cuda")
        k = torch.randn(T, hkv, hd, dtype=torch.bfloat16, device="cuda")
        v = torch.randn_like(k)
        cu = torch.arange(0, nseq + 1, dtype=torch.int32, device="cuda") * L
        lens = [L] * nseq
        scale = 1.0 / math.sqrt(hd)
        t = bench(lambda: ops.attention_prefill_varlen(q, k, v, cu, lens, scale))
        flops = nseq * 2 * 2 * hq * (L * (L + 1) / 2) * hd
        print(f"nseq={nseq:3d} L={L:6d}  {t:9.1f}us  {flops/t/1e6:7.1f} TF/s")


if __name__ == "__main__":
    main()
