#!/usr/bin/env python3
"""Decode-attention microbench: achieved KV-stream TB/s at serving shapes."""

import os
import sys
import time

import torch

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
    hd = 128
    for hq, hkv, name in ((28, 4, "qwen7b"), (64, 8, "llama70b-tp1-ish")):
        print(f"--- {name}: hq={hq} hkv={hkv} hd={hd}")
        for S, L in ((16, 512), (64, 527), (256, 600), (64, 4096), (8, 16384)):
            bs = 16
            nb = (L + bs - 1) // bs
            total = S * nb + 1
            q = torch.randn(S, hq, hd, dtype=torch.bfloat16, device="cuda")
            kc = torch.randn(total, hkv, bs, hd, dtype=torch.bfloat16,
                             device="cuda")
            vc = torch.randn_like(kc)
            bt = torch.arange(1, total, dtype=torch.int32,
                              device="cuda").reshape(S, nb)
            sl = torch.full((S,), L, dtype=torch.int32, device="cuda")
            nparts = ops.decode_num_partitions(S, hkv, nb)
            t = bench(lambda: ops.attention_decode_paged(
                q, kc, vc, bt, sl, hd ** -0.5))
            kv_bytes = S * L * hkv * hd * 2 * 2
            print(f"S={S:4d} L={L:6d} nparts={nparts:3d}  {t:8.1f}us  "
                  f"{kv_bytes / t / 1e6:5.2f} TB/s")


if __name__ == "__main__":
    main()
