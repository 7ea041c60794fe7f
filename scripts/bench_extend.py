#!/usr/bin/env python3
"""Extend-attention (paged chunked-prefill) microbench: old 64-row kernel
vs the 8-wave 32x32 ladder kernel at serving shapes. Effective TF counts
only the causally-visible (q,k) pairs — tile-masked waste shows up as a
lower number, same convention as bench_prefill."""

import math
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import arks_amd.ops as ops
from arks_amd.ops import build_prefill_tiles


def bench(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / iters * 1e6


def tiles256(q_lens, kv_lens, device):
    """Work-sorted unsplit 256-row tiles [n,4] (part=0, nparts=1)."""
    t = []
    for i, n in enumerate(q_lens):
        off = kv_lens[i] - n
        for q0 in range(0, n, 256):
            t.append((i, q0, min(kv_lens[i], off + q0 + 256)))
    t.sort(key=lambda x: -x[2])
    return torch.tensor([(x[0], x[1], 0, 1) for x in t], dtype=torch.int32,
                        device=device).reshape(-1, 4)


def main():
    torch.manual_seed(0)
    hq, hkv, hd = 28, 4, 128
    bs = 16
    # (nseq, q_len, kv_len): bench-shaped chunks + long-context chunks
    for nseq, qlen, kvlen in ((8, 1024, 1024), (8, 512, 1024), (4, 2048, 2048),
                              (1, 8192, 8192), (16, 1024, 1024),
                              (1, 256, 8192), (2, 256, 16384)):
        S, Tq = nseq, nseq * qlen
        nb = (kvlen + bs - 1) // bs
        q = torch.randn(Tq, hq, hd, dtype=torch.bfloat16, device="cuda")
        kc = torch.randn(S * nb + 1, hkv, bs, hd, dtype=torch.bfloat16,
                         device="cuda")
        vc = torch.randn_like(kc)
        bt = (torch.arange(S * nb, dtype=torch.int32, device="cuda") + 1
              ).reshape(S, nb)
        cu = torch.arange(0, S + 1, dtype=torch.int32, device="cuda") * qlen
        kvl = torch.full((S,), kvlen, dtype=torch.int32, device="cuda")
        scale = 1.0 / math.sqrt(hd)
        off = kvlen - qlen
        # visible pairs per seq: sum over q rows of (off + row + 1)
        pairs = S * (qlen * off + qlen * (qlen + 1) / 2)
        flops = 2 * 2 * hq * hd * pairs
        out = torch.empty_like(q)
        t64i = build_prefill_tiles([qlen] * S, "cuda")
        t256 = tiles256([qlen] * S, [kvlen] * S, "cuda")
        nat = ops._native()
        t_old = bench(lambda: nat.attention_extend_paged(
            out, q, kc, vc, bt, kvl, cu, t64i, scale, 0))
        ws0 = torch.empty(0, dtype=torch.float32, device="cuda")
        t_new = bench(lambda: nat.attention_extend_paged2(
            out, q, kc, vc, bt, kvl, cu, t256, ws0, scale, 0))
        # split path: prebuilt tiles (as the engine does once per batch)
        tiles = ops.build_extend_tiles([qlen] * S, [kvlen] * S, False,
                                       "cuda", num_q_heads=hq)
        t_split = bench(lambda: ops.attention_extend_paged(
            q, kc, vc, bt, kvl, cu, [qlen] * S, scale, tiles=tiles))
        print(f"S={nseq:3d} q={qlen:5d} kv={kvlen:5d}  "
              f"old {t_old:9.1f}us {flops/t_old/1e6:7.1f}TF  "
              f"new {t_new:9.1f}us {flops/t_new/1e6:7.1f}TF  "
              f"split {t_split:9.1f}us {flops/t_split/1e6:7.1f}TF  "
              f"x{t_old/t_new:.2f}/x{t_old/t_split:.2f}")


if __name__ == "__main__":
    main()
