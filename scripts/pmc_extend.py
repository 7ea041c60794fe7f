#!/usr/bin/env python3
"""Run ONLY the ladder extend kernel at a serving shape (for rocprofv3 --pmc)."""

import math
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import arks_amd.ops as ops


def main():
    torch.manual_seed(0)
    hq, hkv, hd = 28, 4, 128
    S, qlen, kvlen = int(os.environ.get("PMC_S", 8)), 1024, 1024
    bs = 16
    Tq = S * qlen
    nb = (kvlen + bs - 1) // bs
    q = torch.randn(Tq, hq, hd, dtype=torch.bfloat16, device="cuda")
    kc = torch.randn(S * nb + 1, hkv, bs, hd, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn_like(kc)
    bt = (torch.arange(S * nb, dtype=torch.int32, device="cuda") + 1).reshape(S, nb)
    cu = torch.arange(0, S + 1, dtype=torch.int32, device="cuda") * qlen
    kvl = torch.full((S,), kvlen, dtype=torch.int32, device="cuda")
    tiles = []
    for i in range(S):
        for q0 in range(0, qlen, 256):
            tiles.append((i, q0, 0, 1))
    t256 = torch.tensor(tiles, dtype=torch.int32, device="cuda").reshape(-1, 4)
    out = torch.empty_like(q)
    ws0 = torch.empty(0, dtype=torch.float32, device="cuda")
    nat = ops._native()
    for _ in range(20):
        nat.attention_extend_paged2(out, q, kc, vc, bt, kvl, cu, t256, ws0,
                                    1.0 / math.sqrt(hd), 0)
    torch.cuda.synchronize()


if __name__ == "__main__":
    main()
