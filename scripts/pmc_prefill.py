#!/usr/bin/env python3
"""Run ONLY attn_prefill at a serving shape (for rocprofv3 --pmc)."""

import math
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import arks_amd.ops as ops


def main():
    torch.manual_seed(0)
    hq, hkv, hd = 28, 4, 128
    nseq, L = 4, 2048
    T = nseq * L
    q = torch.randn(T, hq, hd, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(T, hkv, hd, dtype=torch.bfloat16, device="cuda")
    v = torch.randn_like(k)
    cu = torch.arange(0, nseq + 1, dtype=torch.int32, device="cuda") * L
    for _ in range(30):
        ops.attention_prefill_varlen(q, k, v, cu, [L] * nseq, 1.0 / math.sqrt(hd))
    torch.cuda.synchronize()


if __name__ == "__main__":
    main()
