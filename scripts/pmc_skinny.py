#!/usr/bin/env python3
"""Run ONLY the skinny_gemm kernel on the down-proj decode shape (for
rocprofv3 --pmc counter collection). Variant/nsplits via env SK_VARIANT /
SK_NS (defaults: the r2 best LDS config, kc128/pf1 ns=16)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import arks_amd.ops as O


def main():
    torch.manual_seed(0)
    variant = int(os.environ.get("SK_VARIANT", "0"))
    nsplits = int(os.environ.get("SK_NS", "16"))
    M, N, K = 64, 3584, 18944
    a = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.05
    k_per_split = -(-(-(-K // nsplits)) // 32) * 32
    nsp = -(-K // k_per_split)
    out = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
    part = O._skinny_ws(nsp, N, M, a.device)
    native = O._native()
    for _ in range(50):
        native.skinny_gemm_v(out, part, a, w, None, k_per_split, nsp,
                             variant, False)
    torch.cuda.synchronize()


if __name__ == "__main__":
    main()
