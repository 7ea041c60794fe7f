#!/usr/bin/env python3
"""Run ONLY the skinny_gemm kernel on the two worst shapes (for rocprofv3
--pmc counter collection)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import arks_amd.ops as ops


def main():
    torch.manual_seed(0)
    for K, N in ((18944, 3584), (3584, 37888)):
        a = torch.randn(64, K, dtype=torch.bfloat16, device="cuda")
        w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.05
        for _ in range(30):
            ops.skinny_gemm(a, w)
        torch.cuda.synchronize()


if __name__ == "__main__":
    main()
