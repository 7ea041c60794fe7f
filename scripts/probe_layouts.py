#!/usr/bin/env python3
"""Dump tr16 read mapping + verify mfma32 probe on hardware."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from arks_amd.ops import _native

n = _native()
# mfma 32x32x16 layout
a = torch.randn(32, 16, dtype=torch.bfloat16, device="cuda")
b = torch.randn(16, 32, dtype=torch.bfloat16, device="cuda")
d = torch.zeros(32, 32, dtype=torch.float32, device="cuda")
n.mfma_probe32(d, a, b)
ref = a.float() @ b.float()
err = (d - ref).abs().max().item()
print("mfma32 probe max err:", err, "OK" if err < 2e-2 else "LAYOUT WRONG")

# tr16 mapping for a few strides
for stride in (8, 16, 64):
    out = torch.zeros(64 * 4, dtype=torch.int16, device="cuda")
    n.tr16_probe(out, stride)
    o = out.cpu().view(64, 4).tolist()
    print(f"--- stride_bytes={stride}: lane -> LDS u16 indices read")
    for lane in (0, 1, 2, 3, 4, 15, 16, 17, 31, 32, 33, 48, 63):
        print(f"lane {lane:2d}: {o[lane]}")
