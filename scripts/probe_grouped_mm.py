#!/usr/bin/env python3
"""Probe torch._grouped_mm on ROCm: correctness vs per-segment mm + timing
at Qwen3-30B MoE prefill shapes."""
import time

import torch

def main():
    torch.manual_seed(0)
    E, H, I2 = 128, 2048, 3072
    counts = torch.randint(100, 1400, (E,))
    total = int(counts.sum())
    offs = torch.cumsum(counts, 0).to(torch.int32).cuda()
    x = torch.randn(total, H, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(E, H, I2, dtype=torch.bfloat16, device="cuda")
    try:
        y = torch._grouped_mm(x, w, offs=offs)
    except Exception as e:
        print("grouped_mm FAILED:", e)
        return
    # correctness vs per-segment mm
    y_ref = torch.empty_like(y)
    s = 0
    for e in range(E):
        c = int(counts[e])
        y_ref[s:s+c] = x[s:s+c] @ w[e]
        s += c
    ok = torch.allclose(y.float(), y_ref.float(), atol=2e-1, rtol=2e-2)
    print("correct:", ok, "shape:", tuple(y.shape))
    # timing
    for name, fn in (("grouped_mm", lambda: torch._grouped_mm(x, w, offs=offs)),):
        for _ in range(5): fn()
        torch.cuda.synchronize(); t0 = time.time()
        for _ in range(20): fn()
        torch.cuda.synchronize()
        dt = (time.time()-t0)/20
        fl = 2*total*H*I2
        print(f"{name}: {dt*1e6:.0f} us  {fl/dt/1e12:.0f} TF")
    # padded bmm comparison (current path)
    cap = int(counts.max())
    xp = torch.randn(E, cap, H, dtype=torch.bfloat16, device="cuda")
    for _ in range(5): torch.bmm(xp, w)
    torch.cuda.synchronize(); t0 = time.time()
    for _ in range(20): torch.bmm(xp, w)
    torch.cuda.synchronize()
    dt = (time.time()-t0)/20
    print(f"padded bmm (cap={cap}): {dt*1e6:.0f} us  useful {2*total*H*I2/dt/1e12:.0f} TF")

if __name__ == "__main__":
    main()
