#!/usr/bin/env python3
"""Microbench the decode down-proj GEMM (M=64, N=3584, K=18944):
hipBLASLt (with the engine's TunableOp table) vs the skinny split-K kernel
at several nsplits. The decode profile (profiles/data, r2) shows this GEMM
at 46.4 us in-graph = 2.9 TB/s vs the 21.6 us HBM floor."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
os.environ.setdefault(
    "PYTORCH_TUNABLEOP_FILENAME",
    os.path.join(os.path.dirname(__file__), "..", "arks_amd", "data",
                 "tunableop_gfx950.csv"),
)

import torch  # noqa: E402
import torch.nn.functional as F  # noqa: E402

from arks_amd import ops  # noqa: E402


def timeit(fn, iters=200):
    for _ in range(20):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    torch.manual_seed(0)
    dev = "cuda"
    M, N, K = 64, 3584, 18944
    x = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
    w = torch.randn(N, K, dtype=torch.bfloat16, device=dev) * 0.02
    nbytes = w.numel() * 2

    ref = F.linear(x, w)
    us = timeit(lambda: F.linear(x, w))
    print(f"hipBLASLt            : {us:7.1f} us  {nbytes / us / 1e6:5.2f} TB/s")

    # force the skinny kernel regardless of the dispatch threshold
    import arks_amd.ops as O
    native = O._native()
    vnames = {0: "kc128/pf1", 1: "kc128/pf2", 2: "kc128/pf3",
              3: "kc256/pf1", 4: "kc256/pf2", 5: "direct   ", 6: "wavepriv ", 7: "kc128/ab3"}
    for variant in (0, 6, 7):
        for nsplits in (8, 16, 24, 32):
            k_per_split = -(-(-(-K // nsplits)) // 32) * 32
            nsp = -(-K // k_per_split)
            out = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
            part = (O._skinny_ws(nsp, N, M, x.device) if nsp > 1
                    else x.new_empty(0, dtype=torch.float32))

            def run():
                native.skinny_gemm_v(out, part, x, w, None, k_per_split,
                                     nsp, variant, False)

            run()
            err = (out.float() - ref.float()).abs().max().item()
            us = timeit(run)
            print(f"skinny {vnames[variant]} ns={nsp:3d}: {us:7.1f} us  "
                  f"{nbytes / us / 1e6:5.2f} TB/s  maxerr={err:.3e}")

    # fused silu_mul + down-proj vs the two-kernel sequence
    gu = torch.randn(M, 2 * K, dtype=torch.bfloat16, device=dev)
    act = torch.empty(M, K, dtype=torch.bfloat16, device=dev)
    native.silu_mul(act, gu)
    fref = F.linear(act, w)

    def two_kernel():
        native.silu_mul(act, gu)
        return F.linear(act, w)

    us = timeit(two_kernel)
    print(f"silu_mul + hipBLASLt : {us:7.1f} us  (sequence)")
    for variant in (0, 6, 7):
        for nsplits in (8, 16, 24, 32):
            k_per_split = -(-(-(-K // nsplits)) // 32) * 32
            nsp = -(-K // k_per_split)
            out = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
            part = O._skinny_ws(nsp, N, M, x.device)

            def runf():
                native.skinny_gemm_v(out, part, gu, w, None, k_per_split,
                                     nsp, variant, True)

            runf()
            err = (out.float() - fref.float()).abs().max().item()
            us = timeit(runf)
            print(f"fused  {vnames[variant]} ns={nsp:3d}: {us:7.1f} us  "
                  f"{nbytes / us / 1e6:5.2f} TB/s  maxerr={err:.3e}")


if __name__ == "__main__":
    main()
