import sys, time
sys.path.insert(0, ".")
import torch
import arks_amd.ops as O
native = O._native()
def timeit(fn, iters=300):
    for _ in range(30): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6
torch.manual_seed(0)
for name, K, N in (("qkv", 3584, 4608), ("o", 3584, 3584)):
    x = torch.randn(64, K, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.02
    us0 = timeit(lambda: O.skinny_gemm(x, w))
    nb = w.numel() * 2
    print(f"{name}: dispatch {us0:6.1f} us {nb/us0/1e6:.2f} TB/s")
    for variant in (0, 7):
        for ns in (8, 12, 16):
            kps = -(-(-(-K // ns)) // 32) * 32
            nsp = -(-K // kps)
            out = torch.empty(64, N, dtype=torch.bfloat16, device="cuda")
            part = O._skinny_ws(nsp, N, 64, x.device)
            f = lambda: native.skinny_gemm_v(out, part, x, w, None, kps, nsp, variant, False)
            us = timeit(f)
            print(f"{name}: v{variant} ns={nsp:2d} {us:6.1f} us {nb/us/1e6:.2f} TB/s")
