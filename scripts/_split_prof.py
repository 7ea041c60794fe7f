import math, os, sys, time
sys.path.insert(0, "/root/repo")
import torch
import arks_amd.ops as ops

torch.manual_seed(0)
hq, hkv, hd, bs = 28, 4, 128, 16
S, qlen, kvlen = 1, 256, 8192
nb = (kvlen + bs - 1) // bs
q = torch.randn(S * qlen, hq, hd, dtype=torch.bfloat16, device="cuda")
kc = torch.randn(S * nb + 1, hkv, bs, hd, dtype=torch.bfloat16, device="cuda")
vc = torch.randn_like(kc)
bt = (torch.arange(S * nb, dtype=torch.int32, device="cuda") + 1).reshape(S, nb)
cu = torch.arange(0, S + 1, dtype=torch.int32, device="cuda") * qlen
kvl = torch.full((S,), kvlen, dtype=torch.int32, device="cuda")
scale = 1.0 / math.sqrt(hd)
tiles = ops.build_extend_tiles([qlen]*S, [kvlen]*S, False, "cuda", num_q_heads=hq)
print("t256 rows:", tiles[1].tolist()[:10], "ws_rows:", tiles[4])
for _ in range(30):
    ops.attention_extend_paged(q, kc, vc, bt, kvl, cu, [qlen]*S, scale, tiles=tiles)
torch.cuda.synchronize()
