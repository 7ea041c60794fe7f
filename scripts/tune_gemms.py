#!/usr/bin/env python3
"""Offline GEMM algorithm tuning for the serving hot shapes (TunableOp).

Runs every (M = decode capture sizes + prefill chunk sizes) x (the model's
five GEMM shapes) through torch.linear with TunableOp tuning enabled and
writes the merged result table, which ships as
arks_amd/data/tunableop_gfx950.csv and is loaded read-only at engine start.

Usage (on a GPU box):
    python scripts/tune_gemms.py --model qwen2.5-7b --out arks_amd/data/tunableop_gfx950.csv
"""

from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from arks_amd.config import PRESET_CONFIGS

DECODE_MS = [1, 2, 4, 8, 16, 24, 32, 48, 64, 96, 128, 192, 256]
PREFILL_MS = [512, 1024, 2048, 4096, 8192, 16384]


def shapes_for(cfg, tp: int = 1):
    H = cfg.hidden_size
    I = cfg.intermediate_size
    hd = cfg.head_dim
    nq = cfg.num_attention_heads // tp
    nkv = cfg.num_key_value_heads // tp
    return [
        ("qkv", H, (nq + 2 * nkv) * hd, cfg.attention_bias),
        ("o", nq * hd, H, False),
        ("gate_up", H, 2 * I // tp, False),
        ("down", I // tp, H, False),
        ("lm_head", H, cfg.vocab_size, False),
    ]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="qwen2.5-7b")
    ap.add_argument("--tp", type=int, default=1)
    ap.add_argument("--fp8", action="store_true",
                    help="also tune rowwise-scaled fp8 _scaled_mm shapes")
    ap.add_argument("--moe", action="store_true",
                    help="also tune the MoE strided-batch bmm shapes "
                         "(NN layout, pre-transposed expert weights)")
    ap.add_argument("--out", default="arks_amd/data/tunableop_gfx950.csv")
    args = ap.parse_args()

    assert torch.cuda.is_available()
    import torch.cuda.tunable as tunable

    tunable.enable(True)
    tunable.tuning_enable(True)
    tunable.set_filename(args.out)  # flushed via atexit / set_filename
    if os.path.exists(args.out):
        tunable.read_file(args.out)  # extend the existing table

    cfg = PRESET_CONFIGS[args.model]
    dev = torch.device("cuda")
    for name, in_f, out_f, bias in shapes_for(cfg, args.tp):
        w = torch.randn(out_f, in_f, dtype=torch.bfloat16, device=dev) * 0.01
        b = (
            torch.randn(out_f, dtype=torch.bfloat16, device=dev) * 0.01
            if bias
            else None
        )
        ms = DECODE_MS + (PREFILL_MS if name != "lm_head" else [])
        for m in ms:
            x = torch.randn(m, in_f, dtype=torch.bfloat16, device=dev) * 0.01
            F.linear(x, w, b)  # tuning happens on first call per shape
            torch.cuda.synchronize()
            print(f"tuned {name} M={m}", flush=True)
        if args.fp8:
            w8 = (w.float() * 10).clamp(-448, 448).to(torch.float8_e4m3fn)
            sb = torch.full((1, out_f), 0.1, device=dev)
            for m in ms:
                x8 = torch.randn(m, in_f, device=dev).clamp(-4, 4).to(
                    torch.float8_e4m3fn)
                sa = torch.full((m, 1), 0.1, device=dev)
                torch._scaled_mm(x8, w8.t(), scale_a=sa, scale_b=sb,
                                 bias=b, out_dtype=torch.bfloat16)
                torch.cuda.synchronize()
                print(f"tuned fp8 {name} M={m}", flush=True)
        del w, b
        torch.cuda.empty_cache()

    if args.moe and cfg.num_local_experts:
        # grouped/dense expert bmms: [E, rows, H] @ [E, H, 2I] and
        # [E, rows, I] @ [E, I, H] (NN, contiguous — the engine layout).
        E = cfg.num_local_experts // args.tp
        I = cfg.moe_intermediate_size or cfg.intermediate_size
        H = cfg.hidden_size
        w13 = torch.randn(E, H, 2 * I, dtype=torch.bfloat16, device=dev) * 0.01
        w2 = torch.randn(E, I, H, dtype=torch.bfloat16, device=dev) * 0.01
        # decode dense path rows (= batch <= 64) and typical prefill caps
        for rows in [1, 2, 4, 8, 16, 32, 64, 128, 256, 512, 768, 1024]:
            x = torch.randn(E, rows, H, dtype=torch.bfloat16, device=dev) * 0.01
            h = torch.bmm(x, w13)[..., :I].contiguous()
            torch.bmm(h, w2)
            torch.cuda.synchronize()
            print(f"tuned moe bmm rows={rows}", flush=True)
        del w13, w2
        torch.cuda.empty_cache()

    # results are flushed to the filename at interpreter exit
    print(f"tuning done; results flush to {args.out} at exit")


if __name__ == "__main__":
    main()
