#!/usr/bin/env python3
"""Serving load test against the real HTTP stack (output tok/s
+ TTFT percentiles at a fixed request rate, through the OpenAI API rather
than engine.step).

Starts `python -m arks_amd.server` in-process-adjacent (subprocess), fires
`--num-requests` chat completions at `--qps` (Poisson arrivals), streams the
responses, and reports throughput + TTFT/E2E percentiles.

    python scripts/load_test.py --model preset:qwen2.5-7b --qps 16 \
        --num-requests 128 --input-len 512 --output-len 64
"""

from __future__ import annotations

import argparse
import asyncio
import json
import os
import random
import subprocess
import sys
import time

import httpx


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="preset:qwen2.5-7b")
    p.add_argument("--port", type=int, default=18080)
    p.add_argument("--qps", type=float, default=16.0)
    p.add_argument("--num-requests", type=int, default=128)
    p.add_argument("--input-len", type=int, default=512)
    p.add_argument("--output-len", type=int, default=64)
    p.add_argument("--server-args", default="", help="extra server flags")
    p.add_argument("--prompt-style", choices=["random", "repetitive"],
                   default="random")
    p.add_argument("--no-spawn", action="store_true",
                   help="assume a server is already on --port")
    return p.parse_args()


async def one_request(client, args, rid, results):
    # ByteTokenizer-friendly prompt of the requested token length
    if args.prompt_style == "repetitive":
        # periodic text: exercises prompt-lookup speculation
        # (--server-args "--speculative ngram") and the prefix cache
        unit = "".join(random.choice("abcdefgh ") for _ in range(16))
        prompt = (unit * (args.input_len // 16 + 1))[: args.input_len]
    else:
        prompt = "".join(
            random.choice("abcdefgh ") for _ in range(args.input_len))
    t0 = time.time()
    ttft = None
    n_tok = 0
    async with client.stream("POST", "/v1/completions", json={
        "model": "m", "prompt": prompt, "max_tokens": args.output_len,
        "temperature": 0.0, "ignore_eos": True, "stream": True,
        "stream_options": {"include_usage": True},
    }) as r:
        async for line in r.aiter_lines():
            if not line.startswith("data: ") or line == "data: [DONE]":
                continue
            chunk = json.loads(line[6:])
            if chunk.get("choices") and chunk["choices"][0].get("text"):
                if ttft is None:
                    ttft = time.time() - t0
            if chunk.get("usage"):
                n_tok = chunk["usage"]["completion_tokens"]
    results.append({"ttft": ttft, "e2e": time.time() - t0, "tokens": n_tok})


async def run_load(args):
    async with httpx.AsyncClient(
        base_url=f"http://127.0.0.1:{args.port}", timeout=600.0
    ) as client:
        for _ in range(600):
            try:
                if (await client.get("/health")).status_code == 200:
                    break
            except Exception:
                pass
            await asyncio.sleep(1)
        else:
            raise RuntimeError("server never became healthy")

        random.seed(0)
        results: list[dict] = []
        tasks = []
        t_start = time.time()
        for i in range(args.num_requests):
            tasks.append(asyncio.create_task(
                one_request(client, args, i, results)))
            await asyncio.sleep(random.expovariate(args.qps))
        await asyncio.gather(*tasks)
        wall = time.time() - t_start

    ttfts = sorted(r["ttft"] for r in results if r["ttft"] is not None)
    e2es = sorted(r["e2e"] for r in results)
    total_tok = sum(r["tokens"] for r in results)

    def pct(xs, p):
        return xs[min(len(xs) - 1, int(len(xs) * p))] if xs else None

    out = {
        "metric": "serving_output_tok_s",
        "value": round(total_tok / wall, 2),
        "qps": args.qps,
        "num_requests": args.num_requests,
        "input_len": args.input_len,
        "output_len": args.output_len,
        "wall_s": round(wall, 2),
        "ttft_ms": {"p50": round(1000 * pct(ttfts, 0.50), 1),
                    "p90": round(1000 * pct(ttfts, 0.90), 1),
                    "p99": round(1000 * pct(ttfts, 0.99), 1)},
        "e2e_ms": {"p50": round(1000 * pct(e2es, 0.50), 1),
                   "p99": round(1000 * pct(e2es, 0.99), 1)},
    }
    print(json.dumps(out))


def main():
    args = parse_args()
    proc = None
    if not args.no_spawn:
        cmd = [
            sys.executable, "-m", "arks_amd.server",
            "--model", args.model, "--served-model-name", "m",
            "--port", str(args.port), "--max-num-seqs", "256",
        ] + (args.server_args.split() if args.server_args else [])
        proc = subprocess.Popen(
            cmd, cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
            stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL,
        )
    try:
        asyncio.run(run_load(args))
    finally:
        if proc is not None:
            proc.terminate()
            try:
                proc.wait(timeout=20)
            except subprocess.TimeoutExpired:
                proc.kill()


if __name__ == "__main__":
    main()
