"""arks_amd serving entrypoint.

CLI contract matches what the Arks operator composes for its runtime images
(reference arksapplication_controller.go:941-1014): --model,
--served-model-name, --tensor-parallel-size, --port.

    python -m arks_amd.server --port 8080 --model /models/models/ns/name \
        --served-model-name qwen --tensor-parallel-size 8

TP > 1: if not already under torchrun, re-execs itself via
torch.distributed.run with one rank per GPU; rank 0 serves HTTP, other
ranks follow broadcasts (arks_amd/server/async_engine.py).
"""

from __future__ import annotations

import argparse
import os
import sys


def parse_args(argv=None):
    p = argparse.ArgumentParser(prog="arks_amd.server")
    p.add_argument("--model", required=True,
                   help="model directory, or preset:<name> for random-init")
    p.add_argument("--served-model-name", default=None)
    p.add_argument("--tensor-parallel-size", type=int, default=1)
    p.add_argument("--host", default="0.0.0.0")
    p.add_argument("--port", type=int, default=8080)
    p.add_argument("--max-model-len", type=int, default=8192)
    p.add_argument("--max-num-seqs", type=int, default=256)
    p.add_argument("--gpu-memory-utilization", type=float, default=0.90)
    p.add_argument("--kv-cache-blocks", type=int, default=None)
    p.add_argument("--enforce-eager", action="store_true")
    p.add_argument("--disaggregation-mode", choices=["prefill", "decode"],
                   default=None)
    p.add_argument("--quantization", choices=["fp8"], default=None)
    p.add_argument("--kv-cache-dtype", choices=["auto", "fp8"], default="auto")
    p.add_argument("--speculative", choices=["ngram", "draft"], default=None,
                   help="speculative decoding: ngram (prompt lookup) or "
                        "draft (a smaller draft model, --draft-model)")
    p.add_argument("--draft-model", default=None,
                   help="draft checkpoint path (or preset:<name>); must "
                        "share the target tokenizer")
    p.add_argument("--num-speculative-tokens", type=int, default=4)
    p.add_argument("--no-prefix-cache", action="store_true",
                   help="disable the content-addressed prefix/radix cache")
    # multi-node group flags (LWS leader/worker topology): workers join the
    # leader's torch.distributed rendezvous
    p.add_argument("--leader-address", default=None)
    p.add_argument("--node-rank", type=int, default=0)
    return p.parse_args(argv)


def build_engine_config(args):
    from arks_amd.config import EngineConfig

    preset = None
    model_path = None
    if args.model.startswith("preset:"):
        preset = args.model.split(":", 1)[1]
    else:
        model_path = args.model
    return EngineConfig(
        model_path=model_path,
        preset=preset,
        served_model_name=args.served_model_name or args.model,
        tensor_parallel_size=args.tensor_parallel_size,
        max_model_len=args.max_model_len,
        max_num_seqs=args.max_num_seqs,
        gpu_memory_utilization=args.gpu_memory_utilization,
        kv_cache_blocks=args.kv_cache_blocks,
        enforce_eager=args.enforce_eager,
        quantization=args.quantization,
        kv_cache_dtype=args.kv_cache_dtype,
        speculative=args.speculative,
        draft_model=args.draft_model,
        num_speculative_tokens=args.num_speculative_tokens,
        enable_prefix_caching=not args.no_prefix_cache,
    )


def main(argv=None):
    args = parse_args(argv)
    tp = args.tensor_parallel_size
    if tp > 1 and "RANK" not in os.environ:
        # LWS leader/worker topology: workers join the leader's rendezvous
        # (reference arksapplication_controller.go:982-1014 wires
        # LWS_LEADER_ADDRESS / LWS_GROUP_SIZE / LWS_WORKER_INDEX; our
        # command builders pass them as --leader-address / --node-rank).
        nnodes = int(os.environ.get("LWS_GROUP_SIZE", "1"))
        master = args.leader_address or "127.0.0.1"
        if tp % nnodes != 0:
            raise SystemExit(
                f"tensor-parallel-size {tp} must divide across "
                f"LWS_GROUP_SIZE {nnodes} nodes"
            )
        os.execv(
            sys.executable,
            [
                sys.executable, "-m", "torch.distributed.run",
                f"--nnodes={nnodes}", f"--nproc-per-node={tp // nnodes}",
                f"--node-rank={args.node_rank}",
                f"--master-addr={master}", "--master-port=29517",
                "-m", "arks_amd.server",
            ] + sys.argv[1:],
        )

    from arks_amd.parallel import comm as tp_comm

    if tp > 1:
        tp_comm.init_tp()

    cfg = build_engine_config(args)
    rank = int(os.environ.get("RANK", 0))
    if tp > 1 and rank != 0:
        from arks_amd.server.async_engine import worker_loop

        worker_loop(cfg)
        return

    import uvicorn

    from arks_amd.server.api import create_app
    from arks_amd.server.async_engine import AsyncEngine
    from arks_amd.server.tokenizer import load_tokenizer

    served = args.served_model_name or os.path.basename(args.model.rstrip("/"))
    engine = AsyncEngine(cfg, model_name=served,
                         disagg_mode=args.disaggregation_mode)
    tok = load_tokenizer(
        cfg.model_path, engine.model_cfg.vocab_size, engine.model_cfg.eos_token_id
    )
    app = create_app(engine, served, tok, disagg_mode=args.disaggregation_mode)
    uvicorn.run(app, host=args.host, port=args.port, log_level="info")


if __name__ == "__main__":
    main()
