"""The driver's bench contract, exercised for real: `bench.py` launched
under `torch.distributed.run` with world_size 2 (gloo, CPU, tiny preset) —
the exact shape of the driver's round-end SCALE run (which has never had
an 8-GPU node to run on), minus the GPUs. Asserts the single JSON line,
dp token aggregation across ranks, and the contract fields."""

import json
import os
import socket
import subprocess
import sys


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _run_bench(mode: str):
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ, MASTER_ADDR="127.0.0.1")
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", "--nproc-per-node", "2",
        "--master-addr", "127.0.0.1", "--master-port", str(_free_port()),
        os.path.join(repo, "bench.py"),
        "--gpus", "2", "--steps", "3", "--warmup", "1",
        "--batch", "2", "--input-len", "32", "--model", "tiny",
        "--parallel", mode,
    ]
    return subprocess.run(cmd, cwd=repo, env=env, capture_output=True,
                          text=True, timeout=420)


def test_bench_dp2_gloo_contract():
    out = _run_bench("dp")
    assert out.returncode == 0, f"bench failed:\n{out.stdout}\n{out.stderr}"
    lines = [ln for ln in out.stdout.splitlines()
             if ln.startswith("{") and '"metric"' in ln]
    assert len(lines) == 1, f"expected exactly one JSON line:\n{out.stdout}"
    rec = json.loads(lines[0])
    assert rec["metric"] == "output_tok_s"
    assert rec["n_gpus"] == 2
    assert rec["steps"] == 3 and rec["warmup"] == 1
    assert rec["scaling"] == "weak"
    assert rec["higher_is_better"] is True
    assert rec["dtype"] == "bf16" and rec["data"] == "synthetic"
    assert rec["value"] > 0 and rec["ms_per_step"] > 0
    # dp aggregation: whole-job tokens = batch * steps * world / elapsed —
    # value must reflect BOTH ranks (2 * 2 * 3 tokens over the window)
    expected_tokens = 2 * 2 * 3
    approx_tokens = rec["value"] * (rec["ms_per_step"] * 3 / 1000.0)
    assert abs(approx_tokens - expected_tokens) / expected_tokens < 0.05


def test_bench_tp2_gloo_contract():
    """The TP bench path (one engine sharded over both ranks): rank 0
    emits the JSON line with strong scaling and un-multiplied tokens."""
    out = _run_bench("tp")
    assert out.returncode == 0, f"bench failed:\n{out.stdout}\n{out.stderr}"
    lines = [ln for ln in out.stdout.splitlines()
             if ln.startswith("{") and '"metric"' in ln]
    assert len(lines) == 1, f"expected exactly one JSON line:\n{out.stdout}"
    rec = json.loads(lines[0])
    assert rec["n_gpus"] == 2
    assert rec["scaling"] == "strong"
    assert rec["config"]["parallelism"] == "tp2"
    assert rec["config"]["global_batch"] == 2
    assert rec["value"] > 0
