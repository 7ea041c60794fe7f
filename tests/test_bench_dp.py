"""bench.py distributed contract: the driver launches it under torchrun with
one rank per GPU; here we run world_size=2 on CPU (gloo) with the tiny preset
and check the rank-0 JSON line aggregates both replicas."""

import json
import os
import subprocess
import sys

import pytest


@pytest.mark.timeout(600)
def test_bench_dp2_gloo():
    env = dict(os.environ)
    env.setdefault("MASTER_ADDR", "127.0.0.1")
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node=2",
            "--master-addr=127.0.0.1", "--master-port=29531",
            "bench.py", "--gpus", "2", "--steps", "2", "--warmup", "1",
            "--model", "tiny", "--batch", "2", "--input-len", "16",
        ],
        capture_output=True, text=True, timeout=560,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        env=env,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    r = json.loads(line)
    assert r["metric"] == "output_tok_s"
    assert r["config"]["parallelism"] == "dp2"
    assert r["config"]["global_batch"] == 4  # replicas add up
    assert r["scaling"] == "weak"
