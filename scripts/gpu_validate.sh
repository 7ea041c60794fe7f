set -x
cd /root/repo
timeout 420 python -m pytest tests -m gpu -q 2>&1 | tail -6 | tee gpurun_out/pytest_gpu.log
timeout 300 python scripts/bench_skinny.py 2>&1 | tee gpurun_out/bench_skinny.log
timeout 240 python bench.py --steps 16 --warmup 4 --batch 64 --input-len 512 2>&1 | tail -1 | tee gpurun_out/bench_b64.json
timeout 240 python bench.py --steps 16 --warmup 4 --batch 256 --input-len 512 2>&1 | tail -1 | tee gpurun_out/bench_b256.json
timeout 240 python bench.py --steps 16 --warmup 4 --batch 64 --input-len 512 --quantization fp8 2>&1 | tail -1 | tee gpurun_out/bench_b64_fp8.json
