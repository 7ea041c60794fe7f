set -x
cd /root/repo
timeout 420 python -m pytest tests -m gpu -q 2>&1 | tail -3 | tee gpurun_out/pytest_gpu.log
timeout 240 python bench.py --steps 16 --warmup 4 --batch 64 --input-len 512 2>&1 | tail -1 | tee gpurun_out/bench_b64.json
timeout 240 python bench.py --steps 16 --warmup 4 --batch 256 --input-len 512 2>&1 | tail -1 | tee gpurun_out/bench_b256.json
timeout 240 python bench.py --steps 16 --warmup 4 --batch 8 --input-len 512 2>&1 | tail -1 | tee gpurun_out/bench_b8.json
timeout 300 python bench.py --steps 16 --warmup 4 --batch 16 --input-len 8192 2>&1 | tail -1 | tee gpurun_out/bench_b16_8k.json
timeout 300 python bench.py --steps 16 --warmup 4 --batch 64 --input-len 512 --quantization fp8 2>&1 | tail -1 | tee gpurun_out/bench_b64_fp8.json
timeout 300 python bench.py --steps 16 --warmup 4 --batch 256 --input-len 512 --quantization fp8 2>&1 | tail -1 | tee gpurun_out/bench_b256_fp8.json
