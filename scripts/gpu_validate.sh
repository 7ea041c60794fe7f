set -x
cd /root/repo
python -c "import arks_amd.ops as o; print('native', o.native_available())" 2>&1 | tail -1
timeout 420 python -m pytest tests -m gpu -q 2>&1 | tail -6 | tee gpurun_out/pytest_gpu.log
timeout 240 python bench.py --steps 16 --warmup 4 --batch 64 --input-len 512 2>&1 | tail -1 | tee gpurun_out/bench_b64.json
timeout 240 python bench.py --steps 16 --warmup 4 --batch 256 --input-len 512 2>&1 | tail -1 | tee gpurun_out/bench_b256.json
timeout 240 python bench.py --steps 16 --warmup 4 --batch 128 --input-len 1024 2>&1 | tail -1 | tee gpurun_out/bench_b128_1k.json
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof -o r2 -- python /root/repo/bench.py --steps 8 --warmup 2 --batch 64 --input-len 512 > /root/repo/gpurun_out/rocprof_bench.log 2>&1
tail -2 /root/repo/gpurun_out/rocprof_bench.log
ls /root/repo/gpurun_out/prof 2>/dev/null | head -3
