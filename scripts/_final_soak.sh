cd /root/repo
set -x
timeout 240 python -m pytest tests/ -q -m gpu > gpurun_out/soak_suite.txt 2>&1; grep -E "passed|failed" gpurun_out/soak_suite.txt
timeout 100 python -c "import __graft_entry__ as g; g.smoke(); print('SMOKE-OK')" 2>&1 | tail -1
GPU_FUZZ_PRECISION=mx6 GPU_FUZZ_N=1088 GPU_FUZZ_SEED=77 timeout 150 python scripts/gpu_fuzz.py 2>&1 | tail -1
GPU_FUZZ_PRECISION=mx4 GPU_FUZZ_N=1088 GPU_FUZZ_SEED=78 timeout 150 python scripts/gpu_fuzz.py 2>&1 | tail -1
GPU_FUZZ_PRECISION=bf16 GPU_FUZZ_N=1344 GPU_FUZZ_BS=320 GPU_FUZZ_SEED=79 timeout 150 python scripts/gpu_fuzz.py 2>&1 | tail -1
