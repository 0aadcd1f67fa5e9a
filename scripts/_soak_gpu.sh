cd /root/repo
set -x
timeout 900 python -m pytest tests/ -q -m gpu 2>&1 | tail -2
timeout 300 python -c "import __graft_entry__ as g; g.smoke(); print('SMOKE-OK')" 2>&1 | tail -1
timeout 400 python bench.py --gpus 1 --steps 6 --warmup 2 --precision mx6 2>&1 | tail -1 | head -c 300; echo
GPU_FUZZ_PRECISION=bf16 GPU_FUZZ_N=250 timeout 500 python scripts/gpu_fuzz.py 2>&1 | tail -1
GPU_FUZZ_PRECISION=mx8  GPU_FUZZ_N=150 timeout 400 python scripts/gpu_fuzz.py 2>&1 | tail -1
GPU_FUZZ_PRECISION=mx6  GPU_FUZZ_N=150 timeout 400 python scripts/gpu_fuzz.py 2>&1 | tail -1
GPU_FUZZ_PRECISION=mx4  GPU_FUZZ_N=150 timeout 400 python scripts/gpu_fuzz.py 2>&1 | tail -1
GPU_FUZZ_PRECISION=bf16 GPU_FUZZ_BS=512 GPU_FUZZ_N=120 GPU_FUZZ_SEED=991 timeout 400 python scripts/gpu_fuzz.py 2>&1 | tail -1
MATREL_DETERMINISTIC=1 GPU_FUZZ_PRECISION=bf16 GPU_FUZZ_N=120 GPU_FUZZ_SEED=313 timeout 400 python scripts/gpu_fuzz.py 2>&1 | tail -1
