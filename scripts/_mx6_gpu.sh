cd /root/repo
set -x
timeout 420 python -m pytest tests/test_gpu_kernels.py -q -k "mx6" 2>&1 | tail -6
timeout 420 python scripts/mx6_ab.py 2>&1 | tail -12
GPU_FUZZ_PRECISION=mx6 GPU_FUZZ_N=60 timeout 420 python scripts/gpu_fuzz.py 2>&1 | tail -4
timeout 420 python bench.py --gpus 1 --steps 6 --warmup 2 --precision mx6 2>&1 | tail -2
timeout 120 ./scripts/mfma_rate 2>&1 | tail -8
