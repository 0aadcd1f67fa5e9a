cd /tmp && export TMPDIR=/tmp
timeout 400 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/mx4_bench_trace -o tr --output-format csv -- bash -c 'cd /root/repo && timeout 300 python bench.py --gpus 1 --steps 3 --warmup 1 --precision mx4' 2>&1 | tail -2
cd /root/repo
MATREL_DETERMINISTIC=1 GPU_FUZZ_PRECISION=bf16 GPU_FUZZ_N=1088 GPU_FUZZ_SEED=91 timeout 150 python scripts/gpu_fuzz.py 2>&1 | tail -1
