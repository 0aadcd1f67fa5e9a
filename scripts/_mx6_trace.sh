cd /tmp && export TMPDIR=/tmp
timeout 400 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/mx6_bench_trace2 -o tr --output-format csv -- bash -c 'cd /root/repo && timeout 300 python bench.py --gpus 1 --steps 3 --warmup 1 --precision mx6' 2>&1 | tail -2
