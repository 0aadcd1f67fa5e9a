cd /root/repo
set -x
for ex in ols_regression pagerank relational_analytics observability_tour checkpointed_power_iteration triangle_counting; do
  timeout 300 python examples/$ex.py > gpurun_out/ex_$ex.log 2>&1; echo "$ex rc=$?"
done
# near-capacity regression (caught the refcycle OOM in round 2): 163840^2
# bf16 at block 16384 - multi-block orchestration at ~268 GB footprint
timeout 900 python bench.py --gpus 1 --steps 3 --warmup 1 --n-size 163840 --block-size 16384 2>&1 | tail -1 | head -c 220; echo
# capacity at mx6 (quantized multi-block path at scale)
timeout 900 python bench.py --gpus 1 --steps 3 --warmup 1 --n-size 131072 --block-size 16384 --precision mx6 2>&1 | tail -1 | head -c 220; echo
timeout 600 python scripts/gpu_stress.py 2>&1 | tail -3
