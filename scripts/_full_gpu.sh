cd /root/repo
set -x
timeout 900 python -m pytest tests/ -q -m gpu 2>&1 | tail -4
timeout 300 python -c "import __graft_entry__ as g; g.smoke(); print('SMOKE-OK')" 2>&1 | tail -2
timeout 600 python bench.py --gpus 1 --steps 8 --warmup 3 2>&1 | tail -1
timeout 600 python bench.py --gpus 1 --steps 6 --warmup 2 --precision mx8 2>&1 | tail -1
timeout 600 python bench.py --gpus 1 --steps 6 --warmup 2 --precision mx6 2>&1 | tail -1
timeout 600 python bench.py --gpus 1 --steps 6 --warmup 2 --precision mx4 2>&1 | tail -1
