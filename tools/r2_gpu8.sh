#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
echo "=== full GPU suite ==="
python -m pytest tests/ -q -m gpu 2>&1 | tail -3
echo "=== smoke ==="
python -c "import __graft_entry__ as g; g.smoke(); print('smoke ok')" 2>&1 | tail -2
echo "=== 7B mbs12 probe ==="
timeout 900 python bench.py --gpus 1 --steps 4 --warmup 2 --micro-batch-size 12 --global-batch 12 2>&1 | tail -2
echo "=== 7B mbs8 gbs16 (grad accumulation) ==="
timeout 900 python bench.py --gpus 1 --steps 4 --warmup 2 --global-batch 16 2>&1 | tail -2
