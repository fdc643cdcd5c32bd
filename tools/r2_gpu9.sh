#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
echo "=== full GPU suite ==="
python -m pytest tests/ -q -m gpu 2>&1 | tail -2
echo "=== headline bench steps=20 ==="
timeout 1800 python bench.py --gpus 1 --steps 20 --warmup 5 2>&1 | tail -2
echo "=== per-model numbers ==="
timeout 1200 python bench.py --gpus 1 --steps 4 --warmup 2 --model mistral-7b 2>&1 | tail -1
timeout 1200 python bench.py --gpus 1 --steps 6 --warmup 2 --model falcon-7b 2>&1 | tail -1
echo "=== decode final ==="
timeout 900 python tools/bench_decode.py --model llama2-7b --tokens 128 --prompt 32 2>&1 | tail -3
echo "=== rocprof kernel stats (2 steps) ==="
export TMPDIR=/tmp
cd /tmp
timeout 1200 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/r2prof -o r2 -- python /root/repo/bench.py --gpus 1 --steps 2 --warmup 1 > /root/repo/gpurun_out/r2prof.log 2>&1
echo rocprof rc $?
ls /root/repo/gpurun_out/r2prof/ 2>/dev/null
