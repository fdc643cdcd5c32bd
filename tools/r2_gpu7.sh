#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests/test_ops_gpu.py::TestGemv -x -q -m gpu 2>&1 | tail -2
echo "=== decode with GEMV ==="
timeout 900 python tools/bench_decode.py --model llama2-7b --tokens 64 --prompt 32 2>&1 | tail -4
echo "=== decode rocprof stats ==="
export TMPDIR=/tmp
cd /tmp
timeout 900 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/decprof -o dec -- python /root/repo/tools/bench_decode.py --model llama2-7b --tokens 16 --prompt 16 > /root/repo/gpurun_out/decprof.log 2>&1
tail -3 /root/repo/gpurun_out/decprof.log
ls /root/repo/gpurun_out/decprof/ || true
cd /root/repo
echo "=== training-path sanity (quick bench) ==="
timeout 900 python bench.py --gpus 1 --steps 6 --warmup 3 2>&1 | tail -1
