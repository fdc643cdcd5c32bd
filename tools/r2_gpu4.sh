#!/bin/bash
# round-2 GPU call 4: dropout parity + mb2 kernel A/B + bench
set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests/test_ops_gpu.py -x -q -m gpu 2>&1 | tail -3
echo "=== FA numerics with mb2 kernel ==="
MEGATRON_AMD_FA_FWD_WAVES=82 python -m pytest tests/test_ops_gpu.py::TestFlashAttention -x -q -m gpu 2>&1 | tail -3
echo "=== FA microbench default(12) vs mb2(82) ==="
timeout 300 python tools/bench_kernels.py fa
MEGATRON_AMD_FA_FWD_WAVES=82 timeout 300 python tools/bench_kernels.py fa
echo "=== bench steps=12 ==="
timeout 900 python bench.py --gpus 1 --steps 12 --warmup 4 2>&1 | tail -2
echo "=== bench with mb2 ==="
MEGATRON_AMD_FA_FWD_WAVES=82 timeout 900 python bench.py --gpus 1 --steps 12 --warmup 4 2>&1 | tail -1
