#!/bin/bash
# round-2 GPU call 5: dropout debug, mistral-32k mbs2, fp8 check, decode baseline
set -x
cd /root/repo
mkdir -p gpurun_out
echo "=== dropout tests (full output) ==="
python -m pytest tests/test_ops_gpu.py -q -m gpu -k "dropout" 2>&1 | tail -40
echo "=== mistral 32k mbs2 with chunked loss ==="
timeout 1200 python bench.py --gpus 1 --steps 4 --warmup 2 --model mistral-7b 2>&1 | tail -2
echo "=== fp8 quick check ==="
timeout 900 python bench.py --gpus 1 --steps 6 --warmup 3 --dtype fp8 2>&1 | tail -1
echo "=== decode baseline ==="
timeout 900 python tools/bench_decode.py --model llama2-7b --tokens 64 --prompt 32 2>&1 | tail -6
