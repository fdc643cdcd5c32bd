#!/bin/bash
# round-2 GPU call 6: dropout fix validation, mistral mbs1-vs-mbs2+recompute, fp8 tune+bench
set -x
cd /root/repo
mkdir -p gpurun_out
echo "=== dropout tests ==="
python -m pytest tests/test_ops_gpu.py -q -m gpu -k "dropout" 2>&1 | tail -8
echo "=== mistral 32k mbs1 (new kernels, chunked loss) ==="
timeout 1500 python bench.py --gpus 1 --steps 4 --warmup 2 --model mistral-7b 2>&1 | tail -1
echo "=== mistral 32k mbs2 + full recompute ==="
timeout 1500 python bench.py --gpus 1 --steps 4 --warmup 2 --model mistral-7b --micro-batch-size 2 --global-batch 2 --recompute 2>&1 | tail -1
echo "=== fp8 TunableOp sweep (fp8 shapes only) ==="
PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1 \
PYTORCH_TUNABLEOP_FILENAME=profiles/tunableop_results.csv \
PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS=500 \
PYTORCH_TUNABLEOP_MAX_TUNING_ITERATIONS=300 \
TUNE_FP8=1 timeout 1200 python tools/tune_gemms.py 2>&1 | tail -3
grep -c "" profiles/tunableop_results0.csv; grep -i "scaled\|fp8\|Float8" profiles/tunableop_results0.csv | head -12
cp profiles/tunableop_results0.csv gpurun_out/tunableop_full.csv
echo "=== fp8 bench with tuned table ==="
timeout 900 python bench.py --gpus 1 --steps 6 --warmup 3 --dtype fp8 2>&1 | tail -1
