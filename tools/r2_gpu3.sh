#!/bin/bash
# round-2 GPU call 3: bank-conflict fix validation + PMC recheck + TunableOp retune
set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests/test_ops_gpu.py -x -q -m gpu 2>&1 | tail -3
echo "=== FA microbench (conflict-free layout) ==="
timeout 300 python tools/bench_kernels.py fa
echo "=== PMC recheck: conflicts + waits ==="
export TMPDIR=/tmp
cd /tmp
rocprofv3 --pmc SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE SQ_WAIT_ANY SQ_WAVE_CYCLES -d /root/repo/gpurun_out/pmc3 -o fa3 -- python /root/repo/tools/fa_only.py 2>&1 | tail -1
cd /root/repo
echo "=== TunableOp retune of Default rows ==="
grep -v ",Default," profiles/tunableop_results0.csv > profiles/tunableop_results_retune.csv || true
PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1 \
PYTORCH_TUNABLEOP_FILENAME=profiles/tunableop_results_retune.csv \
PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS=1000 \
PYTORCH_TUNABLEOP_MAX_TUNING_ITERATIONS=500 \
TUNE_FP8=0 timeout 1200 python tools/tune_gemms.py 2>&1 | tail -3
grep -c "" profiles/tunableop_results_retune0.csv 2>/dev/null; grep "22016\|11008" profiles/tunableop_results_retune0.csv 2>/dev/null
cp profiles/tunableop_results_retune0.csv gpurun_out/tunableop_retuned.csv 2>/dev/null || true
echo "=== bench (with retuned table if better) ==="
timeout 900 python bench.py --gpus 1 --steps 12 --warmup 4 2>&1 | tail -2
