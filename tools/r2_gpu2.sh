#!/bin/bash
# round-2 GPU call 2: measure (PMC on FA fwd), deepen wgrad algo search,
# finish TunableOp sweep, A/B wave-skip
set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests/test_ops_gpu.py -x -q -m gpu 2>&1 | tail -3
echo "=== FA microbench v12+skip (default) ==="
timeout 300 python tools/bench_kernels.py fa
echo "=== wgrad: deeper algo search ==="
MEGATRON_AMD_WGRAD_VERBOSE=1 MEGATRON_AMD_WGRAD_ALGOS=100 timeout 600 python - <<'PY'
import torch, time
from megatron_amd.ops import ext
mod = ext.load(required=True)
K = 32768
for (out_dim, in_dim) in [(12288, 4096), (4096, 4096), (22016, 4096), (4096, 11008), (32000, 4096)]:
    inp = torch.randn(K, in_dim, device="cuda", dtype=torch.bfloat16)
    g = torch.randn(K, out_dim, device="cuda", dtype=torch.bfloat16)
    mg = torch.zeros(out_dim, in_dim, device="cuda", dtype=torch.float32)
    mod.wgrad_gemm_accum_fp32(inp, g, mg)
    torch.cuda.synchronize(); t0 = time.time()
    for _ in range(10): mod.wgrad_gemm_accum_fp32(inp, g, mg)
    torch.cuda.synchronize(); dt = (time.time()-t0)/10
    print(f"wgrad {out_dim}x{in_dim}: {dt*1e3:.3f} ms {2*out_dim*in_dim*K/dt/1e12:.0f} TF", flush=True)
PY
echo "=== TunableOp full sweep (bounded) ==="
cp profiles/tunableop_results0.csv /tmp/old_tune.csv
PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1 \
PYTORCH_TUNABLEOP_FILENAME=profiles/tunableop_results.csv \
PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS=300 \
PYTORCH_TUNABLEOP_MAX_TUNING_ITERATIONS=200 \
TUNE_FP8=0 timeout 900 python tools/tune_gemms.py 2>&1 | tail -8
cat profiles/tunableop_results0.csv
echo "=== PMC on FA fwd (pass 1: wait buckets) ==="
cat > /tmp/fa_only.py <<'PY2'
import torch, math, sys, os
sys.path.insert(0, "/root/repo")
from megatron_amd.ops import ext
mod = ext.load(required=True)
B, S, H, D = 4, 4096, 32, 128
q = torch.randn(B, S, H, D, device="cuda", dtype=torch.bfloat16)
k = torch.randn(B, S, H, D, device="cuda", dtype=torch.bfloat16)
v = torch.randn(B, S, H, D, device="cuda", dtype=torch.bfloat16)
for _ in range(5):
    mod.flash_attn_fwd(q, k, v, True, 1.0/math.sqrt(D), -1)
torch.cuda.synchronize()
PY2
export TMPDIR=/tmp
cd /tmp
rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY -d /root/repo/gpurun_out/pmc1 -o fa1 -- python /tmp/fa_only.py 2>&1 | tail -2
rocprofv3 --pmc SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE SQ_VALU_MFMA_BUSY_CYCLES SQ_WAIT_INST_LDS -d /root/repo/gpurun_out/pmc2 -o fa2 -- python /tmp/fa_only.py 2>&1 | tail -2
cd /root/repo
python - <<'PY3'
import glob, csv, collections
for d in ["gpurun_out/pmc1", "gpurun_out/pmc2"]:
    for f in glob.glob(d + "/**/*.csv", recursive=True):
        agg = collections.defaultdict(float)
        with open(f) as fh:
            for row in csv.DictReader(fh):
                kn = row.get("Kernel_Name", "")[:40]
                cn = row.get("Counter_Name"); cv = row.get("Counter_Value")
                if cn and "fa_fwd" in kn:
                    agg[cn] += float(cv)
        if agg:
            print(f)
            for k, v in sorted(agg.items()):
                print(f"  {k}: {v:.3e}")
PY3
echo "=== bench steps=12 ==="
timeout 900 python bench.py --gpus 1 --steps 12 --warmup 4 2>&1 | tail -2
