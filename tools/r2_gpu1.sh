#!/bin/bash
# round-2 GPU call 1: validate FA v8 + hipblasLt wgrad, A/B microbench, quick bench
set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests/test_ops_gpu.py -x -q -m gpu 2>&1 | tail -5
echo "=== FA microbench: v8 (default) ==="
timeout 300 python tools/bench_kernels.py fa
echo "=== FA microbench: v4 (12-wave) ==="
MEGATRON_AMD_FA_FWD_WAVES=12 timeout 300 python tools/bench_kernels.py fa
echo "=== wgrad tune (verbose) ==="
MEGATRON_AMD_WGRAD_VERBOSE=1 timeout 300 python - <<'PY'
import torch, time
from megatron_amd.ops import ext
mod = ext.load(required=True)
K = 32768
for (out_dim, in_dim) in [(12288, 4096), (4096, 4096), (22016, 4096), (4096, 11008), (32000, 4096)]:
    inp = torch.randn(K, in_dim, device="cuda", dtype=torch.bfloat16)
    g = torch.randn(K, out_dim, device="cuda", dtype=torch.bfloat16)
    mg = torch.zeros(out_dim, in_dim, device="cuda", dtype=torch.float32)
    mod.wgrad_gemm_accum_fp32(inp, g, mg)  # tunes
    # correctness
    ref = (g[:256].float().t() @ inp[:256].float())
    mg2 = torch.zeros(out_dim, in_dim, device="cuda", dtype=torch.float32)
    mod.wgrad_gemm_accum_fp32(inp[:256].contiguous(), g[:256].contiguous(), mg2)
    err = (mg2 - ref).abs().max().item() / ref.abs().max().item()
    torch.cuda.synchronize(); t0 = time.time()
    for _ in range(10): mod.wgrad_gemm_accum_fp32(inp, g, mg)
    torch.cuda.synchronize(); dt = (time.time()-t0)/10
    print(f"wgrad {out_dim}x{in_dim}xK{K}: {dt*1e3:.3f} ms  {2*out_dim*in_dim*K/dt/1e12:.0f} TF  relerr {err:.2e}", flush=True)
PY
echo "=== quick bench ==="
timeout 900 python bench.py --gpus 1 --steps 8 --warmup 3 2>&1 | tail -3
