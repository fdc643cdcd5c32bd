"""Microbench: hand wgrad GEMM vs hipBLASLt path on the Llama-7B shapes."""
import os
import sys
import time

sys.path.append(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from megatron_amd.ops import ext

mod = ext.load(required=True)
K = 32768
for (M, N) in [(12288, 4096), (4096, 4096), (22016, 4096), (4096, 11008),
               (32000, 4096)]:
    g = torch.randn(K, M, device="cuda", dtype=torch.bfloat16)
    x = torch.randn(K, N, device="cuda", dtype=torch.bfloat16)
    mg = torch.zeros(M, N, device="cuda", dtype=torch.float32)
    # correctness on a K-slice
    ref = g[:256].float().t() @ x[:256].float()
    mg2 = torch.zeros(M, N, device="cuda", dtype=torch.float32)
    assert mod.wgrad_gemm_hand(x[:256].contiguous(), g[:256].contiguous(), mg2)
    err = (mg2 - ref).abs().max().item() / ref.abs().max().item()

    def timeit(fn, iters=10):
        fn()
        torch.cuda.synchronize()
        t0 = time.time()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        return (time.time() - t0) / iters

    t_hand = timeit(lambda: mod.wgrad_gemm_hand(x, g, mg))
    t_lib = timeit(lambda: mod.wgrad_gemm_accum_fp32(x, g, mg))
    fl = 2.0 * M * N * K
    print(f"{M}x{N}xK{K}: hand {t_hand*1e3:.3f} ms ({fl/t_hand/1e12:.0f} TF) "
          f"lib {t_lib*1e3:.3f} ms ({fl/t_lib/1e12:.0f} TF)  relerr {err:.2e}",
          flush=True)
