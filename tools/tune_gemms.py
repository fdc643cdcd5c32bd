"""Offline hipBLASLt TunableOp tuning for the flagship GEMM shapes.

Run on a GPU box; writes profiles/tunableop_results00.csv which bench.py
then loads in read-only mode.

  PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1 \
  PYTORCH_TUNABLEOP_FILENAME=profiles/tunableop_results.csv \
  python tools/tune_gemms.py
"""

import os
import sys

sys.path.append(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    torch.manual_seed(0)
    dev = "cuda"
    M = 8 * 4096  # mbs 8 x seq 4096
    H, FFN, QKV = 4096, 11008, 3 * 4096
    V = 32000
    shapes = [
        # (fwd: X[M,K] @ W[K,N]) and the matching dgrad/wgrad orientations
        (M, H, QKV), (M, H, H), (M, H, 2 * FFN), (M, FFN, H), (M, H, V),
    ]
    tune_fp8 = os.environ.get("TUNE_FP8", "1") == "1" and hasattr(
        torch, "_scaled_mm"
    )
    for (m, k, n) in shapes:
        x = torch.randn(m, k, device=dev, dtype=torch.bfloat16)
        w = torch.randn(n, k, device=dev, dtype=torch.bfloat16)
        g = torch.randn(m, n, device=dev, dtype=torch.bfloat16)
        for _ in range(3):
            y = x @ w.t()          # fwd
            dx = g @ w             # dgrad
            dw = g.t() @ x         # wgrad
        torch.cuda.synchronize()
        print(f"tuned bf16 {m}x{k}x{n}", flush=True)

        if tune_fp8:
            # fp8 recipe GEMMs (megatron_amd/fp8.py): fwd X8@W8^T and
            # dgrad dY8@(W^T)8^T — TunableOp also covers ScaledGemm
            one = torch.ones(1, device=dev)
            x8 = x.to(torch.float8_e4m3fn)
            w8 = w.to(torch.float8_e4m3fn)
            wt8 = w.t().contiguous().to(torch.float8_e4m3fn)
            g8 = g.to(torch.float8_e4m3fn)
            # wgrad orientation (fp8_wgrad): dy^T [N,M] @ x[M,K]-col, f32 out
            gt8 = g.t().contiguous().to(torch.float8_e4m3fn)
            xt8_w = x.t().contiguous().to(torch.float8_e4m3fn)
            for _ in range(3):
                torch._scaled_mm(x8, w8.t(), scale_a=one, scale_b=one,
                                 out_dtype=torch.bfloat16)
                torch._scaled_mm(g8, wt8.t(), scale_a=one, scale_b=one,
                                 out_dtype=torch.bfloat16)
                torch._scaled_mm(gt8, xt8_w.t(), scale_a=one, scale_b=one,
                                 out_dtype=torch.float32)
            torch.cuda.synchronize()
            print(f"tuned fp8 {m}x{k}x{n}", flush=True)
    print("done")


if __name__ == "__main__":
    main()
