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
    for (m, k, n) in shapes:
        x = torch.randn(m, k, device=dev, dtype=torch.bfloat16)
        w = torch.randn(n, k, device=dev, dtype=torch.bfloat16)
        g = torch.randn(m, n, device=dev, dtype=torch.bfloat16)
        for _ in range(3):
            y = x @ w.t()          # fwd
            dx = g @ w             # dgrad
            dw = g.t() @ x         # wgrad
        torch.cuda.synchronize()
        print(f"tuned {m}x{k}x{n}", flush=True)
    print("done")


if __name__ == "__main__":
    main()
