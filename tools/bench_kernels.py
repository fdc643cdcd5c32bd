"""Microbenchmarks for the HIP kernels (run on a GPU box).

  python tools/bench_kernels.py [fa|norms|all]

Reports per-kernel time and effective TFLOP/s / TB/s against roofline.
"""

from __future__ import annotations

import math
import os
import sys
import time

import torch

sys.path.append(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def _time(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / iters


def bench_fa():
    from megatron_amd.ops import ext

    mod = ext.load(required=True)
    B, S, H, HKV, D = 4, 4096, 32, 32, 128
    scale = 1.0 / math.sqrt(D)
    q = torch.randn(B, S, H, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, S, HKV, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, S, HKV, D, device="cuda", dtype=torch.bfloat16)

    t_fwd = _time(lambda: mod.flash_attn_fwd(q, k, v, True, scale, -1))
    out, lse = mod.flash_attn_fwd(q, k, v, True, scale, -1)
    dout = torch.randn_like(out)
    t_bwd = _time(
        lambda: mod.flash_attn_bwd(dout, q, k, v, out, lse, True, scale, -1)
    )

    # causal FLOPs: fwd 2 matmuls, bwd 5 matmuls, each 2*S^2*D*H*B*0.5
    mm = 2 * S * S * D * H * B * 0.5
    fwd_fl = 2 * mm
    bwd_fl = 5 * mm
    print(f"FA fwd  B{B} S{S} H{H} D{D}: {t_fwd * 1e3:.2f} ms  "
          f"{fwd_fl / t_fwd / 1e12:.0f} TF")
    print(f"FA bwd  B{B} S{S} H{H} D{D}: {t_bwd * 1e3:.2f} ms  "
          f"{bwd_fl / t_bwd / 1e12:.0f} TF")

    # GQA variant (llama 70B shape)
    HKV2 = 8
    k2 = torch.randn(B, S, HKV2, D, device="cuda", dtype=torch.bfloat16)
    v2 = torch.randn(B, S, HKV2, D, device="cuda", dtype=torch.bfloat16)
    t_fwd2 = _time(lambda: mod.flash_attn_fwd(q, k2, v2, True, scale, -1))
    print(f"FA fwd GQA8: {t_fwd2 * 1e3:.2f} ms  "
          f"{fwd_fl / t_fwd2 / 1e12:.0f} TF")


def bench_norms():
    from megatron_amd.ops import ext

    mod = ext.load(required=True)
    rows, H = 4 * 4096, 4096
    x = torch.randn(rows, H, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(H, device="cuda", dtype=torch.bfloat16)
    t = _time(lambda: mod.rmsnorm_fwd(x, w, 1e-5))
    gb = rows * H * 2 * 2 / 1e9
    print(f"rmsnorm fwd {rows}x{H}: {t * 1e6:.0f} us  {gb / t / 1e3:.2f} TB/s")

    y, inv = mod.rmsnorm_fwd(x, w, 1e-5)
    dy = torch.randn_like(x)
    t = _time(lambda: mod.rmsnorm_bwd(dy, x, w, inv))
    gb = rows * H * 2 * 3 / 1e9
    print(f"rmsnorm bwd: {t * 1e6:.0f} us  {gb / t / 1e3:.2f} TB/s")


def bench_glu():
    from megatron_amd.ops import ext

    mod = ext.load(required=True)
    rows, F = 4 * 4096, 11008
    x = torch.randn(rows, 2 * F, device="cuda", dtype=torch.bfloat16)
    t = _time(lambda: mod.glu_fwd(x, 3))
    gb = rows * 3 * F * 2 / 1e9
    print(f"swiglu fwd {rows}x2x{F}: {t * 1e6:.0f} us  {gb / t / 1e3:.2f} TB/s")


def bench_softmax():
    from megatron_amd.ops import ext

    mod = ext.load(required=True)
    b, np_, sq = 4, 32, 2048
    x = torch.randn(b, np_, sq, sq, device="cuda", dtype=torch.bfloat16)
    t = _time(lambda: mod.scaled_masked_softmax_fwd(x, None, 1.0, True))
    gb = b * np_ * sq * sq * 2 * 2 / 1e9
    print(f"softmax causal {b}x{np_}x{sq}^2: {t * 1e3:.2f} ms  "
          f"{gb / t / 1e3:.2f} TB/s")


if __name__ == "__main__":
    which = sys.argv[1] if len(sys.argv) > 1 else "all"
    torch.manual_seed(0)
    if which in ("fa", "all"):
        bench_fa()
    if which in ("norms", "all"):
        bench_norms()
        bench_glu()
    if which in ("softmax", "all"):
        bench_softmax()
