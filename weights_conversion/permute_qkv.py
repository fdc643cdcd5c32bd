"""QKV rotary-layout permutation between HF (half-rotation / rotate_half) and
Megatron/Meta (interleaved complex pairs) conventions.

Behavioral contract matches the reference's weights_conversion/utils/
permute_qkv.py:12-29 so checkpoints convert identically: the fused QKV weight
is laid out per KV group as [q_1..q_nq, k, v] (each head_dim rows); every q
and k head has its rows re-interleaved so that HF's [first-half | second-half]
rotary pairing becomes the interleaved (even, odd) pairing used by
megatron_amd's RoPE kernel. v heads are untouched.
"""

from __future__ import annotations

import torch


def permute_qkv(qkv_w: torch.Tensor, dim: int, n_heads: int,
                n_heads_kv: int, revert: bool = False) -> torch.Tensor:
    head_dim = dim // n_heads
    n_qs_per_kv = n_heads // n_heads_kv
    n_groups = qkv_w.size(0) // head_dim // (n_qs_per_kv + 2)

    def permute(x):
        if revert:
            # interleaved -> HF halves
            return (
                x.view(head_dim // 2, 2, dim)
                .transpose(0, 1)
                .reshape(head_dim, dim)
            )
        # HF halves -> interleaved
        return (
            x.view(2, head_dim // 2, dim)
            .transpose(0, 1)
            .reshape(head_dim, dim)
        )

    groups = torch.chunk(qkv_w, n_groups, dim=0)
    new = []
    for group in groups:
        *qs, k, v = torch.split(group, head_dim, dim=0)
        assert len(qs) == n_qs_per_kv, f"{len(qs)}, {n_qs_per_kv}"
        new += [permute(q) for q in qs] + [permute(k), v]
    return torch.cat(new, dim=0)
