"""Rotary position embeddings with linear position-interpolation scaling.

Reference: megatron/model/positional_embeddings.py:7-51 (complex-multiply RoPE
in fp32; scaling divides positions by rope_scaling_factor). We precompute
cos/sin tables (fp32) once and keep them resident in HBM; the per-layer
application is a fused HIP kernel (ops/csrc/rope.hip) over interleaved pairs.
"""

from __future__ import annotations

from typing import Optional

import torch

from ..ops import functional as ops_f


def precompute_freqs(dim: int, end: int, theta: float = 10000.0,
                     scaling_factor: float = 1.0, device=None):
    """cos/sin tables (end, dim/2) fp32."""
    freqs = 1.0 / (
        theta ** (torch.arange(0, dim, 2, device=device)[: dim // 2].float() / dim)
    )
    t = torch.arange(end, device=device).float() / scaling_factor
    angles = torch.outer(t, freqs)  # (end, dim/2)
    return torch.cos(angles), torch.sin(angles)


def apply_rotary_emb(query: torch.Tensor, key: torch.Tensor,
                     cos: torch.Tensor, sin: torch.Tensor,
                     position_ids: Optional[torch.Tensor] = None):
    """query (s, b, np, hn), key (s, b, nkv, hn). cos/sin tables indexed by
    absolute position; position_ids (b, s) selects rows for inference with
    KV-cache offsets."""
    if position_ids is None:
        s = query.shape[0]
        c, sn = cos[:s], sin[:s]
        q = ops_f.apply_rope(query, c, sn)
        k = ops_f.apply_rope(key, c, sn)
    else:
        # gather per-position tables: (s, b, hn/2) — inference path, small s
        c = cos[position_ids].transpose(0, 1)  # (s, b, dim/2)
        sn = sin[position_ids].transpose(0, 1)
        q = _apply_rope_positional(query, c, sn)
        k = _apply_rope_positional(key, c, sn)
    return q, k


def _apply_rope_positional(x, cos, sin):
    xf = x.float()
    x1 = xf[..., 0::2]
    x2 = xf[..., 1::2]
    c = cos.unsqueeze(2)  # (s, b, 1, hn/2)
    s = sin.unsqueeze(2)
    o1 = x1 * c - x2 * s
    o2 = x2 * c + x1 * s
    return torch.stack([o1, o2], dim=-1).flatten(-2).to(x.dtype)
