"""RMSNorm / LayerNorm modules backed by the CDNA4 HIP kernels.

The reference left RMSNorm as unfused Python (fused_layer_norm.py:125-139) —
on Llama/Mistral it sits on the critical path twice per layer, so here it is a
single fused HBM-bound kernel (ops/csrc/norms.hip). Both norms tag their
parameters `sequence_parallel` when SP is on so the optimizer all-reduces
their grads across the TP group (reference fused_layer_norm.py:98-99,132 and
optimizer.py:257-277).
"""

from __future__ import annotations

import torch
from torch.nn import Parameter

from ..ops import functional as ops_f


class RMSNorm(torch.nn.Module):
    def __init__(self, dim: int, eps: float = 1e-5, sequence_parallel: bool = False,
                 params_dtype: torch.dtype = torch.float32):
        super().__init__()
        self.eps = eps
        self.weight = Parameter(torch.ones(dim, dtype=params_dtype))
        if sequence_parallel:
            self.weight.sequence_parallel = True

    def forward(self, x):
        return ops_f.rmsnorm(x, self.weight, self.eps)

    def forward_res(self, x):
        """(normed, residual_passthrough) — the residual branch's gradient
        is folded into the norm backward in-kernel (no autograd fan-in add)."""
        return ops_f.rmsnorm_res(x, self.weight, self.eps)


class LayerNorm(torch.nn.Module):
    def __init__(self, dim: int, eps: float = 1e-5, sequence_parallel: bool = False,
                 params_dtype: torch.dtype = torch.float32):
        super().__init__()
        self.eps = eps
        self.weight = Parameter(torch.ones(dim, dtype=params_dtype))
        self.bias = Parameter(torch.zeros(dim, dtype=params_dtype))
        if sequence_parallel:
            self.weight.sequence_parallel = True
            self.bias.sequence_parallel = True

    def forward(self, x):
        return ops_f.layernorm(x, self.weight, self.bias, self.eps)

    def forward_res(self, x):
        return ops_f.layernorm_res(x, self.weight, self.bias, self.eps)


def get_norm(cfg, sequence_parallel=None):
    sp = cfg.sequence_parallel if sequence_parallel is None else sequence_parallel
    if cfg.use_rms_norm:
        return RMSNorm(cfg.hidden_size, eps=cfg.layernorm_epsilon,
                       sequence_parallel=sp, params_dtype=cfg.params_dtype)
    return LayerNorm(cfg.hidden_size, eps=cfg.layernorm_epsilon,
                     sequence_parallel=sp, params_dtype=cfg.params_dtype)
