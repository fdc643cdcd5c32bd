"""Gradient clipping by global L2 norm + zero counting.

Reference: megatron/optimizer/clip_grads.py:16-136 (apex multi_tensor_l2norm /
multi_tensor_scale). Here the norm and scale run as torch._foreach_* fused
HIP kernels over the grad list — one kernel per dtype group — with the
model-parallel all-reduce on top.
"""

from __future__ import annotations

import torch

from ..models.module import param_is_not_shared
from ..utils import param_is_not_tensor_parallel_duplicate


def clip_grad_norm_fp32(parameters, grads_for_norm, max_norm, norm_type=2,
                        model_parallel_group=None, flat_buffers=None,
                        defer_scale=False):
    """flat_buffers: optional list of contiguous fp32 grad buffers that tile
    exactly the same elements as grads_for_norm (valid when no param is
    excluded as a TP duplicate or PP-shared copy, i.e. TP=PP=1, and padding
    regions are zero). The L2 norm and the clip scale then run as ONE flat
    kernel per buffer instead of a multi-tensor chain over every param."""
    if isinstance(parameters, torch.Tensor):
        parameters = [parameters]
    if isinstance(grads_for_norm, torch.Tensor):
        grads_for_norm = [grads_for_norm]

    if flat_buffers and norm_type == 2:
        total_norm = sum(b.pow(2).sum() for b in flat_buffers)
        torch.distributed.all_reduce(
            total_norm, op=torch.distributed.ReduceOp.SUM,
            group=model_parallel_group,
        )
        total_norm = total_norm.item() ** 0.5
        clip_coeff = min(1.0, float(max_norm) / (total_norm + 1.0e-6))
        if defer_scale:
            # caller folds the coefficient into the optimizer kernel
            return total_norm, clip_coeff
        if clip_coeff < 1.0:
            for b in flat_buffers:
                b.mul_(clip_coeff)
        return total_norm

    grads = []
    for param in parameters:
        if param.grad is not None:
            assert param.grad.type() in (
                "torch.cuda.FloatTensor", "torch.FloatTensor"
            )
            grads.append(param.grad.detach())

    max_norm = float(max_norm)
    norm_type = float(norm_type)
    total_norm = 0.0

    if norm_type == torch.inf:
        total_norm = max(grad.abs().max() for grad in grads_for_norm)
        total_norm_cuda = torch.tensor(
            [float(total_norm)], dtype=torch.float,
            device=grads[0].device if grads else "cpu",
        )
        torch.distributed.all_reduce(
            total_norm_cuda, op=torch.distributed.ReduceOp.MAX,
            group=model_parallel_group,
        )
        total_norm = total_norm_cuda[0].item()
    else:
        if grads_for_norm:
            norms = torch._foreach_norm(grads_for_norm, norm_type)
            total_norm = torch.stack(norms).pow(norm_type).sum()
        else:
            device = grads[0].device if grads else (
                "cuda" if torch.cuda.is_available() else "cpu"
            )
            total_norm = torch.zeros(1, dtype=torch.float, device=device).squeeze()
        torch.distributed.all_reduce(
            total_norm, op=torch.distributed.ReduceOp.SUM,
            group=model_parallel_group,
        )
        total_norm = total_norm.item() ** (1.0 / norm_type)

    clip_coeff = max_norm / (total_norm + 1.0e-6)
    if clip_coeff < 1.0 and grads:
        torch._foreach_mul_(grads, clip_coeff)
    return total_norm


def count_zeros_fp32(parameters, model_parallel_group=None):
    if isinstance(parameters, torch.Tensor):
        parameters = [parameters]
    total_num_zeros = None
    for param in parameters:
        grad_not_none = param.grad is not None
        is_not_shared = param_is_not_shared(param)
        is_not_tp_duplicate = param_is_not_tensor_parallel_duplicate(param)
        if grad_not_none and is_not_shared and is_not_tp_duplicate:
            grad = param.grad.detach()
            num_zeros = grad.numel() - torch.count_nonzero(grad)
            total_num_zeros = (
                num_zeros if total_num_zeros is None else total_num_zeros + num_zeros
            )
    if total_num_zeros is None:
        device = "cuda" if torch.cuda.is_available() else "cpu"
        total_num_zeros = torch.zeros(1, dtype=torch.float, device=device).squeeze()
    else:
        total_num_zeros = total_num_zeros.float()
    torch.distributed.all_reduce(
        total_num_zeros, op=torch.distributed.ReduceOp.SUM,
        group=model_parallel_group,
    )
    return total_num_zeros.item()
