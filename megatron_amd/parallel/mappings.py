"""Autograd-paired TP region mappings (the f/g conjugate ops of the Megatron
paper; reference megatron/core/tensor_parallel/mappings.py:13-278).

All collectives are RCCL over the xGMI mesh via torch.distributed ("nccl"
backend on ROCm IS RCCL). On the MI355X node the TP group's 2/4/8 GPUs are a
fully-connected point-to-point mesh, so the all-gather / reduce-scatter pairs
used by sequence parallelism decompose into direct per-link transfers — prefer
them over all-reduce wherever the layer structure allows (they move the same
bytes but each link carries only its shard).
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from . import state as ps
from .utils import split_tensor_along_last_dim


# --- raw collectives -------------------------------------------------------

def _reduce(input_: torch.Tensor) -> torch.Tensor:
    if ps.get_tensor_model_parallel_world_size() == 1:
        return input_
    # collectives need dense storage; autograd can hand us stride-0
    # expanded grads (e.g. from .sum().backward())
    if not input_.is_contiguous():
        input_ = input_.contiguous()
    dist.all_reduce(input_, group=ps.get_tensor_model_parallel_group())
    return input_


def _split_along_last_dim(input_: torch.Tensor) -> torch.Tensor:
    world_size = ps.get_tensor_model_parallel_world_size()
    if world_size == 1:
        return input_
    input_list = split_tensor_along_last_dim(input_, world_size)
    rank = ps.get_tensor_model_parallel_rank()
    return input_list[rank].contiguous()


def _split_along_first_dim(input_: torch.Tensor) -> torch.Tensor:
    world_size = ps.get_tensor_model_parallel_world_size()
    if world_size == 1:
        return input_
    dim_size = input_.size()[0]
    assert dim_size % world_size == 0
    local_dim_size = dim_size // world_size
    rank = ps.get_tensor_model_parallel_rank()
    dim_offset = rank * local_dim_size
    return input_[dim_offset : dim_offset + local_dim_size].contiguous()


def _gather_along_last_dim(input_: torch.Tensor) -> torch.Tensor:
    world_size = ps.get_tensor_model_parallel_world_size()
    if world_size == 1:
        return input_
    last_dim = input_.dim() - 1
    rank = ps.get_tensor_model_parallel_rank()
    tensor_list = [torch.empty_like(input_) for _ in range(world_size)]
    tensor_list[rank] = input_
    dist.all_gather(tensor_list, input_, group=ps.get_tensor_model_parallel_group())
    return torch.cat(tensor_list, dim=last_dim).contiguous()


def _gather_along_first_dim(input_: torch.Tensor) -> torch.Tensor:
    world_size = ps.get_tensor_model_parallel_world_size()
    if world_size == 1:
        return input_
    dim_size = list(input_.size())
    dim_size[0] = dim_size[0] * world_size
    output = ps.get_global_memory_buffer().get_tensor(dim_size, input_.dtype, "mpu")
    dist.all_gather_into_tensor(
        output, input_.contiguous(), group=ps.get_tensor_model_parallel_group()
    )
    return output


def _reduce_scatter_along_first_dim(input_: torch.Tensor) -> torch.Tensor:
    world_size = ps.get_tensor_model_parallel_world_size()
    if world_size == 1:
        return input_
    dim_size = list(input_.size())
    assert dim_size[0] % world_size == 0
    dim_size[0] = dim_size[0] // world_size
    output = torch.empty(dim_size, dtype=input_.dtype, device=input_.device)
    dist.reduce_scatter_tensor(
        output, input_.contiguous(), group=ps.get_tensor_model_parallel_group()
    )
    return output


# --- autograd wrappers -----------------------------------------------------

class _CopyToModelParallelRegion(torch.autograd.Function):
    """f: identity fwd, all-reduce bwd."""

    @staticmethod
    def forward(ctx, input_):
        return input_

    @staticmethod
    def backward(ctx, grad_output):
        return _reduce(grad_output)


class _ReduceFromModelParallelRegion(torch.autograd.Function):
    """g: all-reduce fwd, identity bwd."""

    @staticmethod
    def forward(ctx, input_):
        return _reduce(input_)

    @staticmethod
    def backward(ctx, grad_output):
        return grad_output


class _ScatterToModelParallelRegion(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input_):
        return _split_along_last_dim(input_)

    @staticmethod
    def backward(ctx, grad_output):
        return _gather_along_last_dim(grad_output)


class _GatherFromModelParallelRegion(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input_):
        return _gather_along_last_dim(input_)

    @staticmethod
    def backward(ctx, grad_output):
        return _split_along_last_dim(grad_output)


class _ScatterToSequenceParallelRegion(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input_):
        return _split_along_first_dim(input_)

    @staticmethod
    def backward(ctx, grad_output):
        return _gather_along_first_dim(grad_output)


class _GatherFromSequenceParallelRegion(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input_, tensor_parallel_output_grad=True):
        ctx.tensor_parallel_output_grad = tensor_parallel_output_grad
        return _gather_along_first_dim(input_)

    @staticmethod
    def backward(ctx, grad_output):
        if ctx.tensor_parallel_output_grad:
            return _reduce_scatter_along_first_dim(grad_output), None
        return _split_along_first_dim(grad_output), None


class _ReduceScatterToSequenceParallelRegion(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input_):
        return _reduce_scatter_along_first_dim(input_)

    @staticmethod
    def backward(ctx, grad_output):
        return _gather_along_first_dim(grad_output)


# --- public API ------------------------------------------------------------

def copy_to_tensor_model_parallel_region(input_):
    return _CopyToModelParallelRegion.apply(input_)


def reduce_from_tensor_model_parallel_region(input_):
    return _ReduceFromModelParallelRegion.apply(input_)


def scatter_to_tensor_model_parallel_region(input_):
    return _ScatterToModelParallelRegion.apply(input_)


def gather_from_tensor_model_parallel_region(input_):
    return _GatherFromModelParallelRegion.apply(input_)


def scatter_to_sequence_parallel_region(input_):
    return _ScatterToSequenceParallelRegion.apply(input_)


def gather_from_sequence_parallel_region(input_, tensor_parallel_output_grad=True):
    return _GatherFromSequenceParallelRegion.apply(input_, tensor_parallel_output_grad)


def reduce_scatter_to_sequence_parallel_region(input_):
    return _ReduceScatterToSequenceParallelRegion.apply(input_)
