"""Small TP helpers (reference megatron/core/tensor_parallel/utils.py:9-108 and
megatron/core/utils.py)."""

from __future__ import annotations

from typing import List

import torch

from . import state as ps


def ensure_divisibility(numerator: int, denominator: int) -> None:
    assert numerator % denominator == 0, f"{numerator} not divisible by {denominator}"


def divide(numerator: int, denominator: int) -> int:
    ensure_divisibility(numerator, denominator)
    return numerator // denominator


def split_tensor_along_last_dim(
    tensor: torch.Tensor, num_partitions: int, contiguous_split_chunks: bool = False
) -> List[torch.Tensor]:
    last_dim = tensor.dim() - 1
    last_dim_size = divide(tensor.size()[last_dim], num_partitions)
    tensor_list = torch.split(tensor, last_dim_size, dim=last_dim)
    if contiguous_split_chunks:
        return [chunk.contiguous() for chunk in tensor_list]
    return list(tensor_list)


def split_tensor_into_1d_equal_chunks(tensor: torch.Tensor, new_buffer=False):
    """This rank's 1/tp slice of the flattened tensor (used by distributed
    activation recompute and pipeline scatter-gather)."""
    partition_size = divide(tensor.numel(), ps.get_tensor_model_parallel_world_size())
    start_index = partition_size * ps.get_tensor_model_parallel_rank()
    end_index = start_index + partition_size
    if new_buffer:
        data = torch.empty(
            partition_size,
            dtype=tensor.dtype,
            device=tensor.device,
            requires_grad=False,
        )
        data.copy_(tensor.view(-1)[start_index:end_index])
    else:
        data = tensor.view(-1)[start_index:end_index]
    return data


def gather_split_1d_tensor(tensor: torch.Tensor) -> torch.Tensor:
    """All-gather a tensor split with split_tensor_into_1d_equal_chunks.

    Runs under no_grad on a detached input: the collective writes the
    output in place (and flattens via views), which autograd rejects in a
    grad-enabled context — the TP>1 x PP>1 1F1B steady state calls this
    with grad mode on (caught by test_tp2_pp2_train_step)."""
    numel_gathered = tensor.numel() * ps.get_tensor_model_parallel_world_size()
    with torch.no_grad():
        gathered = torch.empty(
            numel_gathered, dtype=tensor.dtype,
            device=tensor.device, requires_grad=False,
        )
        torch.distributed.all_gather_into_tensor(
            gathered, tensor.detach(),
            group=ps.get_tensor_model_parallel_group(),
        )
    return gathered


class VocabUtility:
    """Vocab range helpers for the vocab-parallel embedding / CE
    (reference core/tensor_parallel/utils.py:60-108)."""

    @staticmethod
    def vocab_range_from_per_partition_vocab_size(
        per_partition_vocab_size: int, rank: int, world_size: int
    ):
        index_f = rank * per_partition_vocab_size
        index_l = index_f + per_partition_vocab_size
        return index_f, index_l

    @staticmethod
    def vocab_range_from_global_vocab_size(global_vocab_size, rank, world_size):
        per_partition_vocab_size = divide(global_vocab_size, world_size)
        return VocabUtility.vocab_range_from_per_partition_vocab_size(
            per_partition_vocab_size, rank, world_size
        )


# --- viewless-tensor machinery (reference megatron/core/utils.py:44-124) ---


class MakeViewlessTensor(torch.autograd.Function):
    @staticmethod
    def forward(ctx, inp, requires_grad):
        return torch.empty(
            (1,), dtype=inp.dtype, device=inp.device, requires_grad=requires_grad
        ).set_(inp.data)

    @staticmethod
    def backward(ctx, grad_output):
        return grad_output, None


def make_viewless_tensor(inp, requires_grad, keep_graph):
    if inp._base is None:
        return inp
    if keep_graph:
        return MakeViewlessTensor.apply(inp, requires_grad)
    out = torch.empty(
        (1,), dtype=inp.dtype, device=inp.device, requires_grad=requires_grad
    )
    out.data = inp.data
    return out


def assert_viewless_tensor(tensor, extra_msg=None):
    if isinstance(tensor, list):
        for t in tensor:
            assert_viewless_tensor(t)
        return tensor
    if not isinstance(tensor, torch.Tensor):
        return tensor
    assert tensor._base is None, f"Ensure tensor._base is None before setting data. {extra_msg}"
    return tensor


def safely_set_viewless_tensor_data(tensor, new_data_tensor):
    assert_viewless_tensor(tensor)
    tensor.data = new_data_tensor
