"""Broadcast a batch dict from TP-rank-0 to the rest of the TP group
(reference megatron/core/tensor_parallel/data.py:65-105): only TP rank 0 runs
the dataloader; sizes go first, then one flattened int64 payload."""

from __future__ import annotations

import torch

from . import state as ps

_MAX_DATA_DIM = 5


def _check_data_types(keys, data, target_dtype):
    for key in keys:
        assert data[key].dtype == target_dtype, (
            f"{key} has dtype {data[key].dtype} != {target_dtype}"
        )


def _build_key_size_numel_dictionaries(keys, data):
    max_dim = _MAX_DATA_DIM
    sizes = [0 for _ in range(max_dim) for _ in keys]

    if ps.get_tensor_model_parallel_rank() == 0:
        offset = 0
        for key in keys:
            assert data[key].dim() < max_dim, "you should increase MAX_DATA_DIM"
            size = data[key].size()
            for i, s in enumerate(size):
                sizes[i + offset] = s
            offset += max_dim

    device = torch.cuda.current_device() if torch.cuda.is_available() else "cpu"
    sizes_cuda = torch.tensor(sizes, dtype=torch.long, device=device)
    torch.distributed.broadcast(
        sizes_cuda, ps.get_tensor_model_parallel_src_rank(),
        group=ps.get_tensor_model_parallel_group(),
    )

    sizes_cpu = sizes_cuda.cpu()
    key_size = {}
    key_numel = {}
    total_numel = 0
    offset = 0
    for key in keys:
        i = 0
        size = []
        numel = 1
        while sizes_cpu[offset + i] > 0:
            this_size = sizes_cpu[offset + i]
            size.append(int(this_size))
            numel *= int(this_size)
            i += 1
        key_size[key] = size
        key_numel[key] = numel
        total_numel += numel
        offset += max_dim
    return key_size, key_numel, total_numel


def broadcast_data(keys, data, datatype):
    key_size, key_numel, total_numel = _build_key_size_numel_dictionaries(keys, data)
    device = torch.cuda.current_device() if torch.cuda.is_available() else "cpu"
    if ps.get_tensor_model_parallel_rank() == 0:
        _check_data_types(keys, data, datatype)
        flatten_data = torch.cat(
            [data[key].contiguous().view(-1) for key in keys], dim=0
        ).to(device)
    else:
        flatten_data = torch.empty(total_numel, device=device, dtype=datatype)

    torch.distributed.broadcast(
        flatten_data, ps.get_tensor_model_parallel_src_rank(),
        group=ps.get_tensor_model_parallel_group(),
    )

    output = {}
    offset = 0
    for key in keys:
        size = key_size[key]
        numel = key_numel[key]
        output[key] = flatten_data.narrow(0, offset, numel).view(size)
        offset += numel
    return output
