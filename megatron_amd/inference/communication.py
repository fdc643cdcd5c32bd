"""Inference-time pipeline p2p + first/last-stage broadcasts
(reference megatron/text_generation/communication.py:13-186)."""

from __future__ import annotations

import torch

from .. import parallel as mpu


def recv_from_prev_pipeline_rank_(recv_buffer):
    recv_prev_op = torch.distributed.P2POp(
        torch.distributed.irecv, recv_buffer,
        mpu.get_pipeline_model_parallel_prev_rank(),
    )
    reqs = torch.distributed.batch_isend_irecv([recv_prev_op])
    for req in reqs:
        req.wait()


def send_to_next_pipeline_rank(tensor):
    send_next_op = torch.distributed.P2POp(
        torch.distributed.isend, tensor.contiguous(),
        mpu.get_pipeline_model_parallel_next_rank(),
    )
    reqs = torch.distributed.batch_isend_irecv([send_next_op])
    for req in reqs:
        req.wait()


def _is_cuda(tensor):
    assert torch.is_tensor(tensor)
    return tensor.is_cuda


def broadcast_from_last_pipeline_stage(size, dtype, tensor=None):
    is_last_stage = mpu.is_pipeline_last_stage()
    if mpu.get_pipeline_model_parallel_world_size() == 1:
        return tensor
    device = torch.cuda.current_device() if torch.cuda.is_available() else "cpu"
    if is_last_stage:
        assert tensor is not None
        tensor = tensor.contiguous()
    else:
        tensor = torch.empty(size, dtype=dtype, device=device)
    src = mpu.get_pipeline_model_parallel_last_rank()
    group = mpu.get_pipeline_model_parallel_group()
    torch.distributed.broadcast(tensor, src, group)
    return tensor


def broadcast_from_last_to_first_pipeline_stage(size, dtype, tensor=None):
    is_last_stage = mpu.is_pipeline_last_stage()
    is_first_stage = mpu.is_pipeline_first_stage()
    if is_first_stage and is_last_stage:
        return tensor
    if not (is_first_stage or is_last_stage):
        return None
    device = torch.cuda.current_device() if torch.cuda.is_available() else "cpu"
    if is_last_stage:
        assert tensor is not None
        tensor = tensor.contiguous()
    else:
        tensor = torch.empty(size, dtype=dtype, device=device)
    src = mpu.get_pipeline_model_parallel_last_rank()
    group = mpu.get_embedding_group()
    torch.distributed.broadcast(tensor, src, group)
    return tensor


def copy_from_last_to_first_pipeline_stage(size, dtype, tensor=None):
    is_last_stage = mpu.is_pipeline_last_stage()
    is_first_stage = mpu.is_pipeline_first_stage()
    if is_first_stage and is_last_stage:
        return
    if not (is_first_stage or is_last_stage):
        return
    assert tensor is not None
    tensor = tensor.contiguous() if is_last_stage else tensor
    src = mpu.get_pipeline_model_parallel_last_rank()
    group = mpu.get_embedding_group()
    if is_last_stage:
        torch.distributed.broadcast(tensor, src, group)
    else:
        tensor_ = torch.empty(size, dtype=dtype, device=tensor.device)
        torch.distributed.broadcast(tensor_, src, group)
        tensor.copy_(tensor_)


def broadcast_tensor(size, dtype, tensor=None, rank=0):
    device = torch.cuda.current_device() if torch.cuda.is_available() else "cpu"
    if torch.distributed.get_rank() == rank:
        assert tensor is not None
        tensor = tensor.contiguous()
    else:
        tensor = torch.empty(size, dtype=dtype, device=device)
    torch.distributed.broadcast(tensor, rank)
    return tensor


def broadcast_list(size, dtype, list_values=None, rank=0):
    tensor = None
    device = torch.cuda.current_device() if torch.cuda.is_available() else "cpu"
    if torch.distributed.get_rank() == rank:
        tensor = torch.tensor(list_values, dtype=dtype, device=device)
    return broadcast_tensor(size, dtype, tensor=tensor, rank=rank)


def broadcast_int_list(size, int_list=None, rank=0):
    return broadcast_list(size, torch.int64, list_values=int_list, rank=rank)


def broadcast_float_list(size, float_list=None, rank=0):
    return broadcast_list(size, torch.float32, list_values=float_list,
                          rank=rank)
