"""Pipeline-parallel point-to-point communication over RCCL.

Reference: megatron/p2p_communication.py:11-406. Activations/grads move
between pipeline neighbors as batched send/recv (torch.distributed.P2POp +
batch_isend_irecv -> RCCL send/recv over xGMI intra-node, IB/NIC inter-node).

MI355X differences vs the reference:
 - no trailing torch.cuda.synchronize() after each batch (reference
   p2p_communication.py:230-231 'to protect against race condition'): RCCL
   ops are stream-ordered and req.wait() inserts the event dependency into
   the compute stream — the defensive global sync would serialize the whole
   device;
 - the optional scatter-gather optimization (split payload 1/tp before
   send, all-gather after recv) is kept: with SP the tensors are already
   sequence-sharded so it is a no-op there.
"""

from __future__ import annotations

import torch

from . import state as ps
from .utils import (
    divide,
    gather_split_1d_tensor,
    split_tensor_into_1d_equal_chunks,
)


def _communicate_shapes(tensor_send_next, tensor_send_prev, recv_prev, recv_next,
                        cfg):
    """Exchange tensor shapes before payloads (variable seq lengths;
    reference p2p_communication.py:36-98)."""
    recv_prev_shape_tensor = None
    recv_next_shape_tensor = None
    send_prev_shape_tensor = None
    send_next_shape_tensor = None
    device = torch.cuda.current_device() if torch.cuda.is_available() else "cpu"
    if recv_prev:
        recv_prev_shape_tensor = torch.empty(3, device=device, dtype=torch.int64)
    if recv_next:
        recv_next_shape_tensor = torch.empty(3, device=device, dtype=torch.int64)
    if tensor_send_prev is not None:
        send_prev_shape_tensor = torch.tensor(
            tensor_send_prev.size(), device=device, dtype=torch.int64
        )
    if tensor_send_next is not None:
        send_next_shape_tensor = torch.tensor(
            tensor_send_next.size(), device=device, dtype=torch.int64
        )

    ops = []
    if send_prev_shape_tensor is not None:
        ops.append(torch.distributed.P2POp(
            torch.distributed.isend, send_prev_shape_tensor,
            ps.get_pipeline_model_parallel_prev_rank(),
        ))
    if recv_prev_shape_tensor is not None:
        ops.append(torch.distributed.P2POp(
            torch.distributed.irecv, recv_prev_shape_tensor,
            ps.get_pipeline_model_parallel_prev_rank(),
        ))
    if send_next_shape_tensor is not None:
        ops.append(torch.distributed.P2POp(
            torch.distributed.isend, send_next_shape_tensor,
            ps.get_pipeline_model_parallel_next_rank(),
        ))
    if recv_next_shape_tensor is not None:
        ops.append(torch.distributed.P2POp(
            torch.distributed.irecv, recv_next_shape_tensor,
            ps.get_pipeline_model_parallel_next_rank(),
        ))
    if ops:
        reqs = torch.distributed.batch_isend_irecv(ops)
        for req in reqs:
            req.wait()

    recv_prev_shape = [0, 0, 0]
    if recv_prev_shape_tensor is not None:
        recv_prev_shape = recv_prev_shape_tensor.tolist()
    recv_next_shape = [0, 0, 0]
    if recv_next_shape_tensor is not None:
        recv_next_shape = recv_next_shape_tensor.tolist()
    return recv_prev_shape, recv_next_shape


def _communicate(tensor_send_next, tensor_send_prev, recv_prev, recv_next,
                 tensor_shape, cfg, dtype_=None):
    """Batched bidirectional neighbor exchange
    (reference p2p_communication.py:101-251)."""
    tensor_recv_prev = None
    tensor_recv_next = None

    if not cfg.variable_seq_lengths:
        recv_prev_shape = tensor_shape
        recv_next_shape = tensor_shape
    else:
        recv_prev_shape, recv_next_shape = _communicate_shapes(
            tensor_send_next, tensor_send_prev, recv_prev, recv_next, cfg
        )

    split = (
        cfg.scatter_gather_tensors_in_pipeline
        and not cfg.sequence_parallel
        and ps.get_tensor_model_parallel_world_size() > 1
    )
    if split:
        def chunk_shape(shape):
            numel = 1
            for s in shape:
                numel *= s
            return (numel // ps.get_tensor_model_parallel_world_size(),)
    else:
        def chunk_shape(shape):
            return shape

    dtype = cfg.params_dtype
    if cfg.fp32_residual_connection:
        dtype = torch.float
    if dtype_ is not None:
        dtype = dtype_

    # split path: the raw chunk must NOT require grad (it is all-gathered
    # in-place across the TP group first — autograd forbids that on a
    # grad-tracking buffer); grad is re-attached on the gathered tensor
    requires_grad = not split
    device = torch.cuda.current_device() if torch.cuda.is_available() else "cpu"
    if recv_prev:
        tensor_recv_prev = torch.empty(
            chunk_shape(recv_prev_shape), requires_grad=requires_grad,
            device=device, dtype=dtype,
        )
    if recv_next:
        tensor_recv_next = torch.empty(
            chunk_shape(recv_next_shape), requires_grad=requires_grad,
            device=device, dtype=dtype,
        )

    if split:
        if tensor_send_next is not None:
            tensor_send_next = split_tensor_into_1d_equal_chunks(
                tensor_send_next, new_buffer=True
            )
        if tensor_send_prev is not None:
            tensor_send_prev = split_tensor_into_1d_equal_chunks(
                tensor_send_prev, new_buffer=True
            )

    ops = []
    if tensor_send_prev is not None:
        ops.append(torch.distributed.P2POp(
            torch.distributed.isend, tensor_send_prev,
            ps.get_pipeline_model_parallel_prev_rank(),
        ))
    if tensor_recv_prev is not None:
        ops.append(torch.distributed.P2POp(
            torch.distributed.irecv, tensor_recv_prev,
            ps.get_pipeline_model_parallel_prev_rank(),
        ))
    if tensor_send_next is not None:
        ops.append(torch.distributed.P2POp(
            torch.distributed.isend, tensor_send_next,
            ps.get_pipeline_model_parallel_next_rank(),
        ))
    if tensor_recv_next is not None:
        ops.append(torch.distributed.P2POp(
            torch.distributed.irecv, tensor_recv_next,
            ps.get_pipeline_model_parallel_next_rank(),
        ))
    if ops:
        reqs = torch.distributed.batch_isend_irecv(ops)
        for req in reqs:
            req.wait()

    if split:
        if recv_prev:
            tensor_recv_prev = (
                gather_split_1d_tensor(tensor_recv_prev)
                .view(recv_prev_shape)
                .requires_grad_()
            )
        if recv_next:
            tensor_recv_next = (
                gather_split_1d_tensor(tensor_recv_next)
                .view(recv_next_shape)
                .requires_grad_()
            )
    return tensor_recv_prev, tensor_recv_next


# --- public ops (reference p2p_communication.py:254-406) -------------------


def recv_forward(tensor_shape, cfg, dtype_=None, timers=None):
    if ps.is_pipeline_first_stage():
        return None
    if timers:
        timers("forward-recv", log_level=2).start()
    input_tensor, _ = _communicate(None, None, True, False, tensor_shape, cfg,
                                   dtype_=dtype_)
    if timers:
        timers("forward-recv").stop()
    return input_tensor


def recv_backward(tensor_shape, cfg, timers=None):
    if ps.is_pipeline_last_stage():
        return None
    if timers:
        timers("backward-recv", log_level=2).start()
    _, output_tensor_grad = _communicate(None, None, False, True, tensor_shape,
                                         cfg)
    if timers:
        timers("backward-recv").stop()
    return output_tensor_grad


def send_forward(output_tensor, cfg, timers=None):
    if ps.is_pipeline_last_stage():
        return
    if timers:
        timers("forward-send", log_level=2).start()
    _communicate(output_tensor, None, False, False, None, cfg)
    if timers:
        timers("forward-send").stop()


def send_backward(input_tensor_grad, cfg, timers=None):
    if ps.is_pipeline_first_stage():
        return
    if timers:
        timers("backward-send", log_level=2).start()
    _communicate(None, input_tensor_grad, False, False, None, cfg)
    if timers:
        timers("backward-send").stop()


def send_forward_recv_backward(output_tensor, tensor_shape, cfg, timers=None):
    if ps.is_pipeline_last_stage():
        return None
    if timers:
        timers("forward-send-backward-recv", log_level=2).start()
    _, output_tensor_grad = _communicate(
        output_tensor, None, False, True, tensor_shape, cfg
    )
    if timers:
        timers("forward-send-backward-recv").stop()
    return output_tensor_grad


def send_backward_recv_forward(input_tensor_grad, tensor_shape, cfg, timers=None):
    if ps.is_pipeline_first_stage():
        return None
    if timers:
        timers("backward-send-forward-recv", log_level=2).start()
    input_tensor, _ = _communicate(
        None, input_tensor_grad, True, False, tensor_shape, cfg
    )
    if timers:
        timers("backward-send-forward-recv").stop()
    return input_tensor


def send_forward_recv_forward(output_tensor, recv_prev, tensor_shape, cfg,
                              timers=None):
    if timers:
        timers("forward-send-forward-recv", log_level=2).start()
    input_tensor, _ = _communicate(
        output_tensor, None, recv_prev, False, tensor_shape, cfg
    )
    if timers:
        timers("forward-send-forward-recv").stop()
    return input_tensor


def send_backward_recv_backward(input_tensor_grad, recv_next, tensor_shape, cfg,
                                timers=None):
    if timers:
        timers("backward-send-backward-recv", log_level=2).start()
    _, output_tensor_grad = _communicate(
        None, input_tensor_grad, False, recv_next, tensor_shape, cfg
    )
    if timers:
        timers("backward-send-backward-recv").stop()
    return output_tensor_grad


def send_forward_backward_recv_forward_backward(
    output_tensor, input_tensor_grad, recv_prev, recv_next, tensor_shape, cfg,
    timers=None,
):
    if timers:
        timers("forward-backward-send-forward-backward-recv",
               log_level=2).start()
    input_tensor, output_tensor_grad = _communicate(
        output_tensor, input_tensor_grad, recv_prev, recv_next, tensor_shape,
        cfg,
    )
    if timers:
        timers("forward-backward-send-forward-backward-recv").stop()
    return input_tensor, output_tensor_grad
