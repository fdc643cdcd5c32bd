"""Pipeline schedules: no-pipelining grad accumulation, non-interleaved 1F1B,
interleaved 1F1B over virtual model chunks.

Reference: megatron/schedules.py:18-722. Memory tricks kept:
deallocate_output_tensor (free activation payload, keep grad_fn graph node,
:36-53) and custom_backward (call the C++ autograd engine directly so the
deallocated output's shape check is skipped, :55-88).
"""

from __future__ import annotations


import torch
from torch.autograd.variable import Variable

from .. import microbatches
from ..models.enums import ModelType
from ..utils import unwrap_model
from . import p2p, state as ps


def get_forward_backward_func(cfg):
    if ps.get_pipeline_model_parallel_world_size() > 1:
        if cfg.virtual_pipeline_model_parallel_size is not None:
            return forward_backward_pipelining_with_interleaving
        return forward_backward_pipelining_without_interleaving
    return forward_backward_no_pipelining


def deallocate_output_tensor(out):
    """Replace the output tensor's payload with a scalar, keeping the autograd
    graph node alive (reference schedules.py:36-53)."""
    if out is None:
        return
    assert isinstance(out, torch.Tensor)
    assert out._base is None, "counter-productive to free a view of another tensor"
    out.data = torch.empty((1,), device=out.device, dtype=out.dtype)


def custom_backward(output, grad_output):
    """Directly call the autograd engine, skipping the shape-match check that
    torch.autograd.backward would fail on a deallocated output
    (reference schedules.py:55-88)."""
    assert output.numel() == 1, "output should be pseudo-'freed' in schedule"
    assert isinstance(output, torch.Tensor)
    assert isinstance(grad_output, (torch.Tensor, type(None)))
    if grad_output is None:
        assert output.numel() == 1
        grad_output = torch.ones_like(output, memory_format=torch.preserve_format)
    Variable._execution_engine.run_backward(
        tensors=(output,), grad_tensors=(grad_output,), keep_graph=False,
        create_graph=False, inputs=tuple(), allow_unreachable=True,
        accumulate_grad=True,
    )


def forward_step(forward_step_func, data_iterator, model, input_tensor,
                 forward_data_store, cfg, timers, collect_non_loss_data=False):
    """(reference schedules.py:91-139)"""
    if timers is not None:
        timers("forward-compute", log_level=2).start()

    unwrapped_model = unwrap_model(model)
    unwrap_output_tensor = False
    if not isinstance(input_tensor, list):
        input_tensor = [input_tensor]
        unwrap_output_tensor = True

    unwrapped_model.set_input_tensor(input_tensor[0])
    output_tensor, loss_func = forward_step_func(data_iterator, model)
    if ps.is_pipeline_last_stage():
        if not collect_non_loss_data:
            output_tensor = loss_func(output_tensor)
            loss, loss_reduced = output_tensor
            output_tensor = loss / microbatches.get_num_microbatches()
            forward_data_store.append(loss_reduced)
        else:
            data = loss_func(output_tensor, non_loss_data=True)
            forward_data_store.append(data)

    if timers is not None:
        timers("forward-compute").stop()

    if unwrap_output_tensor:
        return output_tensor
    return [output_tensor]


def backward_step(optimizer, input_tensor, output_tensor, output_tensor_grad,
                  cfg, timers):
    """(reference schedules.py:142-202)"""
    if timers is not None:
        timers("backward-compute", log_level=2).start()

    unwrap_input_tensor_grad = False
    if not isinstance(input_tensor, list):
        input_tensor = [input_tensor]
        unwrap_input_tensor_grad = True
    for x in input_tensor:
        if x is not None:
            x.retain_grad()

    if not isinstance(output_tensor, list):
        output_tensor = [output_tensor]
    if not isinstance(output_tensor_grad, list):
        output_tensor_grad = [output_tensor_grad]

    # scale loss on the last stage
    if output_tensor_grad[0] is None and optimizer is not None:
        output_tensor = optimizer.scale_loss(output_tensor[0])
        custom_backward(output_tensor, output_tensor_grad[0])
    else:
        custom_backward(output_tensor[0], output_tensor_grad[0])

    input_tensor_grad = [None]
    if input_tensor is not None:
        input_tensor_grad = []
        for x in input_tensor:
            if x is None:
                input_tensor_grad.append(None)
            else:
                input_tensor_grad.append(x.grad)

    if timers is not None:
        timers("backward-compute").stop()

    if unwrap_input_tensor_grad:
        return input_tensor_grad[0]
    return input_tensor_grad


def forward_backward_no_pipelining(
    forward_step_func, data_iterator, model, optimizer, cfg, timers,
    forward_only, collect_non_loss_data=False,
):
    """Grad-accumulation loop without pipelining (reference schedules.py:213-250)."""
    assert len(model) == 1
    model = model[0]

    import contextlib
    context_handler = torch.no_grad if forward_only else contextlib.nullcontext
    forward_data_store = []
    input_tensor, output_tensor_grad = None, None

    num_microbatches = microbatches.get_num_microbatches()
    for i in range(num_microbatches - 1):
        with context_handler():
            output_tensor = forward_step(
                forward_step_func, data_iterator, model, input_tensor,
                forward_data_store, cfg, timers, collect_non_loss_data,
            )
        if not forward_only:
            backward_step(optimizer, input_tensor, output_tensor,
                          output_tensor_grad, cfg, timers)

    with context_handler():
        output_tensor = forward_step(
            forward_step_func, data_iterator, model, input_tensor,
            forward_data_store, cfg, timers, collect_non_loss_data,
        )
    if not forward_only:
        # arm the bucketed async DP all-reduce for the final backward so
        # grad communication overlaps the remaining backward compute
        if hasattr(model, "enable_grad_sync"):
            model.enable_grad_sync()
        backward_step(optimizer, input_tensor, output_tensor,
                      output_tensor_grad, cfg, timers)
    return forward_data_store


def get_tensor_shapes(rank, model_type, cfg):
    """(reference schedules.py:505-535). SP divides the sequence dim by tp."""
    tensor_shapes = []
    seq_length = cfg.seq_length
    if cfg.sequence_parallel:
        seq_length = seq_length // ps.get_tensor_model_parallel_world_size()
    if model_type == ModelType.encoder_and_decoder:
        decoder_seq_length = cfg.decoder_seq_length
        if cfg.sequence_parallel:
            decoder_seq_length = (
                decoder_seq_length // ps.get_tensor_model_parallel_world_size()
            )
        if ps.is_pipeline_stage_before_split(rank):
            tensor_shapes.append(
                (seq_length, cfg.micro_batch_size, cfg.hidden_size)
            )
        else:
            tensor_shapes.append(
                (decoder_seq_length, cfg.micro_batch_size, cfg.hidden_size)
            )
            tensor_shapes.append(
                (seq_length, cfg.micro_batch_size, cfg.hidden_size)
            )
    else:
        tensor_shapes.append((seq_length, cfg.micro_batch_size, cfg.hidden_size))
    return tensor_shapes


def recv_forward(tensor_shapes, cfg, timers):
    input_tensors = []
    for tensor_shape in tensor_shapes:
        if tensor_shape is None:
            input_tensors.append(None)
        else:
            input_tensors.append(p2p.recv_forward(tensor_shape, cfg,
                                                  timers=timers))
    return input_tensors


def recv_backward(tensor_shapes, cfg, timers):
    output_tensor_grads = []
    for tensor_shape in tensor_shapes:
        if tensor_shape is None:
            output_tensor_grads.append(None)
        else:
            output_tensor_grads.append(
                p2p.recv_backward(tensor_shape, cfg, timers=timers)
            )
    return output_tensor_grads


def send_forward(output_tensors, tensor_shapes, cfg, timers):
    if not isinstance(output_tensors, list):
        output_tensors = [output_tensors]
    for output_tensor, tensor_shape in zip(output_tensors, tensor_shapes):
        if tensor_shape is None:
            continue
        p2p.send_forward(output_tensor, cfg, timers=timers)


def send_backward(input_tensor_grads, tensor_shapes, cfg, timers):
    if not isinstance(input_tensor_grads, list):
        input_tensor_grads = [input_tensor_grads]
    for input_tensor_grad, tensor_shape in zip(input_tensor_grads, tensor_shapes):
        if tensor_shape is None:
            continue
        p2p.send_backward(input_tensor_grad, cfg, timers=timers)


def send_forward_recv_backward(output_tensors, tensor_shapes, cfg, timers):
    if not isinstance(output_tensors, list):
        output_tensors = [output_tensors]
    output_tensor_grads = []
    for output_tensor, tensor_shape in zip(output_tensors, tensor_shapes):
        if tensor_shape is None:
            output_tensor_grads.append(None)
            continue
        output_tensor_grads.append(
            p2p.send_forward_recv_backward(output_tensor, tensor_shape, cfg,
                                           timers=timers)
        )
    return output_tensor_grads


def send_backward_recv_forward(input_tensor_grads, tensor_shapes, cfg, timers):
    if not isinstance(input_tensor_grads, list):
        input_tensor_grads = [input_tensor_grads]
    input_tensors = []
    for input_tensor_grad, tensor_shape in zip(input_tensor_grads, tensor_shapes):
        if tensor_shape is None:
            input_tensors.append(None)
            continue
        input_tensors.append(
            p2p.send_backward_recv_forward(input_tensor_grad, tensor_shape, cfg,
                                           timers=timers)
        )
    return input_tensors


def forward_backward_pipelining_without_interleaving(
    forward_step_func, data_iterator, model, optimizer, cfg, timers,
    forward_only, collect_non_loss_data=False,
):
    """Non-interleaved 1F1B (reference schedules.py:606-722)."""
    assert len(model) == 1
    model = model[0]

    num_microbatches = microbatches.get_num_microbatches()
    pipeline_parallel_size = ps.get_pipeline_model_parallel_world_size()
    pipeline_parallel_rank = ps.get_pipeline_model_parallel_rank()

    num_warmup_microbatches = (
        pipeline_parallel_size - pipeline_parallel_rank - 1
    )
    num_warmup_microbatches = min(num_warmup_microbatches, num_microbatches)
    num_microbatches_remaining = num_microbatches - num_warmup_microbatches

    unwrapped_model = unwrap_model(model)
    model_type = getattr(unwrapped_model, "model_type",
                         ModelType.encoder_or_decoder)
    rank = pipeline_parallel_rank
    recv_tensor_shapes = get_tensor_shapes(rank - 1, model_type, cfg)
    send_tensor_shapes = get_tensor_shapes(rank, model_type, cfg)

    input_tensors = [] if not forward_only else None
    output_tensors = [] if not forward_only else None
    forward_data_store = []

    # warmup
    for i in range(num_warmup_microbatches):
        input_tensor = recv_forward(recv_tensor_shapes, cfg, timers)
        output_tensor = forward_step(
            forward_step_func, data_iterator, model, input_tensor,
            forward_data_store, cfg, timers, collect_non_loss_data,
        )
        send_forward(output_tensor, send_tensor_shapes, cfg, timers)
        if not forward_only:
            input_tensors.append(input_tensor)
            output_tensors.append(output_tensor)
            deallocate_output_tensor(output_tensor[0])

    if num_microbatches_remaining > 0:
        input_tensor = recv_forward(recv_tensor_shapes, cfg, timers)

    # steady state 1F1B
    for i in range(num_microbatches_remaining):
        last_iteration = i == (num_microbatches_remaining - 1)
        output_tensor = forward_step(
            forward_step_func, data_iterator, model, input_tensor,
            forward_data_store, cfg, timers, collect_non_loss_data,
        )
        if forward_only:
            send_forward(output_tensor, send_tensor_shapes, cfg, timers)
            if not last_iteration:
                input_tensor = recv_forward(recv_tensor_shapes, cfg, timers)
        else:
            output_tensor_grad = send_forward_recv_backward(
                output_tensor, send_tensor_shapes, cfg, timers
            )
            input_tensors.append(input_tensor)
            output_tensors.append(output_tensor)
            deallocate_output_tensor(output_tensor[0])

            input_tensor = input_tensors.pop(0)
            output_tensor = output_tensors.pop(0)
            input_tensor_grad = backward_step(
                optimizer, input_tensor, output_tensor, output_tensor_grad,
                cfg, timers,
            )
            if last_iteration:
                input_tensor = None
                send_backward(input_tensor_grad, recv_tensor_shapes, cfg, timers)
            else:
                input_tensor = send_backward_recv_forward(
                    input_tensor_grad, recv_tensor_shapes, cfg, timers
                )

    # cooldown
    if not forward_only:
        for i in range(num_warmup_microbatches):
            input_tensor = input_tensors.pop(0)
            output_tensor = output_tensors.pop(0)
            output_tensor_grad = recv_backward(send_tensor_shapes, cfg, timers)
            input_tensor_grad = backward_step(
                optimizer, input_tensor, output_tensor, output_tensor_grad,
                cfg, timers,
            )
            send_backward(input_tensor_grad, recv_tensor_shapes, cfg, timers)

    return forward_data_store


def forward_backward_pipelining_with_interleaving(
    forward_step_func, data_iterator, model, optimizer, cfg, timers,
    forward_only, collect_non_loss_data=False,
):
    """Interleaved 1F1B over virtual model chunks (reference schedules.py:253-502)."""
    input_tensors = [[] for _ in range(len(model))]
    output_tensors = [[] for _ in range(len(model))]
    forward_data_store = []
    if not forward_only:
        output_tensor_grads = [[] for _ in range(len(model))]

    pipeline_parallel_size = ps.get_pipeline_model_parallel_world_size()
    pipeline_parallel_rank = ps.get_pipeline_model_parallel_rank()

    seq_length = cfg.seq_length
    if cfg.sequence_parallel:
        seq_length = seq_length // ps.get_tensor_model_parallel_world_size()
    tensor_shape = (seq_length, cfg.micro_batch_size, cfg.hidden_size)

    num_model_chunks = len(model)
    num_microbatches = microbatches.get_num_microbatches() * num_model_chunks
    all_warmup_microbatches = False
    if forward_only:
        num_warmup_microbatches = num_microbatches
    else:
        if microbatches.get_num_microbatches() == pipeline_parallel_size:
            num_warmup_microbatches = num_microbatches
            all_warmup_microbatches = True
        else:
            num_warmup_microbatches = (
                pipeline_parallel_size - pipeline_parallel_rank - 1
            ) * 2
            num_warmup_microbatches += (num_model_chunks - 1) * pipeline_parallel_size
            num_warmup_microbatches = min(num_warmup_microbatches, num_microbatches)
    num_microbatches_remaining = num_microbatches - num_warmup_microbatches

    def get_model_chunk_id(microbatch_id, forward):
        microbatch_id_in_group = microbatch_id % (
            pipeline_parallel_size * num_model_chunks
        )
        model_chunk_id = microbatch_id_in_group // pipeline_parallel_size
        if not forward:
            model_chunk_id = num_model_chunks - model_chunk_id - 1
        return model_chunk_id

    def forward_step_helper(microbatch_id):
        model_chunk_id = get_model_chunk_id(microbatch_id, forward=True)
        ps.set_virtual_pipeline_model_parallel_rank(model_chunk_id)

        if ps.is_pipeline_first_stage():
            if len(input_tensors[model_chunk_id]) == len(
                output_tensors[model_chunk_id]
            ):
                input_tensors[model_chunk_id].append(None)
        input_tensor = input_tensors[model_chunk_id][-1]
        output_tensor = forward_step(
            forward_step_func, data_iterator[model_chunk_id],
            model[model_chunk_id], input_tensor, forward_data_store, cfg,
            timers, collect_non_loss_data,
        )
        output_tensors[model_chunk_id].append(output_tensor)
        if forward_only:
            input_tensors[model_chunk_id].pop()
            output_tensors[model_chunk_id].pop()
        return output_tensor

    def backward_step_helper(microbatch_id):
        model_chunk_id = get_model_chunk_id(microbatch_id, forward=False)
        ps.set_virtual_pipeline_model_parallel_rank(model_chunk_id)

        if ps.is_pipeline_last_stage():
            if len(output_tensor_grads[model_chunk_id]) == 0:
                output_tensor_grads[model_chunk_id].append(None)
        input_tensor = input_tensors[model_chunk_id].pop(0)
        output_tensor = output_tensors[model_chunk_id].pop(0)
        output_tensor_grad = output_tensor_grads[model_chunk_id].pop(0)
        return backward_step(
            optimizer, input_tensor, output_tensor, output_tensor_grad, cfg,
            timers,
        )

    # warmup
    ps.set_virtual_pipeline_model_parallel_rank(0)
    input_tensors[0].append(p2p.recv_forward(tensor_shape, cfg, timers=timers))
    for k in range(num_warmup_microbatches):
        output_tensor = forward_step_helper(k)
        next_forward_model_chunk_id = get_model_chunk_id(k + 1, forward=True)
        recv_prev = True
        if ps.is_pipeline_first_stage(ignore_virtual=True):
            if next_forward_model_chunk_id == 0:
                recv_prev = False
        if k == (num_microbatches - 1):
            recv_prev = False
        if ps.is_pipeline_last_stage():
            output_tensor = None

        if (
            k == (num_warmup_microbatches - 1)
            and not forward_only
            and not all_warmup_microbatches
        ):
            input_tensor_grad = None
            recv_next = True
            if ps.is_pipeline_last_stage(ignore_virtual=True):
                recv_next = False
            (
                input_tensor, output_tensor_grad,
            ) = p2p.send_forward_backward_recv_forward_backward(
                output_tensor, input_tensor_grad, recv_prev, recv_next,
                tensor_shape, cfg, timers=timers,
            )
            output_tensor_grads[num_model_chunks - 1].append(output_tensor_grad)
        else:
            input_tensor = p2p.send_forward_recv_forward(
                output_tensor, recv_prev, tensor_shape, cfg, timers=timers
            )
        input_tensors[next_forward_model_chunk_id].append(input_tensor)
        if output_tensor is not None:
            deallocate_output_tensor(output_tensor[0] if isinstance(
                output_tensor, list) else output_tensor)

    # steady state
    for k in range(num_microbatches_remaining):
        forward_k = k + num_warmup_microbatches
        output_tensor = forward_step_helper(forward_k)
        backward_k = k
        input_tensor_grad = backward_step_helper(backward_k)

        forward_model_chunk_id = get_model_chunk_id(forward_k, forward=True)
        ps.set_virtual_pipeline_model_parallel_rank(forward_model_chunk_id)
        if ps.is_pipeline_last_stage():
            output_tensor = None
        backward_model_chunk_id = get_model_chunk_id(backward_k, forward=False)
        ps.set_virtual_pipeline_model_parallel_rank(backward_model_chunk_id)
        if ps.is_pipeline_first_stage():
            input_tensor_grad = None

        recv_prev = True
        if ps.is_pipeline_first_stage(ignore_virtual=True):
            next_forward_model_chunk_id = get_model_chunk_id(
                forward_k - (pipeline_parallel_size - 1), forward=True
            )
            if next_forward_model_chunk_id == (num_model_chunks - 1):
                recv_prev = False
            next_forward_model_chunk_id += 1
        else:
            next_forward_model_chunk_id = get_model_chunk_id(
                forward_k + 1, forward=True
            )

        recv_next = True
        if ps.is_pipeline_last_stage(ignore_virtual=True):
            next_backward_model_chunk_id = get_model_chunk_id(
                backward_k - (pipeline_parallel_size - 1), forward=False
            )
            if next_backward_model_chunk_id == 0:
                recv_next = False
            next_backward_model_chunk_id -= 1
        else:
            next_backward_model_chunk_id = get_model_chunk_id(
                backward_k + 1, forward=False
            )

        if k == (num_microbatches_remaining - 1):
            recv_prev = False

        (
            input_tensor, output_tensor_grad,
        ) = p2p.send_forward_backward_recv_forward_backward(
            output_tensor, input_tensor_grad, recv_prev, recv_next,
            tensor_shape, cfg, timers=timers,
        )
        if output_tensor is not None:
            deallocate_output_tensor(output_tensor[0] if isinstance(
                output_tensor, list) else output_tensor)

        if recv_prev:
            input_tensors[next_forward_model_chunk_id].append(input_tensor)
        if recv_next:
            output_tensor_grads[next_backward_model_chunk_id].append(
                output_tensor_grad
            )

    # cooldown
    if not forward_only:
        if all_warmup_microbatches:
            output_tensor_grads[num_model_chunks - 1].append(
                p2p.recv_backward(tensor_shape, cfg, timers=timers)
            )
        for k in range(num_microbatches_remaining, num_microbatches):
            input_tensor_grad = backward_step_helper(k)
            next_backward_model_chunk_id = get_model_chunk_id(k + 1,
                                                              forward=False)
            recv_next = True
            if ps.is_pipeline_last_stage(ignore_virtual=True):
                if next_backward_model_chunk_id == (num_model_chunks - 1):
                    recv_next = False
            if k == (num_microbatches - 1):
                recv_next = False
            output_tensor_grads[next_backward_model_chunk_id].append(
                p2p.send_backward_recv_backward(
                    input_tensor_grad, recv_next, tensor_shape, cfg,
                    timers=timers,
                )
            )

    return forward_data_store
