"""General utilities (reference megatron/utils.py)."""

from __future__ import annotations

import torch

from .. import parallel as mpu


def unwrap_model(model, module_instances=None):
    """Strip DDP / Float16Module wrappers (reference utils.py:23-35)."""
    from ..models.module import Float16Module
    from ..parallel.ddp import DistributedDataParallel as LocalDDP

    if module_instances is None:
        module_instances = (LocalDDP, Float16Module)
    return_list = True
    if not isinstance(model, list):
        model = [model]
        return_list = False
    unwrapped = []
    for m in model:
        while isinstance(m, module_instances):
            m = m.module
        unwrapped.append(m)
    if not return_list:
        return unwrapped[0]
    return unwrapped


def average_losses_across_data_parallel_group(losses):
    """(reference utils.py:70-79)"""
    averaged = torch.cat([loss.clone().detach().view(1) for loss in losses])
    torch.distributed.all_reduce(averaged, group=mpu.get_data_parallel_group())
    averaged /= torch.distributed.get_world_size(group=mpu.get_data_parallel_group())
    return averaged


def calc_params_l2_norm(model, cfg=None):
    """L2 norm over unique (non-shared, non-duplicated-TP) params
    (reference utils.py:38-67)."""
    from ..models.module import param_is_not_shared

    if not isinstance(model, list):
        model = [model]
    params_data = []
    for model_ in model:
        for param in model_.parameters():
            is_not_shared = param_is_not_shared(param)
            is_not_tp_duplicate = param_is_not_tensor_parallel_duplicate(param)
            if is_not_shared and is_not_tp_duplicate:
                params_data.append(param.data.float())
    if params_data:
        norm2 = torch.stack([p.norm(2) ** 2 for p in params_data]).sum()
    else:
        device = "cuda" if torch.cuda.is_available() else "cpu"
        norm2 = torch.zeros(1, device=device).squeeze()
    torch.distributed.all_reduce(norm2, op=torch.distributed.ReduceOp.SUM,
                                 group=mpu.get_model_parallel_group())
    return norm2.item() ** 0.5


def param_is_not_tensor_parallel_duplicate(param):
    return (
        getattr(param, "model_parallel", False)
        or mpu.get_tensor_model_parallel_rank() == 0
    )


def get_ltor_masks_and_position_ids(
    data, eod_token, reset_position_ids, reset_attention_mask, eod_mask_loss
):
    """Causal mask + loss mask + position ids with optional EOD resets
    (reference utils.py:137-194)."""
    micro_batch_size, seq_length = data.size()

    att_mask_batch = micro_batch_size if reset_attention_mask else 1
    attention_mask = torch.tril(
        torch.ones((att_mask_batch, seq_length, seq_length), device=data.device)
    ).view(att_mask_batch, 1, seq_length, seq_length)

    loss_mask = torch.ones(data.size(), dtype=torch.float, device=data.device)
    if eod_mask_loss:
        loss_mask[data == eod_token] = 0.0

    position_ids = torch.arange(seq_length, dtype=torch.long, device=data.device)
    position_ids = position_ids.unsqueeze(0).expand_as(data)
    if reset_position_ids:
        position_ids = position_ids.clone()

    if reset_position_ids or reset_attention_mask:
        for b in range(micro_batch_size):
            eod_index = position_ids[b, data[b] == eod_token]
            if reset_position_ids:
                eod_index = eod_index.clone()
            prev_index = 0
            for j in range(eod_index.size()[0]):
                i = eod_index[j]
                if reset_attention_mask:
                    attention_mask[b, 0, (i + 1):, : (i + 1)] = 0
                if reset_position_ids:
                    position_ids[b, (i + 1):] -= i + 1 - prev_index
                    prev_index = i + 1

    # invert: True = masked position
    attention_mask = attention_mask < 0.5
    return attention_mask, loss_mask, position_ids


def report_memory(name):
    if not torch.cuda.is_available():
        return
    mega_bytes = 1024.0 * 1024.0
    string = name + " memory (MB)"
    string += f" | allocated: {torch.cuda.memory_allocated() / mega_bytes}"
    string += f" | max allocated: {torch.cuda.max_memory_allocated() / mega_bytes}"
    string += f" | reserved: {torch.cuda.memory_reserved() / mega_bytes}"
    string += f" | max reserved: {torch.cuda.max_memory_reserved() / mega_bytes}"
    if mpu.get_data_parallel_rank() == 0:
        print(f"[Rank {torch.distributed.get_rank()}] {string}", flush=True)


def print_rank_0(message):
    if torch.distributed.is_initialized():
        if torch.distributed.get_rank() == 0:
            print(message, flush=True)
    else:
        print(message, flush=True)


def is_last_rank():
    return torch.distributed.get_rank() == (torch.distributed.get_world_size() - 1)


def print_rank_last(message):
    if torch.distributed.is_initialized():
        if is_last_rank():
            print(message, flush=True)
    else:
        print(message, flush=True)


# --- ADLR autoresume (cluster preemption) hook (reference utils.py:117-134) --

_ADLR_AUTORESUME = None


def get_adlr_autoresume():
    """Lazily import the site-specific AutoResume module if present."""
    global _ADLR_AUTORESUME
    if _ADLR_AUTORESUME is None:
        import os
        import sys

        sys.path.append(os.environ.get("SUBMIT_SCRIPTS", "."))
        try:
            from userlib.auto_resume import AutoResume  # site package

            AutoResume.init()
            _ADLR_AUTORESUME = AutoResume
        except ImportError:
            _ADLR_AUTORESUME = False
    return _ADLR_AUTORESUME or None


def check_adlr_autoresume_termination(iteration, model, optimizer,
                                      opt_param_scheduler, cfg):
    """Save a checkpoint and request resume when the cluster signals
    preemption."""
    import sys

    import torch

    from ..checkpointing import save_checkpoint

    autoresume = get_adlr_autoresume()
    if autoresume is None:
        return
    torch.distributed.barrier()
    if autoresume.termination_requested():
        if cfg.save:
            save_checkpoint(iteration, model, optimizer, opt_param_scheduler,
                            cfg)
        print_rank_0(">>> autoresume termination request found!")
        if torch.distributed.get_rank() == 0:
            autoresume.request_resume()
        print_rank_0(">>> training terminated. Returning")
        sys.exit(0)
