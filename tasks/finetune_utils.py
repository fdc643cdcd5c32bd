"""Generic downstream finetuning driver (reference tasks/finetune_utils.py):
epoch-based train loop over a map-style dataset, built on the same
train_step / evaluate machinery as pretraining."""

from __future__ import annotations

import functools
import sys

import torch

from megatron_amd import global_state
from megatron_amd import microbatches as mb
from megatron_amd import parallel as mpu
from megatron_amd import training as mt
from megatron_amd.checkpointing import load_checkpoint, save_checkpoint
from megatron_amd.config import get_config
from megatron_amd.models import ModelType
from megatron_amd.utils import (
    average_losses_across_data_parallel_group,
    check_adlr_autoresume_termination,
    print_rank_0,
)


def process_batch(batch, fp16=False):
    """dict of numpy/torch arrays -> model inputs on the right device."""
    device = "cuda" if torch.cuda.is_available() else "cpu"
    tokens = batch["text"].long().to(device).contiguous()
    types = batch["types"].long().to(device).contiguous()
    labels = batch["label"].long().to(device).contiguous()
    attention_mask = batch["padding_mask"].float().to(device).contiguous()
    if fp16:
        attention_mask = attention_mask.half()
    return tokens, types, labels, attention_mask


def cross_entropy_loss_func(labels, output_tensor):
    logits = output_tensor
    loss = torch.nn.functional.cross_entropy(
        logits.contiguous().float(), labels
    )
    averaged = average_losses_across_data_parallel_group([loss])
    return loss, {"lm loss": averaged[0]}


def _cross_entropy_forward_step(batch, model):
    """Classification forward with CE loss; `batch` may be the dict itself
    or an iterator over dicts (the schedule passes whichever it was given)."""
    cfg = get_config()
    timers = global_state.get_timers()
    timers("batch-generator", log_level=2).start()
    try:
        batch_ = next(batch)
    except TypeError:
        batch_ = batch
    tokens, types, labels, attention_mask = process_batch(batch_, cfg.fp16)
    timers("batch-generator").stop()
    output_tensor = model(tokens, attention_mask, tokentype_ids=types)
    return output_tensor, functools.partial(cross_entropy_loss_func, labels)


def build_data_loader(dataset, micro_batch_size, num_workers, drop_last,
                      task_collate_fn=None):
    """Per-GPU dataloader over the task dataset, sharded across DP ranks."""
    world_size = mpu.get_data_parallel_world_size()
    rank = mpu.get_data_parallel_rank()
    sampler = torch.utils.data.distributed.DistributedSampler(
        dataset, num_replicas=world_size, rank=rank
    )
    return torch.utils.data.DataLoader(
        dataset, batch_size=micro_batch_size, sampler=sampler, shuffle=False,
        num_workers=num_workers, drop_last=drop_last,
        pin_memory=torch.cuda.is_available(), collate_fn=task_collate_fn,
    )


def _infinite(dataloader):
    while True:
        for batch in dataloader:
            yield batch


def _build_train_valid_dataloaders(train_dataset, valid_dataset, cfg,
                                   task_collate_fn=None):
    print_rank_0("building train and validation dataloaders ...")
    train_dataloader = build_data_loader(
        train_dataset, cfg.micro_batch_size, cfg.num_workers,
        not cfg.keep_last, task_collate_fn,
    )
    cfg.train_iters_per_epoch = len(train_dataloader)
    cfg.train_iters = cfg.epochs * cfg.train_iters_per_epoch
    valid_dataloader = _infinite(build_data_loader(
        valid_dataset, cfg.micro_batch_size, cfg.num_workers,
        not cfg.keep_last, task_collate_fn,
    ))

    # multi-choice datasets collapse their options into the batch dim
    cfg.orig_micro_batch_size = cfg.micro_batch_size
    cfg.orig_global_batch_size = cfg.global_batch_size
    if hasattr(train_dataset, "sample_multiplier"):
        cfg.micro_batch_size *= train_dataset.sample_multiplier
        cfg.global_batch_size *= train_dataset.sample_multiplier
    return train_dataloader, valid_dataloader


def _train(model, optimizer, opt_param_scheduler, forward_step,
           train_dataloader, valid_dataloader, end_of_epoch_callback, cfg):
    timers = global_state.get_timers()
    assert mb.get_num_microbatches() == 1, (
        "finetuning does not support gradient accumulation"
    )
    for m in model:
        m.train()

    losses_dict_sum = {}
    start_epoch = cfg.iteration // cfg.train_iters_per_epoch
    start_iteration = cfg.iteration % cfg.train_iters_per_epoch
    iteration = cfg.iteration
    report_memory_flag = True

    timers("interval-time", log_level=0).start(barrier=True)
    for epoch in range(start_epoch, cfg.epochs):
        print_rank_0(f"working on epoch {epoch + 1} ...")
        train_dataloader.sampler.set_epoch(cfg.seed + epoch)

        for iteration_, batch in enumerate(train_dataloader):
            if iteration_ < start_iteration:
                continue
            start_iteration = 0

            losses_dict, skipped_iter, grad_norm, num_zeros = mt.train_step(
                forward_step, batch, model, optimizer, opt_param_scheduler,
                cfg,
            )
            iteration += 1
            cfg.iteration = iteration

            report_memory_flag = mt.training_log(
                losses_dict, losses_dict_sum,
                optimizer.param_groups[0]["lr"], iteration,
                optimizer.get_loss_scale().item(), report_memory_flag,
                skipped_iter, grad_norm, None, num_zeros, cfg,
            )

            if (
                cfg.adlr_autoresume
                and iteration % cfg.adlr_autoresume_interval == 0
            ):
                check_adlr_autoresume_termination(
                    iteration, model, optimizer, opt_param_scheduler, cfg
                )

            saved = False
            if cfg.save and cfg.save_interval and (
                iteration % cfg.save_interval == 0
            ):
                save_checkpoint(iteration, model, optimizer,
                                opt_param_scheduler, cfg)
                saved = True

            if cfg.eval_interval and iteration % cfg.eval_interval == 0:
                mt.evaluate_and_print_results(
                    f"iteration {iteration}", forward_step, valid_dataloader,
                    model, iteration, cfg, None, False,
                )

            if cfg.exit_interval and iteration % cfg.exit_interval == 0:
                if not saved and cfg.save:
                    save_checkpoint(iteration, model, optimizer,
                                    opt_param_scheduler, cfg)
                torch.distributed.barrier()
                print_rank_0(f"exiting program at iteration {iteration}")
                sys.exit()

        if cfg.save:
            save_checkpoint(iteration, model, optimizer, opt_param_scheduler,
                            cfg)
        if end_of_epoch_callback is not None:
            end_of_epoch_callback(model, epoch)


def finetune(train_valid_datasets_provider, model_provider,
             model_type=ModelType.encoder_or_decoder,
             forward_step=_cross_entropy_forward_step,
             end_of_epoch_callback_provider=None, task_collate_fn=None):
    """Main finetune entry shared by GLUE / RACE (and usable by new tasks)."""
    cfg = get_config()
    assert cfg.rampup_batch_size is None, (
        "batch-size rampup is not supported for finetuning"
    )

    train_dataloader = valid_dataloader = None
    if cfg.epochs > 0:
        train_dataset, valid_dataset = train_valid_datasets_provider()
        train_dataloader, valid_dataloader = _build_train_valid_dataloaders(
            train_dataset, valid_dataset, cfg, task_collate_fn
        )
    else:
        cfg.train_iters = 0

    end_of_epoch_callback = None
    if end_of_epoch_callback_provider is not None:
        end_of_epoch_callback = end_of_epoch_callback_provider()

    model, optimizer, opt_param_scheduler = mt._setup_model_and_optimizer(
        model_provider, model_type, cfg
    )

    # load ONLY the pretrained backbone when starting fresh
    if cfg.iteration == 0 and cfg.pretrained_checkpoint is not None:
        original_load, original_rng = cfg.load, cfg.no_load_rng
        cfg.load = cfg.pretrained_checkpoint
        cfg.no_load_rng = True
        load_checkpoint(model, None, None, cfg, strict=False)
        cfg.load, cfg.no_load_rng = original_load, original_rng
        optimizer.reload_model_params()

    print_rank_0("training ...")
    if cfg.epochs > 0:
        _train(model, optimizer, opt_param_scheduler, forward_step,
               train_dataloader, valid_dataloader, end_of_epoch_callback, cfg)
    elif end_of_epoch_callback is not None:
        print_rank_0("evaluation only mode, setting epoch to -1")
        end_of_epoch_callback(model, epoch=-1, output_predictions=True)
    print_rank_0("done :-)")
    return model
