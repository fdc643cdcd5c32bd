"""Training driver: pretrain() orchestration, model/optimizer build, train
loop, eval loop, logging, exit conditions.

Reference: megatron/training.py:55-966.
"""

from __future__ import annotations

import math
import sys
import time

import torch

from . import global_state
from . import microbatches as mb
from . import parallel as mpu
from .config import get_config
from .checkpointing import load_checkpoint, save_checkpoint
from .initialize import initialize_megatron
from .models.enums import ModelType
from .models.module import Float16Module
from .optim import get_megatron_optimizer, get_optimizer_param_scheduler
from .parallel.ddp import DistributedDataParallel as LocalDDP
from .parallel.schedules import get_forward_backward_func
from .utils import (
    is_last_rank,
    calc_params_l2_norm,
    print_rank_0,
    print_rank_last,
    report_memory,
    unwrap_model,
)

_TRAIN_START_TIME = time.time()


def pretrain(train_valid_test_dataset_provider, model_provider, model_type,
             forward_step_func, extra_args_provider=None, args_defaults=None,
             cfg=None, process_non_loss_data_func=None):
    """Main training entry (reference training.py:55-169)."""
    global _TRAIN_START_TIME
    cfg = initialize_megatron(extra_args_provider, args_defaults, cfg=cfg)

    start_time_tensor = torch.tensor(
        [_TRAIN_START_TIME], dtype=torch.float,
        device="cuda" if torch.cuda.is_available() else "cpu",
    )
    torch.distributed.all_reduce(start_time_tensor,
                                 op=torch.distributed.ReduceOp.MIN)
    _TRAIN_START_TIME = start_time_tensor.item()
    print_rank_0(
        f"time to initialize megatron (seconds): "
        f"{time.time() - _TRAIN_START_TIME:.3f}"
    )

    timers = global_state.get_timers()

    timers("model-and-optimizer-setup", log_level=0).start(barrier=True)
    model, optimizer, opt_param_scheduler = _setup_model_and_optimizer(
        model_provider, model_type, cfg
    )
    timers("model-and-optimizer-setup").stop()

    timers("train/valid/test-data-iterators-setup", log_level=0).start(
        barrier=True
    )
    (
        train_data_iterator, valid_data_iterator, test_data_iterator,
    ) = build_train_valid_test_data_iterators(
        train_valid_test_dataset_provider, cfg
    )
    timers("train/valid/test-data-iterators-setup").stop()

    timers.log(["model-and-optimizer-setup",
                "train/valid/test-data-iterators-setup"], barrier=True)
    print_rank_0("training ...")

    iteration = cfg.iteration
    if cfg.do_train and (cfg.train_iters or 0) > 0:
        iteration = _train(
            forward_step_func, model, optimizer, opt_param_scheduler,
            train_data_iterator, valid_data_iterator, cfg,
            process_non_loss_data_func,
        )
    print_rank_0("after training is done")

    if cfg.do_valid:
        prefix = f"the end of training for val data"
        evaluate_and_print_results(
            prefix, forward_step_func, valid_data_iterator, model, iteration,
            cfg, process_non_loss_data_func, False,
        )
    if cfg.save and iteration != 0:
        save_checkpoint(iteration, model, optimizer, opt_param_scheduler, cfg)
    if cfg.do_test:
        prefix = "the end of training for test data"
        evaluate_and_print_results(
            prefix, forward_step_func, test_data_iterator, model, 0, cfg,
            process_non_loss_data_func, True,
        )
    return model


def get_model(model_provider_func, model_type=ModelType.encoder_or_decoder,
              wrap_with_ddp=True, cfg=None):
    """Build model chunks, move to device, wrap in Float16Module + LocalDDP
    (reference training.py:199-304)."""
    if cfg is None:
        cfg = get_config()
    cfg.model_type = model_type

    if (
        mpu.get_pipeline_model_parallel_world_size() > 1
        and cfg.virtual_pipeline_model_parallel_size is not None
    ):
        assert model_type != ModelType.encoder_and_decoder
        model = []
        for i in range(cfg.virtual_pipeline_model_parallel_size):
            mpu.set_virtual_pipeline_model_parallel_rank(i)
            pre_process = mpu.is_pipeline_first_stage()
            post_process = mpu.is_pipeline_last_stage()
            this_model = model_provider_func(
                pre_process=pre_process, post_process=post_process
            )
            this_model.model_type = model_type
            model.append(this_model)
    else:
        pre_process = mpu.is_pipeline_first_stage()
        post_process = mpu.is_pipeline_last_stage()
        model = model_provider_func(
            pre_process=pre_process, post_process=post_process
        )
        model.model_type = model_type

    if not isinstance(model, list):
        model = [model]

    for param in [p for m in model for p in m.parameters()]:
        if not hasattr(param, "model_parallel"):
            param.model_parallel = False

    if mpu.get_data_parallel_rank() == 0:
        num_params = sum(
            sum(p.nelement() for p in m.parameters()) for m in model
        )
        print(
            f" > number of parameters on (tensor, pipeline) model parallel "
            f"rank ({mpu.get_tensor_model_parallel_rank()}, "
            f"{mpu.get_pipeline_model_parallel_rank()}): {num_params}",
            flush=True,
        )

    if torch.cuda.is_available():
        for model_module in model:
            model_module.cuda(torch.cuda.current_device())

    if cfg.fp16 or cfg.bf16:
        model = [Float16Module(model_module, cfg) for model_module in model]

    if wrap_with_ddp:
        if cfg.DDP_impl == "torch":
            i = torch.cuda.current_device() if torch.cuda.is_available() else None
            model = [
                torch.nn.parallel.DistributedDataParallel(
                    model_module,
                    device_ids=[i] if i is not None else None,
                    output_device=i,
                    process_group=mpu.get_data_parallel_group(),
                )
                for model_module in model
            ]
        elif cfg.DDP_impl == "local":
            model = [
                LocalDDP(
                    model_module,
                    cfg.accumulate_allreduce_grads_in_fp32,
                    cfg.use_contiguous_buffers_in_local_ddp,
                    overlap_grad_reduce=cfg.overlap_grad_reduce,
                    bucket_numel=cfg.overlap_bucket_numel,
                )
                for model_module in model
            ]
            if cfg.data_parallel_random_init:
                for model_module in model:
                    model_module.broadcast_params()
        else:
            raise NotImplementedError(f"Unknown DDP implementation: {cfg.DDP_impl}")
    return model


def _setup_model_and_optimizer(model_provider_func, model_type, cfg,
                               no_wd_decay_cond=None, scale_lr_cond=None,
                               lr_mult=1.0):
    """(reference training.py:353-390)"""
    model = get_model(model_provider_func, model_type, cfg=cfg)
    unwrapped_model = unwrap_model(model, (LocalDDP, Float16Module))

    optimizer = get_megatron_optimizer(model, cfg, no_wd_decay_cond,
                                       scale_lr_cond, lr_mult)
    opt_param_scheduler = get_optimizer_param_scheduler(optimizer, cfg)

    if cfg.load is not None:
        timers = global_state.get_timers()
        timers("load-checkpoint", log_level=0).start(barrier=True)
        cfg.iteration = load_checkpoint(model, optimizer, opt_param_scheduler,
                                        cfg)
        timers("load-checkpoint").stop(barrier=True)
        timers.log(["load-checkpoint"])
    else:
        cfg.iteration = 0

    if (
        len(model) == 1
        and hasattr(unwrapped_model[0], "init_state_dict_from_bert")
    ):
        unwrapped_model[0].init_state_dict_from_bert()

    return model, optimizer, opt_param_scheduler


def train_step(forward_step_func, data_iterator, model, optimizer,
               opt_param_scheduler, cfg):
    """One global step (reference training.py:393-459)."""
    timers = global_state.get_timers()

    for partition in model:
        if hasattr(partition, "zero_grad_buffer"):
            partition.zero_grad_buffer()
    optimizer.zero_grad()

    forward_backward_func = get_forward_backward_func(cfg)
    losses_reduced = forward_backward_func(
        forward_step_func, data_iterator, model, optimizer, cfg, timers,
        forward_only=False,
    )

    if cfg.empty_unused_memory_level >= 1 and torch.cuda.is_available():
        torch.cuda.empty_cache()

    optimizer.reduce_model_grads(timers)

    timers("optimizer", log_level=1).start(
        barrier=cfg.barrier_with_L1_time
    )
    update_successful, grad_norm, num_zeros_in_grad = optimizer.step(timers)
    timers("optimizer").stop()

    if update_successful:
        increment = (
            mb.get_num_microbatches() * cfg.micro_batch_size
            * cfg.data_parallel_size
        )
        opt_param_scheduler.step(increment=increment)
        skipped_iter = 0
    else:
        skipped_iter = 1

    if cfg.empty_unused_memory_level >= 2 and torch.cuda.is_available():
        torch.cuda.empty_cache()

    if mpu.is_pipeline_last_stage(ignore_virtual=True):
        loss_reduced = {}
        for key in losses_reduced[0]:
            losses_reduced_for_key = [x[key] for x in losses_reduced]
            loss_reduced[key] = sum(losses_reduced_for_key) / len(
                losses_reduced_for_key
            )
        return loss_reduced, skipped_iter, grad_norm, num_zeros_in_grad
    return {}, skipped_iter, grad_norm, num_zeros_in_grad


def training_log(loss_dict, total_loss_dict, learning_rate, iteration,
                 loss_scale, report_memory_flag, skipped_iter, grad_norm,
                 params_norm, num_zeros_in_grad, cfg):
    """Console/TB logging (reference training.py:462-641, abbreviated)."""
    timers = global_state.get_timers()
    writer = global_state.get_tensorboard_writer()
    wandb_writer = global_state.get_wandb_writer()

    advanced_iters_key = "advanced iterations"
    skipped_iters_key = "skipped iterations"
    nan_iters_key = "nan iterations"
    if not skipped_iter:
        total_loss_dict[advanced_iters_key] = (
            total_loss_dict.get(advanced_iters_key, 0) + 1
        )
    else:
        total_loss_dict.setdefault(advanced_iters_key, 0)
    total_loss_dict[skipped_iters_key] = (
        total_loss_dict.get(skipped_iters_key, 0) + skipped_iter
    )
    got_nan = False
    for key in loss_dict:
        if not skipped_iter:
            total_loss_dict[key] = (
                total_loss_dict.get(
                    key, torch.tensor([0.0], device=loss_dict[key].device
                                      if torch.is_tensor(loss_dict[key])
                                      else "cpu")
                )
                + loss_dict[key]
            )
        else:
            value = loss_dict[key].float().sum().item() if torch.is_tensor(
                loss_dict[key]) else float(loss_dict[key])
            is_nan = value == float("inf") or value == -float("inf") or value != value
            got_nan = got_nan or is_nan
    total_loss_dict[nan_iters_key] = (
        total_loss_dict.get(nan_iters_key, 0) + int(got_nan)
    )

    batch_size = (
        cfg.micro_batch_size * cfg.data_parallel_size * mb.get_num_microbatches()
    )

    if iteration % cfg.log_interval == 0:
        elapsed_time = timers("interval-time").elapsed(barrier=True)
        elapsed_time_per_iteration = elapsed_time / max(
            1, total_loss_dict[advanced_iters_key] + total_loss_dict[skipped_iters_key]
        )
        seq_len = cfg.seq_length or 0
        tokens_per_sec = (
            batch_size * seq_len / elapsed_time_per_iteration
            if elapsed_time_per_iteration > 0 else 0.0
        )
        log_string = f" iteration {iteration:8d}/{cfg.train_iters:8d} |"
        log_string += (
            f" elapsed time per iteration (ms): "
            f"{elapsed_time_per_iteration * 1000.0:.1f} |"
        )
        log_string += f" tokens/sec: {tokens_per_sec:.1f} |"
        log_string += f" learning rate: {learning_rate:.3E} |"
        log_string += f" global batch size: {batch_size:5d} |"
        for key in total_loss_dict:
            if key not in (advanced_iters_key, skipped_iters_key, nan_iters_key):
                avg = total_loss_dict[key].item() / float(
                    max(1, total_loss_dict[advanced_iters_key])
                )
                log_string += f" {key}: {avg:.6E} |"
                if writer:
                    writer.add_scalar(key, avg, iteration)
                    writer.add_scalar(
                        key + " vs samples", avg, cfg.consumed_train_samples
                    )
                if wandb_writer:
                    wandb_writer.add_scalar(key, avg, iteration)
                total_loss_dict[key] = torch.tensor(
                    [0.0], device=total_loss_dict[key].device
                )
        log_string += f" loss scale: {loss_scale:.1f} |"
        if grad_norm is not None:
            log_string += f" grad norm: {grad_norm:.3f} |"
        if num_zeros_in_grad is not None:
            log_string += f" num zeros: {num_zeros_in_grad:.1f} |"
        if params_norm is not None:
            log_string += f" params norm: {params_norm:.3f} |"
        log_string += (
            f" number of skipped iterations: {total_loss_dict[skipped_iters_key]:3d} |"
        )
        log_string += (
            f" number of nan iterations: {total_loss_dict[nan_iters_key]:3d} |"
        )
        if writer and iteration % cfg.tensorboard_log_interval == 0:
            writer.add_scalar("learning-rate", learning_rate, iteration)
            writer.add_scalar("tokens-per-sec", tokens_per_sec, iteration)
            if cfg.log_timers_to_tensorboard:
                timers.write(
                    ["forward-compute", "backward-compute", "optimizer"],
                    writer, iteration, normalizer=cfg.log_interval,
                )
            if cfg.log_memory_to_tensorboard and torch.cuda.is_available():
                stats = torch.cuda.memory_stats()
                writer.add_scalar(
                    "mem-allocated-bytes",
                    stats.get("allocated_bytes.all.current", 0), iteration,
                )
                writer.add_scalar(
                    "mem-reserved-bytes",
                    stats.get("reserved_bytes.all.current", 0), iteration,
                )
                writer.add_scalar(
                    "mem-max-allocated-bytes",
                    stats.get("allocated_bytes.all.peak", 0), iteration,
                )
        if wandb_writer:
            wandb_writer.add_scalar("learning-rate", learning_rate, iteration)
            wandb_writer.add_scalar("tokens-per-sec", tokens_per_sec,
                                    iteration)
            wandb_writer.flush_all()
        total_loss_dict[advanced_iters_key] = 0
        total_loss_dict[skipped_iters_key] = 0
        total_loss_dict[nan_iters_key] = 0
        print_rank_last(log_string)
        if report_memory_flag and learning_rate > 0.0:
            report_memory(f"(after {iteration} iterations)")
            report_memory_flag = False
    return report_memory_flag


def _train(forward_step_func, model, optimizer, opt_param_scheduler,
           train_data_iterator, valid_data_iterator, cfg,
           process_non_loss_data_func=None):
    """Main loop (reference training.py:654-770)."""
    timers = global_state.get_timers()

    for model_module in model:
        model_module.train()

    total_loss_dict = {}
    iteration = cfg.iteration

    timers("interval-time", log_level=0).start(barrier=True)
    report_memory_flag = True
    while iteration < cfg.train_iters:
        mb.update_num_microbatches(cfg.consumed_train_samples)
        if iteration in (cfg.skip_iters or []):
            # skip backprop on selected iterations (fault-injection hook)
            iteration += 1
            continue
        loss_dict, skipped_iter, grad_norm, num_zeros_in_grad = train_step(
            forward_step_func, train_data_iterator, model, optimizer,
            opt_param_scheduler, cfg,
        )
        iteration += 1
        cfg.iteration = iteration
        new_samples = (
            mpu.get_data_parallel_world_size() * cfg.micro_batch_size
            * mb.get_num_microbatches()
        )
        cfg.consumed_train_samples += new_samples

        params_norm = None
        if cfg.log_params_norm:
            params_norm = calc_params_l2_norm(model)
        report_memory_flag = training_log(
            loss_dict, total_loss_dict,
            optimizer.param_groups[0]["lr"], iteration,
            optimizer.get_loss_scale().item(), report_memory_flag,
            skipped_iter, grad_norm, params_norm, num_zeros_in_grad, cfg,
        )

        if (
            cfg.eval_interval
            and iteration % cfg.eval_interval == 0
            and cfg.do_valid
        ):
            prefix = f"iteration {iteration}"
            evaluate_and_print_results(
                prefix, forward_step_func, valid_data_iterator, model,
                iteration, cfg, process_non_loss_data_func, False,
            )

        # cluster preemption (ADLR autoresume) polling
        if iteration % 50 == 0:
            from .utils import check_adlr_autoresume_termination

            check_adlr_autoresume_termination(iteration, model, optimizer,
                                              opt_param_scheduler, cfg)

        saved_checkpoint = False
        if cfg.exit_signal_handler:
            signal_handler = global_state.get_signal_handler()
            if any(signal_handler.signals_received()):
                save_checkpoint(iteration, model, optimizer,
                                opt_param_scheduler, cfg)
                print_rank_0("exiting program after receiving SIGTERM.")
                sys.exit()

        if cfg.save and cfg.save_interval and iteration % cfg.save_interval == 0:
            save_checkpoint(iteration, model, optimizer, opt_param_scheduler,
                            cfg)
            saved_checkpoint = True

        if cfg.exit_duration_in_mins:
            train_time = (time.time() - _TRAIN_START_TIME) / 60.0
            done_tensor = torch.tensor(
                [train_time > cfg.exit_duration_in_mins], dtype=torch.int,
                device="cuda" if torch.cuda.is_available() else "cpu",
            )
            torch.distributed.all_reduce(done_tensor,
                                         op=torch.distributed.ReduceOp.MAX)
            if done_tensor.item():
                if not saved_checkpoint and cfg.save:
                    save_checkpoint(iteration, model, optimizer,
                                    opt_param_scheduler, cfg)
                print_rank_0(f"exiting program after {train_time} minutes")
                sys.exit()

        if cfg.exit_interval and iteration % cfg.exit_interval == 0:
            if not saved_checkpoint and cfg.save:
                save_checkpoint(iteration, model, optimizer,
                                opt_param_scheduler, cfg)
            torch.distributed.barrier()
            print_rank_0(f"exiting program at iteration {iteration}")
            sys.exit()

    return iteration


def evaluate(forward_step_func, data_iterator, model,
             process_non_loss_data_func, cfg, verbose=False):
    """(reference training.py:773-826)"""
    for model_module in model:
        model_module.eval()

    total_loss_dict = {}
    with torch.no_grad():
        iteration = 0
        while iteration < cfg.eval_iters:
            iteration += 1
            if verbose and iteration % cfg.log_interval == 0:
                print_rank_0(f"Evaluating iter {iteration}/{cfg.eval_iters}")
            forward_backward_func = get_forward_backward_func(cfg)
            timers = global_state.get_timers()
            loss_dicts = forward_backward_func(
                forward_step_func, data_iterator, model, None, cfg, timers,
                forward_only=True,
            )
            if mpu.is_pipeline_last_stage(ignore_virtual=True):
                for loss_dict in loss_dicts:
                    for key in loss_dict:
                        if key not in total_loss_dict:
                            total_loss_dict[key] = torch.tensor(
                                0.0,
                                device=loss_dict[key].device
                                if torch.is_tensor(loss_dict[key]) else "cpu",
                            )
                        total_loss_dict[key] += loss_dict[key]
            cfg.consumed_valid_samples += (
                mpu.get_data_parallel_world_size() * cfg.micro_batch_size
                * mb.get_num_microbatches()
            )
        collected_non_loss_data = None
        if process_non_loss_data_func is not None and is_last_rank():
            # one extra collection pass whose raw outputs go to the
            # post-processor (reference training.py:813-818)
            forward_backward_func = get_forward_backward_func(cfg)
            collected_non_loss_data = forward_backward_func(
                forward_step_func, data_iterator, model, None, cfg,
                global_state.get_timers(), forward_only=True,
                collect_non_loss_data=True,
            )
    for model_module in model:
        model_module.train()
    for key in total_loss_dict:
        total_loss_dict[key] /= cfg.eval_iters * mb.get_num_microbatches()
    return total_loss_dict, collected_non_loss_data


def evaluate_and_print_results(prefix, forward_step_func, data_iterator, model,
                               iteration, cfg, process_non_loss_data_func=None,
                               verbose=False):
    """(reference training.py:829-874)"""
    writer = global_state.get_tensorboard_writer()
    total_loss_dict, collected_non_loss_data = evaluate(
        forward_step_func, data_iterator, model, process_non_loss_data_func,
        cfg, verbose,
    )
    if process_non_loss_data_func is not None and is_last_rank():
        process_non_loss_data_func(collected_non_loss_data, iteration,
                                   writer)
    string = f" validation loss at {prefix} | "
    for key in total_loss_dict:
        value = total_loss_dict[key].item() if torch.is_tensor(
            total_loss_dict[key]) else total_loss_dict[key]
        string += f"{key} value: {value:.6E} | "
        ppl = math.exp(min(20, value))
        string += f"{key} PPL: {ppl:.6E} | "
        if writer:
            writer.add_scalar(f"{key} validation", value, iteration)
            if cfg.log_validation_ppl_to_tensorboard:
                writer.add_scalar(f"{key} validation ppl", ppl, iteration)
    length = len(string) + 1
    print_rank_last("-" * length)
    print_rank_last(string)
    print_rank_last("-" * length)


def cyclic_iter(iter):
    while True:
        for x in iter:
            yield x


def build_train_valid_test_data_iterators(
    build_train_valid_test_datasets_provider, cfg
):
    """(reference training.py:877-966)"""
    train_dataloader, valid_dataloader, test_dataloader = None, None, None
    print_rank_0("> building train, validation, and test datasets ...")

    if mpu.get_tensor_model_parallel_rank() == 0:
        if cfg.train_iters is not None:
            train_samples = cfg.train_iters * cfg.global_batch_size
        elif cfg.train_samples is not None:
            train_samples = cfg.train_samples
        else:
            train_samples = 0
        eval_iters = 0
        test_iters = 0
        if cfg.train_iters and cfg.eval_interval:
            eval_iters = (
                cfg.train_iters // cfg.eval_interval + 1
            ) * cfg.eval_iters
            test_iters = cfg.eval_iters
        train_val_test_num_samples = [
            train_samples,
            eval_iters * cfg.global_batch_size,
            test_iters * cfg.global_batch_size,
        ]
        train_ds, valid_ds, test_ds = build_train_valid_test_datasets_provider(
            train_val_test_num_samples
        )
        from .data.samplers import build_pretraining_data_loader

        train_dataloader = build_pretraining_data_loader(
            train_ds, cfg.consumed_train_samples, cfg
        )
        valid_dataloader = build_pretraining_data_loader(
            valid_ds, cfg.consumed_valid_samples, cfg
        )
        test_dataloader = build_pretraining_data_loader(test_ds, 0, cfg)

        do_train = train_dataloader is not None and (cfg.train_iters or 0) > 0
        do_valid = valid_dataloader is not None and cfg.eval_iters > 0
        do_test = test_dataloader is not None and cfg.eval_iters > 0
        flags = torch.tensor(
            [int(do_train), int(do_valid), int(do_test)], dtype=torch.long,
            device="cuda" if torch.cuda.is_available() else "cpu",
        )
    else:
        flags = torch.tensor(
            [0, 0, 0], dtype=torch.long,
            device="cuda" if torch.cuda.is_available() else "cpu",
        )

    torch.distributed.broadcast(
        flags, mpu.get_tensor_model_parallel_src_rank(),
        group=mpu.get_tensor_model_parallel_group(),
    )
    cfg.do_train = bool(flags[0].item())
    cfg.do_valid = bool(flags[1].item())
    cfg.do_test = bool(flags[2].item())

    dl_type = cfg.dataloader_type
    assert dl_type in ("single", "cyclic")

    def _iter(dl):
        if dl is None:
            return None
        return iter(dl) if dl_type == "single" else iter(cyclic_iter(dl))

    return _iter(train_dataloader), _iter(valid_dataloader), _iter(test_dataloader)
