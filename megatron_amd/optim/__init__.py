"""Optimizer factory (reference megatron/optimizer/__init__.py:63-128)."""

from __future__ import annotations

import torch

from ..models.norms import LayerNorm, RMSNorm
from .adam import FusedAdam, FusedSGD
from .distrib_optimizer import DistributedOptimizer
from .grad_scaler import ConstantGradScaler, DynamicGradScaler
from .optimizer import FP32Optimizer, Float16OptimizerWithFloat16Params
from .scheduler import OptimizerParamScheduler


def _get_params_for_weight_decay_optimization(models):
    """No weight decay on biases and norm weights (reference
    megatron/optimizer/__init__.py:28-60)."""
    weight_decay_params = {"params": []}
    no_weight_decay_params = {"params": [], "weight_decay": 0.0}
    for module in models:
        for module_ in module.modules():
            if isinstance(module_, (LayerNorm, RMSNorm)):
                no_weight_decay_params["params"].extend(
                    [p for p in module_._parameters.values() if p is not None]
                )
            else:
                weight_decay_params["params"].extend(
                    [
                        p
                        for n, p in module_._parameters.items()
                        if p is not None and n != "bias"
                    ]
                )
                no_weight_decay_params["params"].extend(
                    [
                        p
                        for n, p in module_._parameters.items()
                        if p is not None and n == "bias"
                    ]
                )
    return weight_decay_params, no_weight_decay_params


def get_megatron_optimizer(models, cfg, no_wd_decay_cond=None,
                           scale_lr_cond=None, lr_mult=1.0):
    param_groups = [
        g for g in _get_params_for_weight_decay_optimization(models)
        if g["params"]
    ]

    if cfg.optimizer == "adam":
        optimizer = FusedAdam(
            param_groups, lr=cfg.lr, weight_decay=cfg.weight_decay,
            betas=(cfg.adam_beta1, cfg.adam_beta2), eps=cfg.adam_eps,
        )
    elif cfg.optimizer == "sgd":
        optimizer = FusedSGD(
            param_groups, lr=cfg.lr, weight_decay=cfg.weight_decay,
            momentum=cfg.sgd_momentum,
        )
    else:
        raise Exception(f"{cfg.optimizer} optimizer is not supported")

    params_have_main_grad = cfg.DDP_impl == "local"

    if cfg.fp16 or cfg.bf16 or cfg.use_distributed_optimizer:
        grad_scaler = None
        if cfg.loss_scale:
            grad_scaler = ConstantGradScaler(cfg.loss_scale)
        elif cfg.fp16:
            grad_scaler = DynamicGradScaler(
                initial_scale=cfg.initial_loss_scale,
                min_scale=cfg.min_loss_scale,
                growth_factor=2.0, backoff_factor=0.5,
                growth_interval=cfg.loss_scale_window,
                hysteresis=cfg.hysteresis,
            )
        opt_cls = (
            DistributedOptimizer
            if cfg.use_distributed_optimizer
            else Float16OptimizerWithFloat16Params
        )
        return opt_cls(
            optimizer, cfg.clip_grad, cfg.log_num_zeros_in_grad,
            params_have_main_grad, cfg.use_contiguous_buffers_in_local_ddp,
            cfg.fp16, cfg.bf16, cfg.params_dtype, grad_scaler, models, cfg,
        )

    return FP32Optimizer(
        optimizer, cfg.clip_grad, cfg.log_num_zeros_in_grad,
        params_have_main_grad, cfg.use_contiguous_buffers_in_local_ddp,
        models, cfg,
    )


def get_optimizer_param_scheduler(optimizer, cfg):
    """(reference training.py:307-351)"""
    if cfg.train_iters is not None:
        if cfg.lr_decay_iters is None:
            cfg.lr_decay_iters = cfg.train_iters
        lr_decay_steps = cfg.lr_decay_iters * cfg.global_batch_size
        wd_incr_steps = cfg.train_iters * cfg.global_batch_size
        if cfg.lr_warmup_fraction is not None:
            lr_warmup_steps = cfg.lr_warmup_fraction * lr_decay_steps
        else:
            lr_warmup_steps = cfg.lr_warmup_iters * cfg.global_batch_size
    elif cfg.train_samples is not None:
        if cfg.lr_decay_samples is None:
            cfg.lr_decay_samples = cfg.train_samples
        lr_decay_steps = cfg.lr_decay_samples
        wd_incr_steps = cfg.train_samples
        if cfg.lr_warmup_fraction is not None:
            lr_warmup_steps = cfg.lr_warmup_fraction * lr_decay_steps
        else:
            lr_warmup_steps = cfg.lr_warmup_samples
    else:
        raise Exception("either train_iters or train_samples must be set")

    return OptimizerParamScheduler(
        optimizer,
        max_lr=cfg.lr, min_lr=cfg.min_lr,
        lr_warmup_steps=lr_warmup_steps,
        lr_decay_steps=lr_decay_steps,
        lr_decay_style=cfg.lr_decay_style,
        start_wd=cfg.start_weight_decay,
        end_wd=cfg.end_weight_decay,
        wd_incr_steps=wd_incr_steps,
        wd_incr_style=cfg.weight_decay_incr_style,
        use_checkpoint_opt_param_scheduler=cfg.use_checkpoint_opt_param_scheduler,
        override_opt_param_scheduler=cfg.override_opt_param_scheduler,
    )
