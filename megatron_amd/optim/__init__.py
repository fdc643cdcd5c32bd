"""Optimizer factory (reference megatron/optimizer/__init__.py:63-128)."""

from __future__ import annotations

import torch

from ..models.norms import LayerNorm, RMSNorm
from .adam import FusedAdam, FusedSGD
from .distrib_optimizer import DistributedOptimizer
from .grad_scaler import ConstantGradScaler, DynamicGradScaler
from .optimizer import FP32Optimizer, Float16OptimizerWithFloat16Params
from .scheduler import OptimizerParamScheduler


def _get_params_for_weight_decay_optimization(models, no_wd_decay_cond=None,
                                              scale_lr_cond=None,
                                              lr_mult=1.0):
    """Split params into (wd x lr_mult) groups (reference
    megatron/optimizer/__init__.py:28-60 + the training.py:320-350 cond
    hooks): default policy = no weight decay on biases and norm weights;
    `no_wd_decay_cond(name, param)` overrides it, `scale_lr_cond(name,
    param)` moves a param into a group whose lr is multiplied by lr_mult
    (used e.g. to train a task head faster than the backbone)."""
    groups = {
        (True, False): {"params": []},
        (False, False): {"params": [], "weight_decay": 0.0},
        (True, True): {"params": [], "lr_mult": lr_mult},
        (False, True): {"params": [], "weight_decay": 0.0,
                        "lr_mult": lr_mult},
    }
    for module in models:
        for mod_name, module_ in module.named_modules():
            is_norm = isinstance(
                module_, (LayerNorm, RMSNorm, torch.nn.LayerNorm)
            )
            for n, p in module_._parameters.items():
                if p is None:
                    continue
                name = f"{mod_name}.{n}" if mod_name else n
                if no_wd_decay_cond is not None:
                    no_wd = no_wd_decay_cond(name, p)
                else:
                    no_wd = is_norm or n == "bias"
                scale = bool(scale_lr_cond and scale_lr_cond(name, p))
                groups[(not no_wd, scale)]["params"].append(p)
    return tuple(groups.values())


def get_megatron_optimizer(models, cfg, no_wd_decay_cond=None,
                           scale_lr_cond=None, lr_mult=1.0):
    param_groups = [
        g
        for g in _get_params_for_weight_decay_optimization(
            models, no_wd_decay_cond, scale_lr_cond, lr_mult
        )
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
