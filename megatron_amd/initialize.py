"""Runtime initialization: args, torch.distributed (RCCL), model-parallel
groups, RNG (reference megatron/initialize.py:26-275).

On ROCm the "nccl" backend IS RCCL; one process per GPU, device = local rank.
HSA_ENABLE_IPC_MODE_LEGACY=0 must be exported for cross-process dmabuf IPC on
the MI355X pool (host driver constraint).
"""

from __future__ import annotations

import os
import random
import time
from datetime import timedelta

import numpy as np
import torch

from . import global_state
from . import parallel as mpu
from .config import TrainingConfig, parse_config, set_config
from .microbatches import setup_microbatch_calculator


def initialize_megatron(extra_args_provider=None, args_defaults=None,
                        ignore_unknown_args=False, allow_no_cuda=False,
                        cfg: TrainingConfig = None):
    """Parse config, set up distributed + model parallel + RNG.

    Unlike the reference (initialize.py:36 asserts CUDA), CPU/gloo is allowed
    for tests and plumbing runs."""
    if cfg is None:
        cfg = parse_config(extra_args_provider, defaults=args_defaults or {})
    set_config(cfg)

    if cfg.load is not None and cfg.use_checkpoint_args:
        from .checkpointing import load_args_from_checkpoint

        load_args_from_checkpoint(cfg)
        cfg.finalize()

    global_state.set_global_variables(cfg)

    _initialize_distributed(cfg)
    _set_random_seed(cfg.seed, cfg.data_parallel_random_init)
    setup_microbatch_calculator(cfg)
    return cfg


def _initialize_distributed(cfg):
    """(reference initialize.py:124-167)"""
    device_count = torch.cuda.device_count() if torch.cuda.is_available() else 0

    if torch.distributed.is_initialized():
        cfg.rank = torch.distributed.get_rank()
        cfg.world_size = torch.distributed.get_world_size()
    else:
        if cfg.rank == 0:
            print("> initializing torch distributed ...", flush=True)
        if device_count > 0:
            device = cfg.rank % device_count
            if cfg.local_rank is not None:
                assert cfg.local_rank == device, (
                    "expected local-rank to be the same as rank % device-count"
                )
            else:
                cfg.local_rank = device
            torch.cuda.set_device(device)
        backend = cfg.distributed_backend
        if backend == "nccl" and device_count == 0:
            backend = "gloo"
        torch.distributed.init_process_group(
            backend=backend,
            world_size=cfg.world_size, rank=cfg.rank,
            timeout=timedelta(minutes=cfg.distributed_timeout_minutes),
        )

    if device_count > 0:
        if mpu.model_parallel_is_initialized():
            print("model parallel is already initialized")
            return
    if not mpu.model_parallel_is_initialized():
        mpu.initialize_model_parallel(
            cfg.tensor_model_parallel_size,
            cfg.pipeline_model_parallel_size,
            cfg.virtual_pipeline_model_parallel_size,
            cfg.pipeline_model_parallel_split_rank,
        )
        if cfg.rank == 0:
            print(
                f"> initialized tensor model parallel with size "
                f"{mpu.get_tensor_model_parallel_world_size()}"
            )
            print(
                f"> initialized pipeline model parallel with size "
                f"{mpu.get_pipeline_model_parallel_world_size()}"
            )


def _set_random_seed(seed_, data_parallel_random_init=False):
    """(reference initialize.py:179-193)"""
    if seed_ is None or seed_ < 0:
        raise ValueError(f"Seed ({seed_}) should be a positive integer.")
    seed = seed_ + (100 * mpu.get_pipeline_model_parallel_rank())
    if data_parallel_random_init:
        seed = seed + (10 * mpu.get_data_parallel_rank())
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.device_count() > 0:
        mpu.model_parallel_cuda_manual_seed(seed)
    else:
        mpu.model_parallel_cuda_manual_seed(seed)
