"""Checkpoint save/load with the reference's on-disk layout.

Reference: megatron/checkpointing.py:77-740. Layout preserved exactly so
checkpoints interoperate with the reference tooling:

  <save>/iter_NNNNNNN/mp_rank_{tp:02d}[_{pp:03d}]/model_optim_rng.pt
  <save>/latest_checkpointed_iteration.txt          (tracker; or 'release')
  distributed optimizer: model_rng.pt + per-DP-rank optim.pt
  checkpoint_version 3.0
"""

from __future__ import annotations

import os
import random

import numpy as np
import torch

from . import parallel as mpu
from .config import get_config
from .utils import print_rank_0, unwrap_model

_CHECKPOINT_VERSION = 3.0


def get_checkpoint_version():
    return _CHECKPOINT_VERSION


def check_checkpoint_args(checkpoint_args, cfg):
    def _compare(arg_name):
        checkpoint_value = (
            checkpoint_args.get(arg_name)
            if isinstance(checkpoint_args, dict)
            else getattr(checkpoint_args, arg_name, None)
        )
        args_value = getattr(cfg, arg_name)
        error_message = (
            f"{arg_name} value from checkpoint ({checkpoint_value}) is not "
            f"equal to the input argument value ({args_value})."
        )
        assert checkpoint_value == args_value, error_message

    for name in ("num_layers", "hidden_size", "num_attention_heads"):
        _compare(name)


def ensure_directory_exists(filename):
    dirname = os.path.dirname(filename)
    os.makedirs(dirname, exist_ok=True)


def get_checkpoint_name(checkpoints_path, iteration, release=False,
                        use_distributed_optimizer=False, model_only=False):
    """(reference checkpointing.py:77-140)"""
    if release:
        directory = "release"
    else:
        directory = f"iter_{iteration:07d}"
    if mpu.get_pipeline_model_parallel_world_size() == 1:
        common_path = os.path.join(
            checkpoints_path, directory,
            f"mp_rank_{mpu.get_tensor_model_parallel_rank():02d}",
        )
    else:
        common_path = os.path.join(
            checkpoints_path, directory,
            f"mp_rank_{mpu.get_tensor_model_parallel_rank():02d}"
            f"_{mpu.get_pipeline_model_parallel_rank():03d}",
        )
    if use_distributed_optimizer:
        model_name = os.path.join(common_path, "model_rng.pt")
        optim_name = os.path.join(
            common_path + f"_{mpu.get_data_parallel_rank():03d}", "optim.pt"
        )
    else:
        model_name = optim_name = os.path.join(common_path, "model_optim_rng.pt")
    if model_only:
        return model_name
    return model_name, optim_name


def get_checkpoint_tracker_filename(checkpoints_path):
    return os.path.join(checkpoints_path, "latest_checkpointed_iteration.txt")


def read_metadata(tracker_filename):
    iteration = 0
    release = False
    with open(tracker_filename, "r") as f:
        metastring = f.read().strip()
        try:
            iteration = int(metastring)
        except ValueError:
            release = metastring == "release"
            if not release:
                raise ValueError(f"Invalid metadata file {tracker_filename}")
    assert iteration > 0 or release, "error parsing metadata file"

    if torch.distributed.is_initialized():
        iters_cuda = torch.tensor(
            [iteration], dtype=torch.long,
            device="cuda" if torch.cuda.is_available() else "cpu",
        )
        torch.distributed.all_reduce(iters_cuda,
                                     op=torch.distributed.ReduceOp.MAX)
        max_iter = iters_cuda[0].item()
        assert iteration == max_iter, (
            f"iteration mismatch across ranks: {iteration} vs max {max_iter}"
        )
    return iteration, release


def get_rng_state(cfg):
    """Collect rng state across DP group (reference checkpointing.py:217-240)."""
    rng_state = {
        "random_rng_state": random.getstate(),
        "np_rng_state": np.random.get_state(),
        "torch_rng_state": torch.get_rng_state(),
        "cuda_rng_state": (
            torch.cuda.get_rng_state() if torch.cuda.is_available() else None
        ),
        "rng_tracker_states": mpu.get_cuda_rng_tracker().get_states(),
    }
    rng_state_list = None
    if (
        torch.distributed.is_initialized()
        and mpu.get_data_parallel_world_size() > 1
        and cfg.data_parallel_random_init
    ):
        rng_state_list = [None for _ in range(mpu.get_data_parallel_world_size())]
        torch.distributed.all_gather_object(
            rng_state_list, rng_state, group=mpu.get_data_parallel_group()
        )
    else:
        rng_state_list = [rng_state]
    return rng_state_list


def save_checkpoint(iteration, model, optimizer, opt_param_scheduler, cfg=None):
    """(reference checkpointing.py:243-337)"""
    if cfg is None:
        cfg = get_config()
    print_rank_0(
        f"saving checkpoint at iteration {iteration:7d} to {cfg.save}"
    )

    model = unwrap_model(model)
    rng_state = get_rng_state(cfg)

    model_name, optim_name = get_checkpoint_name(
        cfg.save, iteration,
        use_distributed_optimizer=cfg.use_distributed_optimizer,
    )

    # model state (DP rank 0 only)
    if (
        not torch.distributed.is_initialized()
        or mpu.get_data_parallel_rank() == 0
    ):
        state_dict = {}
        state_dict["args"] = _config_to_namespace(cfg)
        state_dict["checkpoint_version"] = _CHECKPOINT_VERSION
        state_dict["iteration"] = iteration
        if len(model) == 1:
            state_dict["model"] = model[0].state_dict_for_save_checkpoint()
        else:
            for i in range(len(model)):
                mpu.set_virtual_pipeline_model_parallel_rank(i)
                state_dict[f"model{i}"] = (
                    model[i].state_dict_for_save_checkpoint()
                )
        if not cfg.no_save_optim and not cfg.use_distributed_optimizer:
            if optimizer is not None:
                state_dict["optimizer"] = optimizer.state_dict()
            if opt_param_scheduler is not None:
                state_dict["opt_param_scheduler"] = (
                    opt_param_scheduler.state_dict()
                )
        if not cfg.no_save_rng:
            state_dict["rng_state"] = rng_state
        ensure_directory_exists(model_name)
        torch.save(state_dict, model_name)

    # distributed optimizer: every DP rank saves its shard
    if cfg.use_distributed_optimizer and not cfg.no_save_optim and (
        optimizer is not None
    ):
        optim_state = {
            "optimizer": optimizer.state_dict(),
            "opt_param_scheduler": (
                opt_param_scheduler.state_dict()
                if opt_param_scheduler is not None else None
            ),
        }
        ensure_directory_exists(optim_name)
        torch.save(optim_state, optim_name)

    if torch.distributed.is_initialized():
        torch.distributed.barrier()
    print_rank_0(
        f"  successfully saved checkpoint at iteration {iteration:7d} "
        f"to {cfg.save}"
    )
    if (
        not torch.distributed.is_initialized()
        or torch.distributed.get_rank() == 0
    ):
        tracker_filename = get_checkpoint_tracker_filename(cfg.save)
        with open(tracker_filename, "w") as f:
            f.write(str(iteration))
    if torch.distributed.is_initialized():
        torch.distributed.barrier()


def _config_to_namespace(cfg):
    import argparse
    import dataclasses

    ns = argparse.Namespace()
    for f in dataclasses.fields(cfg):
        setattr(ns, f.name, getattr(cfg, f.name))
    return ns


def _transpose_first_dim(t, num_splits, num_splits_first, model):
    """QKV ordering fixup for old checkpoint versions
    (reference checkpointing.py:340-377)."""
    input_shape = t.size()
    if num_splits_first:
        intermediate_shape = (
            (num_splits, -1, input_shape[0] // num_splits) + input_shape[1:]
        )
        t = t.view(*intermediate_shape)
        t = t.transpose(0, 1).contiguous()
    else:
        intermediate_shape = (
            (-1, num_splits, input_shape[0] // num_splits) + input_shape[1:]
        )
        t = t.view(*intermediate_shape)
        t = t.transpose(0, 1).contiguous()
    t = t.view(*input_shape)
    return t


def fix_query_key_value_ordering(model, checkpoint_version):
    if checkpoint_version < 2.0:
        raise NotImplementedError(
            "checkpoints with version < 2.0 are not supported"
        )


def load_args_from_checkpoint(cfg):
    """Read args stored in the checkpoint (reference checkpointing.py:482-567)."""
    load_dir = cfg.load
    tracker_filename = get_checkpoint_tracker_filename(load_dir)
    if not os.path.isfile(tracker_filename):
        print_rank_0(
            f"WARNING: could not find checkpoint metadata file {tracker_filename}"
        )
        return cfg
    iteration, release = read_metadata(tracker_filename)
    model_name = get_checkpoint_name(load_dir, iteration, release,
                                     model_only=True)
    state_dict = torch.load(model_name, map_location="cpu",
                            weights_only=False)
    if "args" not in state_dict:
        return cfg
    checkpoint_args = state_dict["args"]
    for key in (
        "num_layers", "hidden_size", "ffn_hidden_size", "num_attention_heads",
        "num_attention_heads_kv", "kv_channels", "max_position_embeddings",
        "make_vocab_size_divisible_by", "padded_vocab_size", "use_bias",
        "use_rms_norm", "use_post_ln", "glu_activation",
        "position_embedding_type", "rope_theta", "rope_scaling_factor",
        "tie_embed_logits", "parallel_attn", "parallel_layernorm",
        "sliding_window_size", "layernorm_epsilon", "tokenizer_type",
    ):
        if hasattr(checkpoint_args, key):
            setattr(cfg, key, getattr(checkpoint_args, key))
    return cfg


def load_checkpoint(model, optimizer, opt_param_scheduler, cfg=None,
                    load_arg="load", strict=True):
    """(reference checkpointing.py:570-696)"""
    if cfg is None:
        cfg = get_config()
    load_dir = getattr(cfg, load_arg)

    model = unwrap_model(model)

    tracker_filename = get_checkpoint_tracker_filename(load_dir)
    if not os.path.isfile(tracker_filename):
        if getattr(cfg, "exit_on_missing_checkpoint", False):
            raise FileNotFoundError(
                f"--exit_on_missing_checkpoint: no metadata file "
                f"{tracker_filename}"
            )
        print_rank_0(
            f"WARNING: could not find the metadata file {tracker_filename}; "
            "will not load any checkpoints and will start from random"
        )
        return 0
    iteration, release = read_metadata(tracker_filename)
    if getattr(cfg, "load_iters", None):
        # load a specific saved iteration instead of the tracker's latest
        iteration, release = int(cfg.load_iters), False

    if cfg.use_distributed_optimizer:
        model_name, optim_name = get_checkpoint_name(
            load_dir, iteration, release, use_distributed_optimizer=True
        )
        print_rank_0(f" loading checkpoint from {load_dir} at iteration {iteration}")
        state_dict = torch.load(model_name, map_location="cpu",
                                weights_only=False)
        optim_state_dict = None
        if not release and not cfg.finetune and not cfg.no_load_optim:
            if os.path.isfile(optim_name):
                optim_state_dict = torch.load(optim_name, map_location="cpu",
                                              weights_only=False)
    else:
        model_name, _ = get_checkpoint_name(load_dir, iteration, release)
        print_rank_0(f" loading checkpoint from {load_dir} at iteration {iteration}")
        state_dict = torch.load(model_name, map_location="cpu",
                                weights_only=False)
        optim_state_dict = state_dict

    checkpoint_version = state_dict.get("checkpoint_version", 0)

    if cfg.finetune or release:
        iteration = 0
    else:
        if "iteration" in state_dict:
            iteration = state_dict["iteration"]
        else:
            raise KeyError("iteration not found in checkpoint")

    if "args" in state_dict and not cfg.finetune:
        check_checkpoint_args(state_dict["args"], cfg)
        cfg.consumed_train_samples = getattr(
            state_dict["args"], "consumed_train_samples", 0
        )
        cfg.consumed_valid_samples = getattr(
            state_dict["args"], "consumed_valid_samples", 0
        )

    # model
    if len(model) == 1:
        model[0].load_state_dict(state_dict["model"], strict=strict)
    else:
        for i in range(len(model)):
            mpu.set_virtual_pipeline_model_parallel_rank(i)
            model[i].load_state_dict(state_dict[f"model{i}"], strict=strict)

    fix_query_key_value_ordering(model, checkpoint_version)

    # optimizer
    if not release and not cfg.finetune and not cfg.no_load_optim:
        try:
            if optimizer is not None and optim_state_dict is not None:
                optimizer.load_state_dict(optim_state_dict["optimizer"])
            if (
                opt_param_scheduler is not None
                and optim_state_dict is not None
                and optim_state_dict.get("opt_param_scheduler") is not None
            ):
                opt_param_scheduler.load_state_dict(
                    optim_state_dict["opt_param_scheduler"]
                )
        except KeyError:
            print_rank_0(
                "Unable to load optimizer from checkpoint; continuing "
                "without restoring optimizer state"
            )

    # rng
    if not release and not cfg.finetune and not cfg.no_load_rng:
        try:
            if "rng_state" in state_dict:
                if cfg.data_parallel_random_init:
                    rng_state = state_dict["rng_state"][
                        mpu.get_data_parallel_rank()
                    ]
                else:
                    rng_state = state_dict["rng_state"][0]
                random.setstate(rng_state["random_rng_state"])
                np.random.set_state(rng_state["np_rng_state"])
                torch.set_rng_state(rng_state["torch_rng_state"])
                if torch.cuda.is_available() and rng_state["cuda_rng_state"] is not None:
                    torch.cuda.set_rng_state(rng_state["cuda_rng_state"])
                if rng_state["rng_tracker_states"]:
                    mpu.get_cuda_rng_tracker().set_states(
                        rng_state["rng_tracker_states"]
                    )
        except KeyError:
            print_rank_0("Unable to load rng state from checkpoint")

    if torch.distributed.is_initialized():
        torch.distributed.barrier()
    print_rank_0(
        f"  successfully loaded checkpoint from {load_dir} "
        f"at iteration {iteration}"
    )
    return iteration
