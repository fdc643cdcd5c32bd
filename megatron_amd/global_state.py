"""Process-wide runtime state (tokenizer, timers, writers, counters).

The reference keeps everything in megatron/global_vars.py:14-21; here the
config is an explicit object (megatron_amd/config.py) and only the genuinely
process-wide services live in this registry.
"""

from __future__ import annotations

from typing import Optional

from .timers import Timers

_TOKENIZER = None
_TENSORBOARD_WRITER = None
_WANDB_WRITER = None
_TIMERS: Optional[Timers] = None
_SIGNAL_HANDLER = None
_TOKENS = 0


def set_global_variables(cfg):
    global _TIMERS
    if cfg.tokenizer_type is not None:
        _build_tokenizer(cfg)
    _set_tensorboard_writer(cfg)
    _TIMERS = Timers(cfg.timing_log_level, cfg.timing_log_option)
    if cfg.exit_signal_handler:
        _set_signal_handler()


def _build_tokenizer(cfg):
    global _TOKENIZER
    from .tokenizer import build_tokenizer

    _TOKENIZER = build_tokenizer(cfg)
    return _TOKENIZER


def rebuild_tokenizer(cfg):
    global _TOKENIZER
    _TOKENIZER = None
    return _build_tokenizer(cfg)


def get_tokenizer():
    assert _TOKENIZER is not None, "tokenizer is not initialized"
    return _TOKENIZER


def set_tokenizer(tokenizer):
    global _TOKENIZER
    _TOKENIZER = tokenizer


def _set_tensorboard_writer(cfg):
    global _TENSORBOARD_WRITER, _WANDB_WRITER
    import torch

    if cfg.tensorboard_dir and cfg.rank == (cfg.world_size - 1):
        try:
            from torch.utils.tensorboard import SummaryWriter

            _TENSORBOARD_WRITER = SummaryWriter(
                log_dir=cfg.tensorboard_dir,
                max_queue=cfg.tensorboard_queue_size,
            )
        except ModuleNotFoundError:
            print("WARNING: tensorboard not available", flush=True)
    if cfg.wandb_logger and cfg.rank == (cfg.world_size - 1):
        from .wandb_logger import WandbTBShim, WandBConfig

        _WANDB_WRITER = WandbTBShim(WandBConfig.from_config(cfg))


def get_tensorboard_writer():
    return _TENSORBOARD_WRITER


def get_wandb_writer():
    return _WANDB_WRITER


def get_timers() -> Timers:
    assert _TIMERS is not None, "timers are not initialized"
    return _TIMERS


def init_timers(log_level=0, log_option="minmax"):
    global _TIMERS
    _TIMERS = Timers(log_level, log_option)
    return _TIMERS


def _set_signal_handler():
    global _SIGNAL_HANDLER
    from .dist_signal_handler import DistributedSignalHandler

    _SIGNAL_HANDLER = DistributedSignalHandler().__enter__()


def get_signal_handler():
    return _SIGNAL_HANDLER


def update_num_tokens(n):
    global _TOKENS
    _TOKENS += int(n)


def get_num_tokens():
    return _TOKENS


def set_num_tokens(n):
    global _TOKENS
    _TOKENS = int(n)
