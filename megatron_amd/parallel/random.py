"""Per-region RNG state tracking + activation recompute.

Reference: megatron/core/tensor_parallel/random.py:64-252. TP ranks need
*different* dropout seeds inside model-parallel regions (activations are
sharded) but *identical* seeds outside (residual stream is replicated). We keep
named device RNG states and fork into them with a context manager.

On ROCm the device generator is the HIP philox generator exposed through
torch.cuda.*_rng_state — same API as CUDA. On CPU (unit tests, gloo) we track
torch's CPU generator instead so the fork semantics are testable without a GPU.
"""

from __future__ import annotations

import contextlib
from typing import Dict

import torch

from . import state as ps
from .utils import (
    divide,
    gather_split_1d_tensor,
    safely_set_viewless_tensor_data,
    split_tensor_into_1d_equal_chunks,
)

_MODEL_PARALLEL_RNG_TRACKER_NAME = "model-parallel-rng"

_ON_DEVICE = torch.cuda.is_available


def _get_state():
    if torch.cuda.is_available():
        return torch.cuda.get_rng_state()
    return torch.get_rng_state()


def _set_state(state):
    if torch.cuda.is_available():
        torch.cuda.set_rng_state(state)
    else:
        torch.set_rng_state(state)


def _manual_seed(seed):
    if torch.cuda.is_available():
        torch.cuda.manual_seed(seed)
    else:
        torch.manual_seed(seed)


class DeviceRNGStatesTracker:
    """Named device RNG states with fork() (reference CudaRNGStatesTracker,
    random.py:64-132)."""

    def __init__(self):
        self.states_: Dict[str, torch.Tensor] = {}
        self.seeds_ = set()

    def reset(self):
        self.states_ = {}
        self.seeds_ = set()

    def get_states(self):
        return dict(self.states_)

    def set_states(self, states):
        self.states_ = states

    def add(self, name, seed):
        if seed in self.seeds_:
            raise Exception(f"seed {seed} already present")
        self.seeds_.add(seed)
        if name in self.states_:
            raise Exception(f"rng state {name} already present")
        orig = _get_state()
        _manual_seed(seed)
        self.states_[name] = _get_state()
        _set_state(orig)

    @contextlib.contextmanager
    def fork(self, name=_MODEL_PARALLEL_RNG_TRACKER_NAME):
        if name not in self.states_:
            raise Exception(f"rng state {name} is not added")
        orig = _get_state()
        _set_state(self.states_[name])
        try:
            yield
        finally:
            self.states_[name] = _get_state()
            _set_state(orig)


_RNG_STATE_TRACKER = DeviceRNGStatesTracker()


def get_cuda_rng_tracker() -> DeviceRNGStatesTracker:
    return _RNG_STATE_TRACKER


def model_parallel_cuda_manual_seed(seed: int) -> None:
    """Seed: default state = seed + pp_rank (same across TP group);
    model-parallel state offset by 2718 + tp_rank (reference random.py:144-172)."""
    offset = seed + 2718
    tensor_model_parallel_seed = offset + ps.get_tensor_model_parallel_rank()
    data_parallel_seed = seed

    _RNG_STATE_TRACKER.reset()
    _manual_seed(data_parallel_seed)
    _RNG_STATE_TRACKER.add(_MODEL_PARALLEL_RNG_TRACKER_NAME, tensor_model_parallel_seed)


# ---------------------------------------------------------------------------
# activation recompute


class CheckpointFunction(torch.autograd.Function):
    """Activation checkpointing with RNG save/restore and optional sharding of
    the saved input across the TP group (reference random.py:175-245)."""

    @staticmethod
    def forward(ctx, run_function, distribute_saved_activations, *args):
        ctx.run_function = run_function
        ctx.distribute_saved_activations = distribute_saved_activations

        ctx.fwd_cpu_rng_state = torch.get_rng_state()
        ctx.fwd_device_rng_state = _get_state()
        ctx.fwd_rng_tracker_states = _RNG_STATE_TRACKER.get_states()

        with torch.no_grad():
            outputs = run_function(*args)

        if distribute_saved_activations:
            ctx.input_0_shape = args[0].data.shape
            safely_set_viewless_tensor_data(
                args[0],
                split_tensor_into_1d_equal_chunks(args[0].data, new_buffer=True),
            )

        ctx.save_for_backward(*args)
        return outputs

    @staticmethod
    def backward(ctx, *grads):
        if not torch.autograd._is_checkpoint_valid():
            raise RuntimeError("checkpointing is not compatible with .grad()")
        inputs = ctx.saved_tensors
        if ctx.distribute_saved_activations:
            safely_set_viewless_tensor_data(
                inputs[0],
                gather_split_1d_tensor(inputs[0].data).view(ctx.input_0_shape),
            )

        bwd_cpu_rng_state = torch.get_rng_state()
        bwd_device_rng_state = _get_state()
        bwd_rng_tracker_states = _RNG_STATE_TRACKER.get_states()

        torch.set_rng_state(ctx.fwd_cpu_rng_state)
        _set_state(ctx.fwd_device_rng_state)
        _RNG_STATE_TRACKER.set_states(ctx.fwd_rng_tracker_states)

        detached_inputs = tuple(
            inp.detach().requires_grad_(inp.requires_grad) for inp in inputs
        )
        with torch.enable_grad():
            outputs = ctx.run_function(*detached_inputs)

        torch.set_rng_state(bwd_cpu_rng_state)
        _set_state(bwd_device_rng_state)
        _RNG_STATE_TRACKER.set_states(bwd_rng_tracker_states)

        if isinstance(outputs, torch.Tensor):
            outputs = (outputs,)
        elif isinstance(outputs, tuple):
            outputs = tuple(o for o in outputs if torch.is_tensor(o))
        torch.autograd.backward(outputs, grads[: len(outputs)])
        grads = tuple(
            inp.grad if isinstance(inp, torch.Tensor) else inp
            for inp in detached_inputs
        )
        return (None, None) + grads


def checkpoint(function, distribute_saved_activations, *args):
    return CheckpointFunction.apply(function, distribute_saved_activations, *args)
