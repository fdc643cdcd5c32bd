"""Micro-batch calculator: constant or ramped global batch size
(reference megatron/microbatches.py:17-144).

The number of microbatches per global step is
global_batch_size / (micro_batch_size * data_parallel_size); with batch-size
rampup the global batch grows from `start` to the target in `increment`
steps spread evenly over `ramp_samples` consumed samples, and every
intermediate size must stay divisible by micro_batch * dp. On MI355X the
288 GB of HBM3E usually lets the micro-batch carry the whole per-GPU batch
(num_microbatches == 1) until pipeline parallelism needs in-flight
microbatches to fill the 1F1B schedule."""

from __future__ import annotations

from abc import ABC, abstractmethod


def build_num_microbatches_calculator(cfg):
    if cfg.rampup_batch_size is None:
        return ConstantNumMicroBatches(
            cfg.global_batch_size, cfg.micro_batch_size, cfg.data_parallel_size
        )
    assert len(cfg.rampup_batch_size) == 3
    start_batch_size, batch_size_increment, ramup_samples = map(
        int, cfg.rampup_batch_size
    )
    return RampupBatchsizeNumMicroBatches(
        start_batch_size, batch_size_increment, ramup_samples,
        cfg.global_batch_size, cfg.micro_batch_size, cfg.data_parallel_size,
    )


class NumMicroBatchesCalculator(ABC):
    def __init__(self):
        self.num_micro_batches = None
        self.current_global_batch_size = None

    def get(self):
        return self.num_micro_batches

    def get_current_global_batch_size(self):
        return self.current_global_batch_size

    @abstractmethod
    def update(self, consumed_samples, consistency_check):
        ...


class ConstantNumMicroBatches(NumMicroBatchesCalculator):
    def __init__(self, global_batch_size, micro_batch_size, data_parallel_size):
        super().__init__()
        micro_batch_times_data_parallel = micro_batch_size * data_parallel_size
        assert global_batch_size % micro_batch_times_data_parallel == 0
        self.num_micro_batches = global_batch_size // micro_batch_times_data_parallel
        assert self.num_micro_batches >= 1
        self.current_global_batch_size = global_batch_size

    def update(self, consumed_samples, consistency_check):
        pass


class RampupBatchsizeNumMicroBatches(NumMicroBatchesCalculator):
    def __init__(self, start_batch_size, batch_size_increment, ramup_samples,
                 global_batch_size, micro_batch_size, data_parallel_size):
        super().__init__()
        self.micro_batch_size = micro_batch_size
        self.data_parallel_size = data_parallel_size
        self.micro_batch_times_data_parallel_size = (
            micro_batch_size * data_parallel_size
        )
        assert self.micro_batch_times_data_parallel_size > 0
        assert start_batch_size > 0
        self.start_batch_size = start_batch_size
        assert global_batch_size > 0
        self.global_batch_size = global_batch_size
        diff_batch_size = self.global_batch_size - self.start_batch_size
        assert diff_batch_size >= 0
        assert batch_size_increment > 0
        self.batch_size_increment = batch_size_increment
        assert diff_batch_size % batch_size_increment == 0, (
            f"global batch - start batch ({diff_batch_size}) must be divisible "
            f"by the increment ({batch_size_increment})"
        )
        num_increments = diff_batch_size // self.batch_size_increment
        self.ramup_samples = ramup_samples
        assert self.ramup_samples >= 0
        self.rampup_samples_per_increment = self.ramup_samples / num_increments

        self.update(0, False)

    def update(self, consumed_samples, consistency_check):
        if consumed_samples > self.ramup_samples:
            self.current_global_batch_size = self.global_batch_size
        else:
            steps = int(consumed_samples / self.rampup_samples_per_increment)
            self.current_global_batch_size = (
                self.start_batch_size + steps * self.batch_size_increment
            )
            assert self.current_global_batch_size <= self.global_batch_size
        if consistency_check:
            assert (
                self.current_global_batch_size
                % self.micro_batch_times_data_parallel_size
                == 0
            )
        self.num_micro_batches = (
            self.current_global_batch_size
            // self.micro_batch_times_data_parallel_size
        )


# module-level instance managed by global_state
_CALCULATOR = None


def setup_microbatch_calculator(cfg):
    global _CALCULATOR
    _CALCULATOR = build_num_microbatches_calculator(cfg)


def get_num_microbatches():
    assert _CALCULATOR is not None
    return _CALCULATOR.get()


def get_current_global_batch_size():
    assert _CALCULATOR is not None
    return _CALCULATOR.get_current_global_batch_size()


def update_num_microbatches(consumed_samples, consistency_check=True):
    assert _CALCULATOR is not None
    _CALCULATOR.update(consumed_samples, consistency_check)
