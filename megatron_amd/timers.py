"""Named timers with device sync and cross-rank reporting
(reference megatron/timers.py:56-304). On ROCm the sync is
torch.cuda.synchronize() (HIP stream sync); log levels 0-2 gate overhead."""

from __future__ import annotations

import time

import torch


class _TimerBase:
    def __init__(self, name):
        self.name = name

    def start(self, barrier=False):
        raise NotImplementedError

    def stop(self, barrier=False):
        raise NotImplementedError

    def reset(self):
        raise NotImplementedError

    def elapsed(self, reset=True, barrier=False):
        raise NotImplementedError


class DummyTimer(_TimerBase):
    def __init__(self):
        super().__init__("dummy")

    def start(self, barrier=False):
        pass

    def stop(self, barrier=False):
        pass

    def reset(self):
        pass

    def elapsed(self, reset=True, barrier=False):
        raise Exception("dummy timer should not be used to calculate elapsed time")


class Timer(_TimerBase):
    def __init__(self, name):
        super().__init__(name)
        self._elapsed = 0.0
        self._started = False
        self._start_time = time.time()

    def _sync(self, barrier):
        if barrier and torch.distributed.is_initialized():
            torch.distributed.barrier()
        if torch.cuda.is_available():
            torch.cuda.synchronize()

    def start(self, barrier=False):
        assert not self._started, f"timer {self.name} has already been started"
        self._sync(barrier)
        self._start_time = time.time()
        self._started = True

    def stop(self, barrier=False):
        assert self._started, f"timer {self.name} is not started"
        self._sync(barrier)
        self._elapsed += time.time() - self._start_time
        self._started = False

    def reset(self):
        self._elapsed = 0.0
        self._started = False

    def elapsed(self, reset=True, barrier=False):
        started_ = self._started
        if self._started:
            self.stop(barrier=barrier)
        elapsed_ = self._elapsed
        if reset:
            self.reset()
        if started_:
            self.start(barrier=barrier)
        return elapsed_


class Timers:
    def __init__(self, log_level=0, log_option="minmax"):
        self._log_level = log_level
        self._log_option = log_option
        self._timers = {}
        self._log_levels = {}
        self._dummy_timer = DummyTimer()
        self._max_log_level = 2

    def __call__(self, name, log_level=None):
        if name in self._timers:
            if log_level is not None:
                assert log_level == self._log_levels[name]
            return self._timers[name]
        if log_level is None:
            log_level = self._max_log_level
        assert log_level <= self._max_log_level
        if log_level > self._log_level:
            return self._dummy_timer
        self._timers[name] = Timer(name)
        self._log_levels[name] = log_level
        return self._timers[name]

    def _get_elapsed_time_all_ranks(self, names, reset, barrier):
        world_size = torch.distributed.get_world_size()
        rank = torch.distributed.get_rank()
        device = "cuda" if torch.cuda.is_available() else "cpu"
        rank_name_to_time = torch.zeros(
            (world_size, len(names)), dtype=torch.float, device=device
        )
        for i, name in enumerate(names):
            if name in self._timers:
                rank_name_to_time[rank, i] = self._timers[name].elapsed(
                    reset=reset
                )
        if device == "cuda":
            torch.distributed.all_gather_into_tensor(
                rank_name_to_time.view(-1),
                rank_name_to_time[rank, :].view(-1),
            )
        else:
            gathered = [torch.zeros_like(rank_name_to_time[rank])
                        for _ in range(world_size)]
            torch.distributed.all_gather(gathered, rank_name_to_time[rank, :])
            rank_name_to_time = torch.stack(gathered)
        return rank_name_to_time

    def log(self, names, rank=None, normalizer=1.0, reset=True, barrier=False):
        if not torch.distributed.is_initialized():
            return
        assert normalizer > 0.0
        name_list = [n for n in names if n in self._timers]
        if not name_list:
            return
        rank_name_to_time = self._get_elapsed_time_all_ranks(
            name_list, reset, barrier
        )
        if self._log_option in ("max", "minmax"):
            string = "(min, max) time across ranks (ms):"
            for i, name in enumerate(name_list):
                t = rank_name_to_time[:, i] * 1000.0 / normalizer
                string += f"\n    {name}: ({t.min().item():.2f}, {t.max().item():.2f})"
        elif self._log_option == "all":
            string = "times across ranks (ms):"
            for i, name in enumerate(name_list):
                string += f"\n  {name}:"
                for r in range(rank_name_to_time.shape[0]):
                    string += (
                        f"\n    rank {r}: "
                        f"{rank_name_to_time[r, i].item() * 1000.0 / normalizer:.2f}"
                    )
        else:
            raise Exception(f"unknown timing log option {self._log_option}")

        if rank is None:
            rank = torch.distributed.get_world_size() - 1
        if rank == torch.distributed.get_rank():
            print(string, flush=True)

    def write(self, names, writer, iteration, normalizer=1.0, reset=False,
              barrier=False):
        assert normalizer > 0.0
        name_list = [n for n in names if n in self._timers]
        if not name_list or writer is None:
            return
        rank_name_to_time = self._get_elapsed_time_all_ranks(
            name_list, reset, barrier
        )
        for i, name in enumerate(name_list):
            value = rank_name_to_time[:, i].max().item() / normalizer
            writer.add_scalar(name + "-time", value, iteration)
