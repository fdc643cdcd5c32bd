"""LR / weight-decay scheduler (reference megatron/optimizer_param_scheduler.py:10-228)."""

from __future__ import annotations

import math


class OptimizerParamScheduler:
    def __init__(self, optimizer, max_lr, min_lr, lr_warmup_steps, lr_decay_steps,
                 lr_decay_style, start_wd, end_wd, wd_incr_steps, wd_incr_style,
                 use_checkpoint_opt_param_scheduler=True,
                 override_opt_param_scheduler=False):
        self.optimizer = optimizer
        self.max_lr = float(max_lr)
        self.min_lr = min_lr
        assert self.min_lr >= 0.0 and self.max_lr >= self.min_lr
        self.lr_warmup_steps = lr_warmup_steps
        self.num_steps = 0
        self.lr_decay_steps = lr_decay_steps
        assert self.lr_decay_steps > 0
        assert self.lr_warmup_steps < self.lr_decay_steps
        self.lr_decay_style = lr_decay_style

        self.start_wd = start_wd
        self.end_wd = end_wd
        assert self.start_wd >= 0.0 and self.end_wd >= self.start_wd
        self.wd_incr_steps = wd_incr_steps
        self.wd_incr_style = wd_incr_style

        self.override_opt_param_scheduler = override_opt_param_scheduler
        self.use_checkpoint_opt_param_scheduler = use_checkpoint_opt_param_scheduler
        if self.override_opt_param_scheduler:
            assert not self.use_checkpoint_opt_param_scheduler

        self.step(0)

    def get_wd(self):
        if self.num_steps > self.wd_incr_steps:
            return self.end_wd
        if self.wd_incr_style == "constant":
            assert self.start_wd == self.end_wd
            return self.end_wd
        incr_ratio = float(self.num_steps) / float(self.wd_incr_steps)
        assert 0.0 <= incr_ratio <= 1.0
        delta_wd = self.end_wd - self.start_wd
        if self.wd_incr_style == "linear":
            coeff = incr_ratio
        elif self.wd_incr_style == "cosine":
            coeff = 0.5 * (math.cos(math.pi * (1 - incr_ratio)) + 1.0)
        else:
            raise Exception(f"{self.wd_incr_style} not supported")
        return self.start_wd + coeff * delta_wd

    def get_lr(self):
        # warmup
        if self.lr_warmup_steps > 0 and self.num_steps <= self.lr_warmup_steps:
            return self.max_lr * float(self.num_steps) / float(self.lr_warmup_steps)
        if self.lr_decay_style == "constant":
            return self.max_lr
        if self.num_steps > self.lr_decay_steps:
            return self.min_lr

        if self.lr_decay_style == "inverse-square-root":
            warmup_steps = max(self.lr_warmup_steps, 1)
            num_steps = max(self.num_steps, 1)
            lr = self.max_lr * warmup_steps ** 0.5 / (num_steps ** 0.5)
            return max(self.min_lr, lr)

        num_steps_ = self.num_steps - self.lr_warmup_steps
        decay_steps_ = self.lr_decay_steps - self.lr_warmup_steps
        decay_ratio = float(num_steps_) / float(decay_steps_)
        assert 0.0 <= decay_ratio <= 1.0
        delta_lr = self.max_lr - self.min_lr
        if self.lr_decay_style == "linear":
            coeff = 1.0 - decay_ratio
        elif self.lr_decay_style == "cosine":
            coeff = 0.5 * (math.cos(math.pi * decay_ratio) + 1.0)
        else:
            raise Exception(f"{self.lr_decay_style} not supported")
        return self.min_lr + coeff * delta_lr

    def step(self, increment=1):
        self.num_steps += increment
        new_lr = self.get_lr()
        new_wd = self.get_wd()
        for group in self.optimizer.param_groups:
            new_lr_mult = group.get("lr_mult", 1.0)
            wd_mult = group.get("wd_mult", 1.0)
            group["lr"] = new_lr * new_lr_mult
            group["weight_decay"] = new_wd * wd_mult

    def state_dict(self):
        return {
            "max_lr": self.max_lr,
            "lr_warmup_steps": self.lr_warmup_steps,
            "num_steps": self.num_steps,
            "lr_decay_style": self.lr_decay_style,
            "lr_decay_steps": self.lr_decay_steps,
            "min_lr": self.min_lr,
            "start_wd": self.start_wd,
            "end_wd": self.end_wd,
            "wd_incr_style": self.wd_incr_style,
            "wd_incr_steps": self.wd_incr_steps,
        }

    def _check_and_set(self, cls_value, sd_value, name):
        if self.override_opt_param_scheduler:
            return cls_value
        if not self.use_checkpoint_opt_param_scheduler:
            assert cls_value == sd_value, (
                f"OptimizerParamScheduler: {name} mismatch "
                f"(class {cls_value} vs checkpoint {sd_value})"
            )
        return sd_value

    def load_state_dict(self, sd):
        self.max_lr = self._check_and_set(self.max_lr, sd["max_lr"], "lr")
        self.min_lr = self._check_and_set(self.min_lr, sd["min_lr"], "min lr")
        self.lr_warmup_steps = self._check_and_set(
            self.lr_warmup_steps, sd["lr_warmup_steps"], "warmup steps"
        )
        self.lr_decay_steps = self._check_and_set(
            self.lr_decay_steps, sd["lr_decay_steps"], "decay steps"
        )
        self.lr_decay_style = self._check_and_set(
            self.lr_decay_style, sd["lr_decay_style"], "decay style"
        )
        num_steps = sd["num_steps"]
        self.step(increment=num_steps - self.num_steps)
        if "start_wd" in sd:
            self.start_wd = self._check_and_set(self.start_wd, sd["start_wd"],
                                                "start wd")
            self.end_wd = self._check_and_set(self.end_wd, sd["end_wd"], "end wd")
            self.wd_incr_steps = self._check_and_set(
                self.wd_incr_steps, sd["wd_incr_steps"], "wd incr steps"
            )
            self.wd_incr_style = self._check_and_set(
                self.wd_incr_style, sd["wd_incr_style"], "wd incr style"
            )
