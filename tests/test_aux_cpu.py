"""Auxiliary subsystems: timers, metrics, the wandb TB shim, and the
distributed SIGTERM consensus handler (reference tests/test_wandb.py +
SURVEY §5 coverage)."""

import os
import signal
import sys
import time
from unittest import mock

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def test_timers_basic(dist_single):
    from megatron_amd.timers import DummyTimer, Timers

    timers = Timers(log_level=1, log_option="minmax")
    t = timers("phase-a", log_level=1)
    t.start()
    time.sleep(0.01)
    t.stop()
    assert timers("phase-a").elapsed(reset=False) >= 0.01

    # above the configured log level -> dummy (zero-cost, no state)
    d = timers("too-detailed", log_level=2)
    assert isinstance(d, DummyTimer)
    d.start()
    d.stop()

    # same name must keep its level
    again = timers("phase-a", log_level=1)
    assert again is not None

    # log() runs the all-rank gather path
    timers.log(["phase-a"])

    class _Writer:
        def __init__(self):
            self.scalars = {}

        def add_scalar(self, name, value, step):
            self.scalars[name] = (value, step)

    timers("phase-b", log_level=0).start()
    timers("phase-b").stop()
    w = _Writer()
    timers.write(["phase-b"], w, iteration=3)
    assert any("phase-b" in k for k in w.scalars)


def test_timer_double_start_raises(dist_single):
    from megatron_amd.timers import Timers

    timers = Timers(log_level=2)
    timers("x").start()
    with pytest.raises(AssertionError):
        timers("x").start()
    timers("x").stop()


def test_metrics(dist_single):
    from megatron_amd import parallel as mpu
    from megatron_amd.metrics import METRICS, MetricInput

    if not mpu.model_parallel_is_initialized():
        mpu.initialize_model_parallel(1, 1)
    s, b, v = 4, 2, 8
    logits = torch.zeros(s, b, v)
    labels = torch.zeros(b, s, dtype=torch.long)
    logits[..., 3] = 5.0  # argmax = 3 everywhere
    labels[:, :2] = 3  # half the positions correct
    inp = MetricInput(
        batch={"labels": labels, "loss_mask": torch.ones(b, s)},
        logits=logits, loss=torch.tensor(0.5),
    )
    assert abs(METRICS["perplexity"](inp)["perplexity"]
               - torch.exp(torch.tensor(0.5)).item()) < 1e-6
    assert METRICS["accuracy"](inp)["accuracy"] == 0.5
    assert METRICS["count_loss_mask"](inp)["count_loss_mask"] == b * s


def test_wandb_shim_batches_by_step():
    from megatron_amd.wandb_logger import WandBConfig, WandbTBShim

    fake_wandb = mock.MagicMock()
    fake_run = mock.MagicMock()
    fake_wandb.init.return_value = fake_run
    with mock.patch.dict(sys.modules, {"wandb": fake_wandb}):
        shim = WandbTBShim(WandBConfig(project="p", name="n"))
        shim.add_scalar("loss", 1.0, 1)
        shim.add_scalar("lr", 0.1, 1)
        fake_wandb.log.assert_not_called()  # same step: buffered
        shim.add_scalar("loss", 0.9, 2)  # new step flushes step 1
        fake_wandb.log.assert_called_once_with(
            {"loss": 1.0, "lr": 0.1}, step=1
        )
        shim.flush_all()
        assert fake_wandb.log.call_count == 2


def test_signal_handler_consensus(dist_single):
    from megatron_amd.dist_signal_handler import DistributedSignalHandler

    with DistributedSignalHandler(sig=signal.SIGUSR1) as handler:
        assert handler.signals_received() == [False]
        os.kill(os.getpid(), signal.SIGUSR1)
        # the handler records the signal; consensus gathers across ranks
        assert handler.signals_received() == [True]
    # original handler restored on exit
    assert signal.getsignal(signal.SIGUSR1) not in (None,)


def _make_sched(style, warmup=10, decay=100, max_lr=1.0, min_lr=0.1):
    import torch as t

    from megatron_amd.optim.scheduler import OptimizerParamScheduler

    p = t.nn.Parameter(t.zeros(1))
    opt = t.optim.SGD([p], lr=max_lr)
    return OptimizerParamScheduler(
        opt, max_lr, min_lr, warmup, decay, style,
        start_wd=0.1, end_wd=0.1, wd_incr_steps=decay,
        wd_incr_style="constant",
    )


def test_lr_schedules():
    import math as _math

    # linear warmup reaches max_lr at warmup end
    s = _make_sched("linear")
    s.step(increment=10)
    assert abs(s.get_lr() - 1.0) < 1e-9
    # linear decay hits min_lr at decay end
    s.step(increment=90)
    assert abs(s.get_lr() - 0.1) < 1e-9

    # cosine: halfway through decay = midpoint of (max, min)
    s = _make_sched("cosine")
    s.step(increment=10 + 45)  # halfway through the 90 decay steps
    expected = 0.1 + (1.0 - 0.1) * 0.5 * (1 + _math.cos(_math.pi * 0.5))
    assert abs(s.get_lr() - expected) < 1e-6

    # constant stays at max after warmup
    s = _make_sched("constant", min_lr=0.0)
    s.step(increment=50)
    assert abs(s.get_lr() - 1.0) < 1e-9

    # state roundtrip
    s = _make_sched("cosine")
    s.step(increment=30)
    sd = s.state_dict()
    s2 = _make_sched("cosine")
    s2.load_state_dict(sd)
    assert abs(s.get_lr() - s2.get_lr()) < 1e-12


def test_dynamic_grad_scaler():
    from megatron_amd.optim.grad_scaler import DynamicGradScaler

    sc = DynamicGradScaler(
        initial_scale=2.0 ** 10, min_scale=1.0, growth_factor=2.0,
        backoff_factor=0.5, growth_interval=2, hysteresis=2,
    )
    s0 = sc.scale.item()
    # hysteresis: first inf does NOT backoff, second does
    sc.update(found_inf=True)
    assert sc.scale.item() == s0
    sc.update(found_inf=True)
    assert sc.scale.item() == s0 * 0.5
    # growth after `growth_interval` clean steps
    sc.update(found_inf=False)
    sc.update(found_inf=False)
    assert sc.scale.item() == s0
    # state roundtrip
    sd = sc.state_dict()
    sc2 = DynamicGradScaler(2.0, 1.0, 2.0, 0.5, 2, 2)
    sc2.load_state_dict(sd)
    assert sc2.scale.item() == sc.scale.item()


def test_clip_grad_inf_norm_and_count_zeros(dist_single):
    from megatron_amd import parallel as mpu
    from megatron_amd.optim.clip_grads import (
        clip_grad_norm_fp32, count_zeros_fp32,
    )

    if not mpu.model_parallel_is_initialized():
        mpu.initialize_model_parallel(1, 1)

    p = torch.nn.Parameter(torch.zeros(4))
    p.grad = torch.tensor([3.0, -7.0, 0.0, 1.0])
    norm = clip_grad_norm_fp32(
        [p], [p.grad], max_norm=100.0, norm_type=torch.inf,
        model_parallel_group=mpu.get_model_parallel_group(),
    )
    assert norm == 7.0

    # L2 with clipping applied
    p.grad = torch.tensor([3.0, 4.0, 0.0, 0.0])  # norm 5
    norm = clip_grad_norm_fp32(
        [p], [p.grad], max_norm=1.0,
        model_parallel_group=mpu.get_model_parallel_group(),
    )
    assert abs(norm - 5.0) < 1e-6
    assert abs(p.grad.norm().item() - 1.0) < 1e-5  # scaled down to max_norm

    p.grad = torch.tensor([3.0, 0.0, 0.0, 1.0])
    zeros = count_zeros_fp32(
        [p], model_parallel_group=mpu.get_model_parallel_group()
    )
    assert zeros == 2


def test_vocab_parallel_ce_label_smoothing(dist_single):
    from megatron_amd import parallel as mpu

    if not mpu.model_parallel_is_initialized():
        mpu.initialize_model_parallel(1, 1)
    torch.manual_seed(0)
    s, b, v = 4, 2, 16
    logits = torch.randn(s, b, v, requires_grad=True)
    target = torch.randint(0, v, (s, b))
    eps = 0.1
    loss = mpu.vocab_parallel_cross_entropy(logits, target,
                                            label_smoothing=eps)
    # NeMo-style convention the reference implements:
    # (1 - a) * CE - a * mean(log_softmax), a = eps * K / (K - 1)
    lp = torch.log_softmax(logits.detach(), dim=-1)
    ce = torch.nn.functional.cross_entropy(
        logits.detach().view(-1, v), target.view(-1), reduction="none"
    ).view(s, b)
    a = eps * v / (v - 1)
    ref = (1 - a) * ce - a * lp.mean(-1)
    assert torch.allclose(loss, ref, atol=1e-5), (loss - ref).abs().max()
    loss.sum().backward()
    lr = logits.detach().clone().requires_grad_(True)
    lp2 = torch.log_softmax(lr, dim=-1)
    ce2 = torch.nn.functional.cross_entropy(
        lr.view(-1, v), target.view(-1), reduction="none"
    ).view(s, b)
    ((1 - a) * ce2 - a * lp2.mean(-1)).sum().backward()
    assert torch.allclose(logits.grad, lr.grad, atol=1e-5), (
        (logits.grad - lr.grad).abs().max()
    )


def test_rampup_batch_size_calculator():
    """Global batch grows from start to target in even increments over the
    rampup samples; every intermediate size divides by micro*dp."""
    from megatron_amd.microbatches import RampupBatchsizeNumMicroBatches

    calc = RampupBatchsizeNumMicroBatches(
        start_batch_size=4, batch_size_increment=4, ramup_samples=100,
        global_batch_size=16, micro_batch_size=2, data_parallel_size=1,
    )
    # 3 increments of 4 over 100 samples -> one every 33.3 samples
    assert calc.get_current_global_batch_size() == 4
    assert calc.get() == 2  # 4 / (2*1)
    calc.update(34, consistency_check=True)
    assert calc.get_current_global_batch_size() == 8
    calc.update(67, consistency_check=True)
    assert calc.get_current_global_batch_size() == 12
    calc.update(101, consistency_check=True)
    assert calc.get_current_global_batch_size() == 16
    assert calc.get() == 8
    # never overshoots the target
    calc.update(10_000, consistency_check=True)
    assert calc.get_current_global_batch_size() == 16
