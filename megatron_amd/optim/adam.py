"""Fused Adam(W) for MI355X.

Replaces apex.optimizers.FusedAdam (reference megatron/optimizer/__init__.py:3).
State and params are fp32. On GPU the update runs as one fused HIP kernel per
param group via the ops extension when the group's tensors are the flat master
buffers, falling back to torch._foreach_* fused kernels otherwise; on CPU it
is the same math through _foreach.
"""

from __future__ import annotations

import math
import torch


class FusedAdam(torch.optim.Optimizer):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.0, adam_w_mode=True):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.adam_w_mode = adam_w_mode

    supports_grad_scale = True

    @torch.no_grad()
    def step(self, closure=None, grad_scale=1.0):
        """grad_scale multiplies grads inside the fused kernel — the
        mixed-precision optimizer passes the deferred clip coefficient here
        so clipping costs no extra memory pass."""
        self.wrote_model_params = False
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for group in self.param_groups:
            params, grads, exp_avgs, exp_avg_sqs = [], [], [], []
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p)
                    state["exp_avg_sq"] = torch.zeros_like(p)
                params.append(p)
                grads.append(p.grad)
                exp_avgs.append(state["exp_avg"])
                exp_avg_sqs.append(state["exp_avg_sq"])
                state["step"] += 1
            if not params:
                continue

            beta1, beta2 = group["betas"]
            lr = group["lr"]
            eps = group["eps"]
            wd = group["weight_decay"]
            step = self.state[params[0]]["step"]
            bias_correction1 = 1 - beta1 ** step
            bias_correction2 = 1 - beta2 ** step

            ext = None
            if params[0].is_cuda:
                from ..ops import ext as _ext
                mod = _ext.load(required=True)
                if hasattr(mod, "fused_adam"):
                    ext = mod
            if ext is not None:
                model_outs = [getattr(p, "model_out", None) for p in params]
                if all(m is not None for m in model_outs):
                    # fused master update + bf16 model-param write-back
                    ext.fused_adam_with_model_copy(
                        [p.view(-1) for p in params],
                        [g.reshape(-1) for g in grads],
                        [m.view(-1) for m in exp_avgs],
                        [v.view(-1) for v in exp_avg_sqs],
                        [m.view(-1) for m in model_outs],
                        lr, beta1, beta2, eps, wd, step,
                        1 if self.adam_w_mode else 0, float(grad_scale),
                    )
                    self.wrote_model_params = True
                else:
                    ext.fused_adam(
                        params, grads, exp_avgs, exp_avg_sqs,
                        lr, beta1, beta2, eps, wd, step,
                        1 if self.adam_w_mode else 0, float(grad_scale),
                    )
                continue

            if grad_scale != 1.0:
                grads = list(torch._foreach_mul(grads, grad_scale))
            if self.adam_w_mode and wd != 0.0:
                torch._foreach_mul_(params, 1 - lr * wd)
            elif wd != 0.0:
                grads = list(torch._foreach_add(grads, params, alpha=wd))

            torch._foreach_mul_(exp_avgs, beta1)
            torch._foreach_add_(exp_avgs, grads, alpha=1 - beta1)
            torch._foreach_mul_(exp_avg_sqs, beta2)
            torch._foreach_addcmul_(exp_avg_sqs, grads, grads, value=1 - beta2)

            step_size = lr / bias_correction1
            denom = torch._foreach_sqrt(exp_avg_sqs)
            torch._foreach_div_(denom, math.sqrt(bias_correction2))
            torch._foreach_add_(denom, eps)
            torch._foreach_addcdiv_(params, exp_avgs, denom, value=-step_size)
        return loss


class FusedSGD(torch.optim.SGD):
    pass
