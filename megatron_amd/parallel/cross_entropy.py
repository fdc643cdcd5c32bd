"""Vocab-parallel cross entropy (reference
megatron/core/tensor_parallel/cross_entropy.py:14-175).

Softmax-CE over logits sharded along the vocab dim: three TP all-reduces
(MAX of logits, SUM of the predicted logit, SUM of exp) — all shaped (s, b),
tiny next to the (s, b, v/tp) logits, so they run as single fused RCCL calls.
"""

from __future__ import annotations

import torch

from . import state as ps
from .utils import VocabUtility


class _VocabParallelCrossEntropy(torch.autograd.Function):
    @staticmethod
    def forward(ctx, vocab_parallel_logits, target, label_smoothing=0.0):
        # local max then TP max-reduce
        logits_max = torch.max(vocab_parallel_logits, dim=-1)[0]
        torch.distributed.all_reduce(
            logits_max, op=torch.distributed.ReduceOp.MAX,
            group=ps.get_tensor_model_parallel_group(),
        )
        vocab_parallel_logits = vocab_parallel_logits - logits_max.unsqueeze(dim=-1)

        # local slice of the target
        partition_vocab_size = vocab_parallel_logits.size()[-1]
        rank = ps.get_tensor_model_parallel_rank()
        world_size = ps.get_tensor_model_parallel_world_size()
        vocab_start_index, vocab_end_index = (
            VocabUtility.vocab_range_from_per_partition_vocab_size(
                partition_vocab_size, rank, world_size
            )
        )
        target_mask = (target < vocab_start_index) | (target >= vocab_end_index)
        masked_target = target.clone() - vocab_start_index
        masked_target[target_mask] = 0

        logits_2d = vocab_parallel_logits.view(-1, partition_vocab_size)
        masked_target_1d = masked_target.view(-1)
        arange_1d = torch.arange(
            start=0, end=logits_2d.size()[0], device=logits_2d.device
        )
        predicted_logits_1d = logits_2d[arange_1d, masked_target_1d].clone()
        predicted_logits = predicted_logits_1d.view_as(target)
        predicted_logits[target_mask] = 0.0
        torch.distributed.all_reduce(
            predicted_logits, op=torch.distributed.ReduceOp.SUM,
            group=ps.get_tensor_model_parallel_group(),
        )

        exp_logits = vocab_parallel_logits
        torch.exp(vocab_parallel_logits, out=exp_logits)
        sum_exp_logits = exp_logits.sum(dim=-1)
        torch.distributed.all_reduce(
            sum_exp_logits, op=torch.distributed.ReduceOp.SUM,
            group=ps.get_tensor_model_parallel_group(),
        )

        loss = torch.log(sum_exp_logits) - predicted_logits

        # normalize FIRST: the smoothing term needs true log-probabilities
        # (reference cross_entropy.py:67-86 divides before smoothing)
        exp_logits.div_(sum_exp_logits.unsqueeze(dim=-1))

        vocab_size = exp_logits.size(-1)
        if label_smoothing > 0:
            # NeMo-style smoothing (reference cross_entropy.py:70-86):
            # (1 - alpha*K/(K-1)) * y_gt + alpha*K/(K-1) * mean_i(y_i)
            assert 1.0 > label_smoothing > 0.0
            smoothing = label_smoothing * vocab_size / (vocab_size - 1)
            log_probs = torch.log(exp_logits)
            mean_log_probs = log_probs.mean(dim=-1)
            loss = (1.0 - smoothing) * loss - smoothing * mean_log_probs

        ctx.label_smoothing, ctx.vocab_size = label_smoothing, vocab_size
        ctx.save_for_backward(exp_logits, target_mask, masked_target_1d)
        return loss

    @staticmethod
    def backward(ctx, grad_output):
        softmax, target_mask, masked_target_1d = ctx.saved_tensors
        label_smoothing, vocab_size = ctx.label_smoothing, ctx.vocab_size

        grad_input = softmax
        partition_vocab_size = softmax.size()[-1]
        grad_2d = grad_input.view(-1, partition_vocab_size)
        arange_1d = torch.arange(start=0, end=grad_2d.size()[0], device=grad_2d.device)
        softmax_update = 1.0 - target_mask.view(-1).float()

        if label_smoothing > 0:
            smoothing = label_smoothing * vocab_size / (vocab_size - 1)
            grad_2d[arange_1d, masked_target_1d] -= (1.0 - smoothing) * softmax_update
            average_grad = 1 / vocab_size
            grad_2d[arange_1d, :] -= smoothing * average_grad
        else:
            grad_2d[arange_1d, masked_target_1d] -= softmax_update

        grad_input.mul_(grad_output.unsqueeze(dim=-1))
        return grad_input, None, None


def vocab_parallel_cross_entropy(vocab_parallel_logits, target, label_smoothing=0.0):
    return _VocabParallelCrossEntropy.apply(
        vocab_parallel_logits, target, label_smoothing
    )


def vocab_parallel_max_indices(vocab_parallel_logits: torch.Tensor) -> torch.Tensor:
    """Global argmax over TP-sharded vocab (used by the accuracy metric;
    reference cross_entropy.py:146-175)."""
    partition_vocab_size = vocab_parallel_logits.size(-1)
    rank = ps.get_tensor_model_parallel_rank()
    world_size = ps.get_tensor_model_parallel_world_size()
    vocab_start_index, _ = VocabUtility.vocab_range_from_per_partition_vocab_size(
        partition_vocab_size, rank, world_size
    )
    local_max, local_indices = torch.max(vocab_parallel_logits, dim=-1)
    local_indices = local_indices + vocab_start_index
    if world_size == 1:
        return local_indices
    all_max = [torch.empty_like(local_max) for _ in range(world_size)]
    all_indices = [torch.empty_like(local_indices) for _ in range(world_size)]
    torch.distributed.all_gather(
        all_max, local_max, group=ps.get_tensor_model_parallel_group()
    )
    torch.distributed.all_gather(
        all_indices, local_indices, group=ps.get_tensor_model_parallel_group()
    )
    all_max = torch.stack(all_max)
    all_indices = torch.stack(all_indices)
    which = torch.max(all_max, dim=0)[1]
    return torch.gather(all_indices, 0, which.unsqueeze(0)).squeeze(0)
