"""Pluggable eval metrics (reference megatron/metrics.py:11-110)."""

from __future__ import annotations

from dataclasses import dataclass
import torch

from . import parallel as mpu


@dataclass
class MetricInput:
    batch: dict
    logits: torch.Tensor
    loss: torch.Tensor


def perplexity(inp: MetricInput):
    return {"perplexity": torch.exp(inp.loss).item()}


def accuracy(inp: MetricInput):
    """Token accuracy on loss-masked positions via TP-sharded argmax
    (reference metrics.py:40-70)."""
    labels = inp.batch["labels"]
    loss_mask = inp.batch["loss_mask"]
    preds = mpu.vocab_parallel_max_indices(inp.logits)  # [s, b]
    preds = preds.transpose(0, 1)  # [b, s]
    correct = ((preds == labels).float() * loss_mask).sum()
    total = loss_mask.sum()
    return {"accuracy": (correct / total).item() if total > 0 else 0.0}


def count_loss_mask(inp: MetricInput):
    return {"count_loss_mask": inp.batch["loss_mask"].sum().item()}


METRICS = {
    "perplexity": perplexity,
    "accuracy": accuracy,
    "count_loss_mask": count_loss_mask,
}
