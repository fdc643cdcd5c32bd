"""Supervised retriever finetuning on NQ (reference
tasks/orqa/supervised/finetune.py, condensed to the single-group case):
in-batch-negative softmax over query·context scores — each question's
positive passage is the diagonal target."""

from __future__ import annotations

import functools
import json

import numpy as np
import torch
import torch.nn.functional as F
from torch.utils.data import Dataset

from megatron_amd import global_state
from megatron_amd.config import get_config
from megatron_amd.global_state import get_tokenizer
from megatron_amd.models import ModelType
from megatron_amd.models.biencoder_model import BiEncoderModel
from megatron_amd.utils import (
    average_losses_across_data_parallel_group,
    print_rank_0,
)

from tasks.data_utils import build_tokens_types_paddings_from_text
from tasks.finetune_utils import finetune


class NQSupervisedDataset(Dataset):
    """jsonl: {'question':..., 'answers':[...], 'positive_context':
    {'text':..., 'title':...}} (DPR-style retriever training data)."""

    def __init__(self, name, datapaths, tokenizer, max_seq_length):
        self.name = name
        self.dataset_name = name
        self.tokenizer = tokenizer
        self.max_seq_length = max_seq_length
        self.samples = []
        for p in datapaths:
            with open(p, encoding="utf-8") as f:
                for line in f:
                    self.samples.append(json.loads(line))
        print_rank_0(f" > {name}: {len(self.samples)} NQ training samples")

    def __len__(self):
        return len(self.samples)

    def __getitem__(self, idx):
        d = self.samples[idx]
        q_ids, q_types, q_pad = build_tokens_types_paddings_from_text(
            d["question"], None, self.tokenizer, self.max_seq_length
        )
        ctx = d["positive_context"]
        c_ids, c_types, c_pad = build_tokens_types_paddings_from_text(
            ctx.get("title", ""), ctx["text"], self.tokenizer,
            self.max_seq_length,
        )
        return {
            "query": np.array(q_ids, dtype=np.int64),
            "query_pad_mask": np.array(q_pad, dtype=np.int64),
            "context": np.array(c_ids, dtype=np.int64),
            "context_pad_mask": np.array(c_pad, dtype=np.int64),
            # keep the uniform key so finetune logging works
            "label": idx,
        }


def _loss_func(batch_size, output_tensor):
    scores = output_tensor  # [b, b] in-batch retrieval scores
    labels = torch.arange(batch_size, device=scores.device)
    loss = F.cross_entropy(scores.float(), labels)
    with torch.no_grad():
        acc = (scores.argmax(-1) == labels).float().mean()
    averaged = average_losses_across_data_parallel_group([loss, acc])
    return loss, {"lm loss": averaged[0], "in-batch acc": averaged[1]}


def _forward_step(batch, model):
    cfg = get_config()
    timers = global_state.get_timers()
    timers("batch-generator", log_level=2).start()
    try:
        batch_ = next(batch)
    except TypeError:
        batch_ = batch
    device = "cuda" if torch.cuda.is_available() else "cpu"
    q = batch_["query"].long().to(device)
    qm = batch_["query_pad_mask"].long().to(device)
    c = batch_["context"].long().to(device)
    cm = batch_["context_pad_mask"].long().to(device)
    timers("batch-generator").stop()
    scores = model(q, qm, c, cm)
    return scores, functools.partial(_loss_func, q.shape[0])


def train_valid_datasets_provider():
    cfg = get_config()
    tokenizer = get_tokenizer()
    train = NQSupervisedDataset("training", cfg.train_data, tokenizer,
                                cfg.seq_length)
    valid = NQSupervisedDataset("validation", cfg.valid_data, tokenizer,
                                cfg.seq_length)
    return train, valid


def model_provider(pre_process=True, post_process=True):
    cfg = get_config()
    print_rank_0("building biencoder (retriever) model ...")
    return BiEncoderModel(
        cfg,
        shared_query_context_model=cfg.biencoder_shared_query_context_model,
    )


def main():
    finetune(train_valid_datasets_provider, model_provider,
             model_type=ModelType.encoder_or_decoder,
             forward_step=_forward_step)
