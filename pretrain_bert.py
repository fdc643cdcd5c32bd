"""BERT pretraining entry point (reference pretrain_bert.py)."""

from __future__ import annotations

import functools

import torch

from megatron_amd import global_state
from megatron_amd.config import get_config
from megatron_amd.models import BertModel, ModelType
from megatron_amd.parallel import broadcast_data
from megatron_amd.training import pretrain
from megatron_amd.utils import average_losses_across_data_parallel_group


def model_provider(pre_process=True, post_process=True):
    cfg = get_config()
    return BertModel(
        cfg, num_tokentypes=2, add_binary_head=cfg.bert_binary_head,
        parallel_output=True, pre_process=pre_process,
        post_process=post_process,
    )


def get_batch(data_iterator):
    keys = ["text", "types", "labels", "is_random", "loss_mask",
            "padding_mask"]
    datatype = torch.int64
    data = next(data_iterator) if data_iterator is not None else None
    data_b = broadcast_data(keys, data, datatype)
    tokens = data_b["text"].long()
    types = data_b["types"].long()
    sentence_order = data_b["is_random"].long()
    loss_mask = data_b["loss_mask"].float()
    lm_labels = data_b["labels"].long()
    padding_mask = data_b["padding_mask"].long()
    return tokens, types, sentence_order, loss_mask, lm_labels, padding_mask


def loss_func(loss_mask, sentence_order, output_tensor):
    lm_loss_, sop_logits = output_tensor
    lm_loss_ = lm_loss_.float()
    loss_mask = loss_mask.float()
    lm_loss = torch.sum(lm_loss_.view(-1) * loss_mask.reshape(-1)) / (
        loss_mask.sum() + 1e-8
    )
    if sop_logits is not None:
        sop_loss = torch.nn.functional.cross_entropy(
            sop_logits.view(-1, 2).float(), sentence_order.view(-1),
            ignore_index=-1,
        )
        loss = lm_loss + sop_loss
        averaged = average_losses_across_data_parallel_group(
            [lm_loss, sop_loss]
        )
        return loss, {"lm loss": averaged[0], "sop loss": averaged[1]}
    averaged = average_losses_across_data_parallel_group([lm_loss])
    return lm_loss, {"lm loss": averaged[0]}


def forward_step(data_iterator, model):
    timers = global_state.get_timers()
    timers("batch-generator", log_level=2).start()
    tokens, types, sentence_order, loss_mask, lm_labels, padding_mask = (
        get_batch(data_iterator)
    )
    timers("batch-generator").stop()

    cfg = get_config()
    if not cfg.bert_binary_head:
        types = None
    output_tensor = model(tokens, padding_mask, tokentype_ids=types,
                          lm_labels=lm_labels)
    return output_tensor, functools.partial(loss_func, loss_mask,
                                            sentence_order)


def train_valid_test_datasets_provider(train_val_test_num_samples):
    cfg = get_config()
    if cfg.data_path is None:
        raise ValueError("--data_path required for BERT pretraining")
    import numpy as np

    from megatron_amd.data.bert_dataset import BertDataset
    from megatron_amd.data.gpt_dataset import (
        get_indexed_dataset_, get_train_valid_test_split_,
    )

    indexed = get_indexed_dataset_(cfg.data_path[0], cfg.data_impl,
                                   not cfg.mmap_warmup)
    splits = get_train_valid_test_split_(cfg.split, indexed.sizes.shape[0])

    def build(index, name):
        if splits[index + 1] <= splits[index]:
            return None
        docs = np.arange(splits[index], splits[index + 1], dtype=np.int32)
        return BertDataset(
            name, indexed, docs, train_val_test_num_samples[index],
            cfg.seq_length, cfg.mask_prob, cfg.short_seq_prob, cfg.seed,
            binary_head=cfg.bert_binary_head,
        )

    return build(0, "train"), build(1, "valid"), build(2, "test")


if __name__ == "__main__":
    pretrain(
        train_valid_test_datasets_provider, model_provider,
        ModelType.encoder_or_decoder, forward_step,
        args_defaults={"tokenizer_type": "BertWordPieceLowerCase",
                       "position_embedding_type": "absolute"},
    )
