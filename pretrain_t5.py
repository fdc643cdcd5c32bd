"""T5 pretraining entry point (reference pretrain_t5.py)."""

from __future__ import annotations

import functools

import torch

from megatron_amd import global_state
from megatron_amd.config import get_config
from megatron_amd.models import ModelType, T5Model
from megatron_amd.parallel import broadcast_data
from megatron_amd.training import pretrain
from megatron_amd.utils import average_losses_across_data_parallel_group


def model_provider(pre_process=True, post_process=True):
    cfg = get_config()
    return T5Model(cfg, num_tokentypes=0, parallel_output=True,
                   pre_process=pre_process, post_process=post_process)


def get_batch(data_iterator):
    keys = ["text_enc", "text_dec", "labels", "loss_mask", "enc_mask",
            "dec_mask"]
    datatype = torch.int64
    data = next(data_iterator) if data_iterator is not None else None
    data_b = broadcast_data(keys, data, datatype)
    tokens_enc = data_b["text_enc"].long()
    tokens_dec = data_b["text_dec"].long()
    labels = data_b["labels"].long()
    loss_mask = data_b["loss_mask"].float()
    enc_mask = data_b["enc_mask"]
    dec_mask = data_b["dec_mask"]
    # pairwise masks
    enc_mask_2d = enc_mask.unsqueeze(1) * enc_mask.unsqueeze(2)
    dec_mask_2d = dec_mask.unsqueeze(1) * dec_mask.unsqueeze(2)
    enc_dec_mask = dec_mask.unsqueeze(2) * enc_mask.unsqueeze(1)
    return (tokens_enc, tokens_dec, loss_mask, labels, enc_mask_2d,
            dec_mask_2d, enc_dec_mask)


def loss_func(loss_mask, output_tensor):
    lm_loss_ = output_tensor.float()
    lm_loss = torch.sum(lm_loss_.view(-1) * loss_mask.reshape(-1)) / (
        loss_mask.sum() + 1e-8
    )
    averaged = average_losses_across_data_parallel_group([lm_loss])
    return lm_loss, {"lm loss": averaged[0]}


def forward_step(data_iterator, model):
    timers = global_state.get_timers()
    timers("batch-generator", log_level=2).start()
    (tokens_enc, tokens_dec, loss_mask, lm_labels, enc_mask, dec_mask,
     enc_dec_mask) = get_batch(data_iterator)
    timers("batch-generator").stop()

    output_tensor = model(
        tokens_enc, tokens_dec, enc_mask, dec_mask, enc_dec_mask,
        lm_labels=lm_labels,
    )
    return output_tensor, functools.partial(loss_func, loss_mask)


def train_valid_test_datasets_provider(train_val_test_num_samples):
    cfg = get_config()
    if cfg.data_path is None:
        raise ValueError("--data_path required for T5 pretraining")
    import numpy as np

    from megatron_amd.data.gpt_dataset import (
        get_indexed_dataset_, get_train_valid_test_split_,
    )
    from megatron_amd.data.t5_dataset import T5Dataset

    indexed = get_indexed_dataset_(cfg.data_path[0], cfg.data_impl,
                                   not cfg.mmap_warmup)
    splits = get_train_valid_test_split_(cfg.split, indexed.sizes.shape[0])

    def build(index, name):
        if splits[index + 1] <= splits[index]:
            return None
        docs = np.arange(splits[index], splits[index + 1], dtype=np.int32)
        return T5Dataset(
            name, indexed, docs, train_val_test_num_samples[index],
            cfg.encoder_seq_length or cfg.seq_length,
            cfg.decoder_seq_length or 128, cfg.mask_prob, cfg.seed,
        )

    return build(0, "train"), build(1, "valid"), build(2, "test")


if __name__ == "__main__":
    pretrain(
        train_valid_test_datasets_provider, model_provider,
        ModelType.encoder_and_decoder, forward_step,
        args_defaults={"tokenizer_type": "SentencePieceTokenizer",
                       "position_embedding_type": "absolute",
                       "vocab_extra_ids": 100},
    )
