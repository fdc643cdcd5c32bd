"""Inverse-Cloze-Task (ICT) retriever pretraining entry
(reference pretrain_ict.py): trains the biencoder with in-batch negatives."""

from __future__ import annotations

import torch
import torch.nn.functional as F

from megatron_amd import global_state
from megatron_amd.config import get_config
from megatron_amd.models import ModelType
from megatron_amd.models.biencoder_model import BiEncoderModel
from megatron_amd.parallel import broadcast_data
from megatron_amd.training import pretrain
from megatron_amd.utils import average_losses_across_data_parallel_group


def model_provider(pre_process=True, post_process=True):
    cfg = get_config()
    return BiEncoderModel(cfg, num_tokentypes=2, projection_dim=128)


def get_batch(data_iterator):
    keys = ["query_tokens", "query_mask", "context_tokens", "context_mask"]
    data = next(data_iterator) if data_iterator is not None else None
    data_b = broadcast_data(keys, data, torch.int64)
    return (data_b["query_tokens"].long(), data_b["query_mask"].long(),
            data_b["context_tokens"].long(), data_b["context_mask"].long())


def loss_func(output_tensor):
    scores = output_tensor.float()
    batch = scores.shape[0]
    labels = torch.arange(batch, device=scores.device)
    loss = F.cross_entropy(scores, labels)
    acc = (scores.argmax(-1) == labels).float().mean()
    averaged = average_losses_across_data_parallel_group([loss, acc])
    return loss, {"retrieval loss": averaged[0], "in-batch acc": averaged[1]}


def forward_step(data_iterator, model):
    timers = global_state.get_timers()
    timers("batch-generator", log_level=2).start()
    query_tokens, query_mask, context_tokens, context_mask = get_batch(
        data_iterator
    )
    timers("batch-generator").stop()
    output = model(query_tokens, query_mask, context_tokens, context_mask)
    return output, loss_func


def train_valid_test_datasets_provider(train_val_test_num_samples):
    """ICT dataset: a sentence (query) vs its surrounding block (context),
    built over an indexed dataset of documents."""
    cfg = get_config()
    if cfg.data_path is None:
        raise ValueError("--data_path required for ICT pretraining")
    import numpy as np

    from megatron_amd.data.gpt_dataset import (
        get_indexed_dataset_, get_train_valid_test_split_,
    )

    indexed = get_indexed_dataset_(cfg.data_path[0], cfg.data_impl, True)
    splits = get_train_valid_test_split_(cfg.split, indexed.sizes.shape[0])

    class ICTDataset(torch.utils.data.Dataset):
        def __init__(self, docs, num_samples, seq_length, seed):
            self.docs = docs
            self.num_samples = max(1, num_samples)
            self.seq = seq_length
            self.seed = seed

        def __len__(self):
            return self.num_samples

        def __getitem__(self, idx):
            rng = np.random.RandomState((self.seed + idx) % 2 ** 31)
            doc = np.asarray(
                indexed[self.docs[rng.randint(len(self.docs))]]
            )
            half = max(1, min(len(doc) // 2, self.seq))
            q = np.zeros(self.seq, dtype=np.int64)
            c = np.zeros(self.seq, dtype=np.int64)
            qlen = min(half, self.seq)
            clen = min(len(doc), self.seq)
            q[:qlen] = doc[:qlen]
            c[:clen] = doc[:clen]
            return {
                "query_tokens": q,
                "query_mask": (q != 0).astype(np.int64),
                "context_tokens": c,
                "context_mask": (c != 0).astype(np.int64),
            }

    def build(i, name):
        if splits[i + 1] <= splits[i]:
            return None
        docs = np.arange(splits[i], splits[i + 1], dtype=np.int32)
        return ICTDataset(docs, train_val_test_num_samples[i],
                          cfg.seq_length, cfg.seed)

    return build(0, "train"), build(1, "valid"), build(2, "test")


if __name__ == "__main__":
    pretrain(
        train_valid_test_datasets_provider, model_provider,
        ModelType.encoder_or_decoder, forward_step,
        args_defaults={"tokenizer_type": "BertWordPieceLowerCase",
                       "position_embedding_type": "absolute"},
    )
