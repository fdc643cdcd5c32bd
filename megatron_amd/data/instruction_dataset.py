"""Instruction-tuning dataset: paired -text/-role indexed datasets with
per-sample role spans; loss only on assistant tokens.

Reference: megatron/data/instruction_dataset.py:26-355. Roles are stored as a
parallel indexed dataset of the same shape with role codes per token span:
system=0, prompter=1, assistant=2 (written by tools/preprocess_instruct_data).
The collator pads to seq_length (16-multiple under variable_seq_lengths) and
emits attention/assistant/pad masks.
"""

from __future__ import annotations

from typing import List

import numpy as np
import torch

from ..config import get_config
from .blendable_dataset import BlendableDataset
from .gpt_dataset import (
    get_datasets_weights_and_num_samples,
    get_train_valid_test_split_,
)
from .indexed_dataset import make_dataset as make_indexed_dataset

ROLE_SYSTEM = 0
ROLE_PROMPTER = 1
ROLE_ASSISTANT = 2


def build_train_valid_test_datasets(data_prefix, data_impl, splits_string,
                                    train_valid_test_num_samples, seq_length,
                                    seed, skip_warmup):
    """(reference instruction_dataset.py:207-268)"""
    if len(data_prefix) == 1:
        return _build_train_valid_test_datasets(
            data_prefix[0], data_impl, splits_string,
            train_valid_test_num_samples, seq_length, seed, skip_warmup,
        )
    prefixes, weights, datasets_train_valid_test_num_samples = (
        get_datasets_weights_and_num_samples(
            data_prefix, train_valid_test_num_samples
        )
    )
    train_datasets, valid_datasets, test_datasets = [], [], []
    for i in range(len(prefixes)):
        train_ds, valid_ds, test_ds = _build_train_valid_test_datasets(
            prefixes[i], data_impl, splits_string,
            datasets_train_valid_test_num_samples[i], seq_length, seed,
            skip_warmup,
        )
        if train_ds:
            train_datasets.append(train_ds)
        if valid_ds:
            valid_datasets.append(valid_ds)
        if test_ds:
            test_datasets.append(test_ds)
    return (
        BlendableDataset(train_datasets, weights) if train_datasets else None,
        BlendableDataset(valid_datasets, weights) if valid_datasets else None,
        BlendableDataset(test_datasets, weights) if test_datasets else None,
    )


def _build_train_valid_test_datasets(data_prefix, data_impl, splits_string,
                                     train_valid_test_num_samples, seq_length,
                                     seed, skip_warmup):
    text_dataset = make_indexed_dataset(data_prefix + "-text", data_impl,
                                        skip_warmup)
    role_dataset = make_indexed_dataset(data_prefix + "-role", data_impl,
                                        skip_warmup)
    total_num_docs = text_dataset.sizes.shape[0]
    splits = get_train_valid_test_split_(splits_string, total_num_docs)

    def build_dataset(index, name):
        dataset = None
        if splits[index + 1] > splits[index]:
            documents = np.arange(
                start=splits[index], stop=splits[index + 1], step=1,
                dtype=np.int32,
            )
            dataset = InstructionDataset(
                name, data_prefix, documents, text_dataset, role_dataset,
                train_valid_test_num_samples[index], seq_length, seed,
            )
        return dataset

    return (
        build_dataset(0, "train"),
        build_dataset(1, "valid"),
        build_dataset(2, "test"),
    )


class InstructionDataset(torch.utils.data.Dataset):
    """(reference instruction_dataset.py:26-204)"""

    def __init__(self, name, data_prefix, documents, text_dataset,
                 role_dataset, num_samples, seq_length, seed):
        self.name = name
        self.text_dataset = text_dataset
        self.role_dataset = role_dataset
        self.seq_length = seq_length
        self.documents = documents
        self.num_samples = num_samples

        np_rng = np.random.RandomState(seed=seed)
        n_docs = len(documents)
        num_epochs = max(1, int(np.ceil(num_samples / max(1, n_docs))))
        idx = np.concatenate(
            [np_rng.permutation(documents) for _ in range(num_epochs)]
        )
        self.sample_map = idx[: max(num_samples, 1)]

    def __len__(self):
        return len(self.sample_map)

    def __getitem__(self, idx):
        doc = int(self.sample_map[idx % len(self.sample_map)])
        tokens = np.array(self.text_dataset[doc], dtype=np.int64)
        roles = np.array(self.role_dataset[doc], dtype=np.int64)
        if len(tokens) > self.seq_length:
            tokens = tokens[: self.seq_length]
            roles = roles[: self.seq_length]
        return {"text": tokens, "role": roles}

    @property
    def collate_fn(self):
        cfg = get_config()
        seq_length = self.seq_length
        variable = cfg.variable_seq_lengths

        def collate(samples: List[dict]):
            max_len = max(len(s["text"]) for s in samples)
            if variable:
                # pad to 16-multiple (reference instruction_dataset.py:327-330)
                pad_to = (max_len + 15) // 16 * 16
            else:
                pad_to = seq_length
            batch_tokens = np.zeros((len(samples), pad_to), dtype=np.int64)
            assistant_mask = np.zeros((len(samples), pad_to), dtype=np.float32)
            pad_mask = np.zeros((len(samples), pad_to), dtype=np.float32)
            for i, s in enumerate(samples):
                n = min(len(s["text"]), pad_to)
                batch_tokens[i, :n] = s["text"][:n]
                assistant_mask[i, :n] = (
                    s["role"][:n] == ROLE_ASSISTANT
                ).astype(np.float32)
                pad_mask[i, :n] = 1.0
            return {
                "text": torch.from_numpy(batch_tokens),
                "assistant_mask": torch.from_numpy(assistant_mask),
                "pad_mask": torch.from_numpy(pad_mask),
            }

        return collate
