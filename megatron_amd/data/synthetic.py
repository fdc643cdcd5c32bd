"""Synthetic token datasets for benchmarks and plumbing tests (no network,
no dataset files in this environment)."""

from __future__ import annotations

import numpy as np
import torch


class SyntheticGPTDataset(torch.utils.data.Dataset):
    def __init__(self, vocab_size, seq_length, num_samples, seed=1234):
        self.vocab_size = vocab_size
        self.seq_length = seq_length
        self.num_samples = max(1, num_samples)
        self.seed = seed

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx):
        rng = np.random.RandomState((self.seed + idx) % (2 ** 31))
        tokens = rng.randint(
            0, self.vocab_size, size=(self.seq_length,), dtype=np.int64
        )
        return {"text": tokens}


def build_synthetic_datasets(cfg, train_val_test_num_samples):
    vocab = cfg.padded_vocab_size or 32000
    seq = cfg.seq_length + 1
    train = SyntheticGPTDataset(vocab, seq, train_val_test_num_samples[0],
                                cfg.seed)
    # 2x margin + floor: a resumed run inherits consumed_valid_samples from
    # the previous run's end-of-training eval, so the exact budget
    # (eval_iters * (train_iters//eval_interval + 1)) can under-count
    valid = SyntheticGPTDataset(
        vocab, seq, max(64, 2 * train_val_test_num_samples[1]), cfg.seed + 1
    )
    test = SyntheticGPTDataset(
        vocab, seq, max(64, 2 * train_val_test_num_samples[2]), cfg.seed + 2
    )
    return train, valid, test
