"""BERT masked-LM dataset (reference megatron/data/bert_dataset.py +
dataset_utils.py masked-LM machinery, condensed: random document pairs with
NSP labels, whole-word-agnostic token masking 15%/80-10-10)."""

from __future__ import annotations

import numpy as np
import torch

from ..global_state import get_tokenizer


class BertDataset(torch.utils.data.Dataset):
    def __init__(self, name, indexed_dataset, documents, num_samples,
                 max_seq_length, masked_lm_prob, short_seq_prob, seed,
                 binary_head=True):
        self.name = name
        self.indexed_dataset = indexed_dataset
        self.documents = documents
        self.num_samples = max(1, num_samples)
        self.max_seq_length = max_seq_length
        self.masked_lm_prob = masked_lm_prob
        self.short_seq_prob = short_seq_prob
        self.seed = seed
        self.binary_head = binary_head

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx):
        rng = np.random.RandomState((self.seed + idx) % 2 ** 31)
        tok = get_tokenizer()
        cls_id, sep_id, mask_id = tok.cls, tok.sep, tok.mask
        vocab_size = tok.vocab_size

        def doc_tokens(i):
            return np.asarray(
                self.indexed_dataset[self.documents[i % len(self.documents)]]
            )

        a = doc_tokens(rng.randint(len(self.documents)))
        if self.binary_head and rng.random() < 0.5:
            b = doc_tokens(rng.randint(len(self.documents)))
            is_next = 0
        else:
            b = a
            is_next = 1

        max_tokens = self.max_seq_length - 3  # [CLS] A [SEP] B [SEP]
        len_a = min(len(a), max_tokens // 2)
        len_b = min(len(b), max_tokens - len_a)
        tokens = np.concatenate(
            [[cls_id], a[:len_a], [sep_id], b[:len_b], [sep_id]]
        ).astype(np.int64)
        tokentypes = np.concatenate(
            [np.zeros(len_a + 2), np.ones(len_b + 1)]
        ).astype(np.int64)

        # masked LM
        labels = np.full_like(tokens, -1)
        loss_mask = np.zeros_like(tokens)
        cand = [
            i for i in range(len(tokens))
            if tokens[i] != cls_id and tokens[i] != sep_id
        ]
        rng.shuffle(cand)
        n_pred = max(1, int(len(cand) * self.masked_lm_prob))
        for i in cand[:n_pred]:
            labels[i] = tokens[i]
            loss_mask[i] = 1
            r = rng.random()
            if r < 0.8:
                tokens[i] = mask_id
            elif r < 0.9:
                tokens[i] = rng.randint(0, vocab_size)

        # pad
        pad = self.max_seq_length - len(tokens)
        pad_mask = np.concatenate([np.ones(len(tokens)), np.zeros(pad)])
        tokens = np.concatenate([tokens, np.zeros(pad, dtype=np.int64)])
        tokentypes = np.concatenate([tokentypes, np.zeros(pad,
                                                          dtype=np.int64)])
        labels = np.concatenate([labels, np.full(pad, -1, dtype=np.int64)])
        loss_mask = np.concatenate([loss_mask, np.zeros(pad,
                                                        dtype=np.int64)])
        return {
            "text": tokens,
            "types": tokentypes,
            "labels": labels,
            "is_random": int(is_next == 0),
            "loss_mask": loss_mask,
            "padding_mask": pad_mask.astype(np.int64),
            "truncated": 0,
        }
