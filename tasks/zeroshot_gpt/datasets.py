"""Zero-shot eval datasets (reference tasks/zeroshot_gpt/datasets.py):
sliding-window LM perplexity chunks (WikiText-103) and LAMBADA last-word
accuracy samples."""

from __future__ import annotations

import json
import math

import numpy as np
import torch

from megatron_amd.config import get_config
from megatron_amd.global_state import get_tokenizer
from megatron_amd.utils import print_rank_0

from tasks.zeroshot_gpt.detokenizer import get_detokenizer


def build_dataset(task):
    if task == "LAMBADA":
        return _build_lambada_dataset()
    if task == "WIKITEXT103":
        return _build_wikitext103_dataset()
    raise NotImplementedError(f"dataset for {task} task is not implemented")


class _LMDataset(torch.utils.data.Dataset):
    """Overlapping seq_len windows over one long token stream; the pad_mask
    marks only the NEW (non-overlapping) positions of each window so every
    token is scored exactly once."""

    def __init__(self, tokens, seq_len, pad_idx, num_original_tokens,
                 num_tokenized_tokens, overlapping_eval=None):
        self.tokens = tokens
        self.seq_len = seq_len
        self.pad_idx = pad_idx
        self.overlapping_eval = max(1, overlapping_eval or seq_len)
        self.num_original_tokens = num_original_tokens
        self.num_tokenized_tokens = num_tokenized_tokens
        targets = max(len(tokens) - 1 - self.overlapping_eval, 0)
        self.total_sequences = max(
            math.ceil(targets / self.overlapping_eval) + 1, 1
        )

    def __len__(self):
        return self.total_sequences

    def __getitem__(self, idx):
        start = idx * self.overlapping_eval
        tokens = list(self.tokens[start : start + self.seq_len + 1])
        pad_mask = [1] * len(tokens)
        if len(tokens) < self.seq_len + 1:
            n_pad = self.seq_len + 1 - len(tokens)
            pad_mask += [0] * n_pad
            tokens += [self.pad_idx] * n_pad
        pad_mask = np.array(pad_mask[1:])
        if self.overlapping_eval != self.seq_len and idx != 0:
            pad_mask[: -self.overlapping_eval] *= 0
        return {"text": np.array(tokens), "pad_mask": pad_mask}


class _LambadaDataset(torch.utils.data.Dataset):
    """Score only the final word of each passage (strict mode re-tokenizes
    the last whitespace word separately, matching the published protocol)."""

    def __init__(self, path, pad_idx, tokenizer, seq_len, strict=False):
        print_rank_0(f"> building lambada dataset from {path} ...")
        self.seq_len = seq_len
        self.pad_idx = pad_idx
        self.tokenizer = tokenizer
        self.strict = strict
        self.tokens = []
        self.labels = []
        with open(path) as f:
            for line in f:
                text = json.loads(line)["text"]
                tokens, labels = self.get_tokens(text)
                self.tokens.append(tokens)
                self.labels.append(labels)

    def get_tokens(self, text):
        if not self.strict:
            tokens = self.tokenizer.tokenize(text)
            return tokens[:-1], [tokens[-1]]
        last_word = text.split()[-1]
        start = text.rfind(last_word)
        context = self.tokenizer.tokenize(text[:start].strip())
        return context, self.tokenizer.tokenize(" " + last_word)

    def __len__(self):
        return len(self.tokens)

    def __getitem__(self, idx):
        tokens = list(self.tokens[idx])
        labels = list(self.labels[idx])
        pad_mask = [0] * len(tokens) + [1] * len(labels)
        tokens = tokens + labels
        if len(tokens) < self.seq_len + 1:
            n_pad = self.seq_len + 1 - len(tokens)
            pad_mask += [0] * n_pad
            tokens += [self.pad_idx] * n_pad
        return {
            "text": np.array(tokens[: self.seq_len + 1]),
            "pad_mask": np.array(pad_mask[1 : self.seq_len + 1]),
        }


def _build_lambada_dataset():
    cfg = get_config()
    tokenizer = get_tokenizer()
    assert len(cfg.valid_data) == 1
    ds = _LambadaDataset(cfg.valid_data[0], tokenizer.eod, tokenizer,
                         cfg.seq_length, getattr(cfg, "strict_lambada", False))
    print_rank_0(f" > found {len(ds)} samples.")
    return ds


def _build_wikitext103_dataset():
    cfg = get_config()
    tokenizer = get_tokenizer()
    assert len(cfg.valid_data) == 1
    with open(cfg.valid_data[0], "rb") as f:
        raw = f.read().decode("utf-8")
    num_original_tokens = len(raw.strip().split(" "))
    detok = get_detokenizer(cfg.valid_data[0])(raw)
    tokens = tokenizer.tokenize(detok)
    print_rank_0(
        f" > number of original tokens: {num_original_tokens}, number of "
        f"detokenized tokens: {len(tokens)}"
    )
    return _LMDataset(tokens, cfg.seq_length, tokenizer.eod,
                      num_original_tokens, len(tokens),
                      cfg.overlapping_eval)
