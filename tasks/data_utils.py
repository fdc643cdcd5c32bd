"""Shared helpers for downstream-task datasets (reference
tasks/data_utils.py:1-105): text cleanup and the [CLS] A [SEP] B [SEP]
packing with token types and padding masks that the BERT-style heads
consume."""

from __future__ import annotations

import re

import numpy as np


def clean_text(text: str) -> str:
    """Collapse newlines/whitespace and normalize sentence-final dots."""
    text = text.replace("\n", " ")
    text = re.sub(r"\s+", " ", text)
    for _ in range(3):
        text = text.replace(" . ", ". ")
    return text


def build_sample(ids, types, paddings, label, unique_id):
    """Package one (or, for multi-choice, a list of) tokenized sequence(s)
    into the numpy dict the finetune batch producer consumes."""
    return {
        "text": np.array(ids, dtype=np.int64),
        "types": np.array(types, dtype=np.int64),
        "padding_mask": np.array(paddings, dtype=np.int64),
        "label": int(label),
        "uid": int(unique_id),
    }


def build_tokens_types_paddings_from_text(text_a, text_b, tokenizer,
                                          max_seq_length):
    a_ids = tokenizer.tokenize(text_a)
    b_ids = tokenizer.tokenize(text_b) if text_b is not None else None
    return build_tokens_types_paddings_from_ids(
        a_ids, b_ids, max_seq_length, tokenizer.cls, tokenizer.sep,
        tokenizer.pad,
    )


def build_tokens_types_paddings_from_ids(a_ids, b_ids, max_seq_length,
                                         cls_id, sep_id, pad_id):
    """[CLS] A [SEP] (B [SEP]) with segment types 0/1, trimmed to
    max_seq_length (keeping a trailing [SEP] when trimmed or when B exists)
    and padded with a 0/1 padding mask."""
    ids = [cls_id] + list(a_ids) + [sep_id]
    types = [0] * len(ids)
    paddings = [1] * len(ids)

    if b_ids is not None:
        ids += list(b_ids)
        types += [1] * len(b_ids)
        paddings += [1] * len(b_ids)

    trimmed = False
    if len(ids) >= max_seq_length:
        ids = ids[: max_seq_length - 1]
        types = types[: max_seq_length - 1]
        paddings = paddings[: max_seq_length - 1]
        trimmed = True

    if b_ids is not None or trimmed:
        ids.append(sep_id)
        types.append(1 if b_ids is not None else 0)
        paddings.append(1)

    pad = max_seq_length - len(ids)
    if pad > 0:
        ids += [pad_id] * pad
        types += [pad_id] * pad
        paddings += [0] * pad
    return ids, types, paddings
