"""GLUE base dataset (reference tasks/glue/data.py): TSV files ->
{text_a, text_b, label, uid} samples, tokenized lazily per item.

Subclasses implement `process_samples_from_single_path` for their TSV
schema (MNLI: sentence pair at columns 8/9, gold label last; QQP: questions
at 3/4, is_duplicate at 5); this base class handles file iteration and the
[CLS] A [SEP] B [SEP] packing via tasks.data_utils at __getitem__ time, so
large train sets tokenize on the fly instead of up front."""

from __future__ import annotations

from abc import ABC, abstractmethod

from torch.utils.data import Dataset

from megatron_amd.utils import print_rank_0

from tasks.data_utils import (
    build_sample,
    build_tokens_types_paddings_from_text,
)


class GLUEAbstractDataset(ABC, Dataset):
    def __init__(self, task_name, dataset_name, datapaths, tokenizer,
                 max_seq_length):
        self.task_name = task_name
        self.dataset_name = dataset_name
        self.tokenizer = tokenizer
        self.max_seq_length = max_seq_length
        print_rank_0(f" > building {task_name} dataset for {dataset_name}:")
        print_rank_0("  > paths: " + " ".join(datapaths))
        self.samples = []
        for datapath in datapaths:
            self.samples.extend(self.process_samples_from_single_path(datapath))
        print_rank_0(f"  >> total number of samples: {len(self.samples)}")

    def __len__(self):
        return len(self.samples)

    def __getitem__(self, idx):
        raw = self.samples[idx]
        ids, types, paddings = build_tokens_types_paddings_from_text(
            raw["text_a"], raw["text_b"], self.tokenizer, self.max_seq_length
        )
        return build_sample(ids, types, paddings, raw["label"], raw["uid"])

    @abstractmethod
    def process_samples_from_single_path(self, datapath):
        """Return a list of {'text_a', 'text_b', 'label', 'uid'} dicts."""
