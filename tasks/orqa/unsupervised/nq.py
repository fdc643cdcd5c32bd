"""Natural Questions open-domain eval data (reference
tasks/orqa/unsupervised/nq.py): question/answers pairs -> packed query
token batches plus the raw answer references."""

from __future__ import annotations

import csv
import json

import numpy as np
import torch
from torch.utils.data import DataLoader, Dataset

from megatron_amd.config import get_config
from megatron_amd.global_state import get_tokenizer
from megatron_amd.utils import print_rank_0

from tasks.data_utils import build_tokens_types_paddings_from_text


def normalize_question(question):
    if question.endswith("?"):
        question = question[:-1]
    return question


class NQDataset(Dataset):
    """Reads jsonl ({'question':..., 'answers':[...]}) or tab-separated
    (question \\t ["ans", ...]) files."""

    def __init__(self, name, datapath, tokenizer, max_seq_length):
        self.name = name
        self.tokenizer = tokenizer
        self.max_seq_length = max_seq_length
        print_rank_0(f" > building NQ dataset for {name} from {datapath}")
        self.samples = []
        with open(datapath, encoding="utf-8") as f:
            if datapath.endswith(".json") or datapath.endswith(".jsonl"):
                for line in f:
                    d = json.loads(line)
                    self.samples.append(
                        (normalize_question(d["question"]), d["answers"])
                    )
            else:
                for row in csv.reader(f, delimiter="\t"):
                    question, answers = row[0], json.loads(row[1])
                    self.samples.append(
                        (normalize_question(question), answers)
                    )
        print_rank_0(f"  >> total number of samples: {len(self.samples)}")

    def __len__(self):
        return len(self.samples)

    def __getitem__(self, idx):
        question, answers = self.samples[idx]
        ids, types, paddings = build_tokens_types_paddings_from_text(
            question, None, self.tokenizer, self.max_seq_length
        )
        return {
            "query": np.array(ids, dtype=np.int64),
            "query_types": np.array(types, dtype=np.int64),
            "query_pad_mask": np.array(paddings, dtype=np.int64),
            "reference": answers,
        }


def _collate(batch):
    return {
        "query": torch.tensor(
            np.stack([b["query"] for b in batch])
        ),
        "query_types": torch.tensor(
            np.stack([b["query_types"] for b in batch])
        ),
        "query_pad_mask": torch.tensor(
            np.stack([b["query_pad_mask"] for b in batch])
        ),
        "reference": [b["reference"] for b in batch],
    }


def get_nq_dataset(qa_data, split):
    cfg = get_config()
    return NQDataset(f"Google NQ {split}", qa_data, get_tokenizer(),
                     cfg.seq_length)


def get_one_epoch_nq_dataloader(dataset, micro_batch_size=None):
    cfg = get_config()
    return DataLoader(
        dataset, batch_size=micro_batch_size or cfg.micro_batch_size,
        shuffle=False, num_workers=cfg.num_workers, collate_fn=_collate,
    )


def process_nq_batch(batch):
    device = "cuda" if torch.cuda.is_available() else "cpu"
    tokens = batch["query"].long().to(device)
    types = batch["query_types"].long().to(device)
    mask = batch["query_pad_mask"].long().to(device)
    return tokens, mask, types, batch["reference"]
