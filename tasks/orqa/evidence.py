"""Open-retrieval evidence corpus (reference
megatron/data/orqa_wiki_dataset.py, condensed): a TSV of
`id \\t text \\t title` passages, packed as [CLS] title [SEP] text [SEP]
for the context encoder."""

from __future__ import annotations

import csv

import numpy as np
import torch
from torch.utils.data import Dataset

from megatron_amd.config import get_config
from megatron_amd.global_state import get_tokenizer
from megatron_amd.utils import print_rank_0

from tasks.data_utils import build_tokens_types_paddings_from_text


class OpenRetrievalEvidenceDataset(Dataset):
    def __init__(self, datapath, tokenizer, max_seq_length):
        print_rank_0(f" > loading evidence from {datapath}")
        self.tokenizer = tokenizer
        self.max_seq_length = max_seq_length
        self.id2text = {}
        self.row_ids = []
        with open(datapath, encoding="utf-8") as f:
            reader = csv.reader(f, delimiter="\t")
            for row in reader:
                if row[0] == "id":
                    continue
                doc_id = int(row[0])
                text, title = row[1], row[2] if len(row) > 2 else ""
                self.id2text[doc_id] = (text, title)
                self.row_ids.append(doc_id)
        print_rank_0(f"  >> loaded {len(self.row_ids)} passages")

    def __len__(self):
        return len(self.row_ids)

    def __getitem__(self, idx):
        doc_id = self.row_ids[idx]
        text, title = self.id2text[doc_id]
        ids, types, paddings = build_tokens_types_paddings_from_text(
            title, text, self.tokenizer, self.max_seq_length
        )
        return {
            "row_id": doc_id,
            "context": np.array(ids, dtype=np.int64),
            "context_types": np.array(types, dtype=np.int64),
            "context_pad_mask": np.array(paddings, dtype=np.int64),
        }


def get_open_retrieval_wiki_dataset():
    cfg = get_config()
    return OpenRetrievalEvidenceDataset(cfg.evidence_data_path,
                                        get_tokenizer(), cfg.seq_length)
