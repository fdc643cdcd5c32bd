"""RACE multiple-choice dataset (reference tasks/race/data.py): each *.txt
file holds JSON lines with an article, questions, 4 options each, and
letter answers. Every question becomes NUM_CHOICES packed sequences
([CLS] question+choice [SEP] article [SEP]) that collapse into the batch
dimension (sample_multiplier)."""

from __future__ import annotations

import glob
import json
import os
import time

from torch.utils.data import Dataset

from megatron_amd.utils import print_rank_0

from tasks.data_utils import (
    build_sample,
    build_tokens_types_paddings_from_ids,
    clean_text,
)

NUM_CHOICES = 4
MAX_QA_LENGTH = 128


class RaceDataset(Dataset):
    def __init__(self, dataset_name, datapaths, tokenizer, max_seq_length,
                 max_qa_length=MAX_QA_LENGTH):
        self.dataset_name = dataset_name
        print_rank_0(f" > building RACE dataset for {dataset_name}:")
        print_rank_0("  > paths: " + " ".join(datapaths))
        self.samples = []
        for datapath in datapaths:
            self.samples.extend(
                process_single_datapath(datapath, tokenizer, max_qa_length,
                                        max_seq_length)
            )
        print_rank_0(f"  >> total number of samples: {len(self.samples)}")
        # each item expands to NUM_CHOICES rows in the batch dimension
        self.sample_multiplier = NUM_CHOICES

    def __len__(self):
        return len(self.samples)

    def __getitem__(self, idx):
        return self.samples[idx]


def process_single_datapath(datapath, tokenizer, max_qa_length,
                            max_seq_length):
    print_rank_0(f"   > working on {datapath}")
    start = time.time()
    filenames = glob.glob(os.path.join(datapath, "*.txt"))
    samples = []
    num_docs = num_questions = 0
    for filename in filenames:
        with open(filename) as f:
            for line in f:
                data = json.loads(line)
                num_docs += 1
                context_ids = tokenizer.tokenize(clean_text(data["article"]))
                questions = data["questions"]
                choices = data["options"]
                answers = data["answers"]
                assert len(questions) == len(answers) == len(choices)

                for qi, question in enumerate(questions):
                    num_questions += 1
                    label = ord(answers[qi]) - ord("A")
                    assert 0 <= label < NUM_CHOICES
                    assert len(choices[qi]) == NUM_CHOICES

                    ids_list, types_list, paddings_list = [], [], []
                    for choice in choices[qi]:
                        # cloze questions substitute the blank; others append
                        qa = (question.replace("_", choice)
                              if "_" in question
                              else " ".join([question, choice]))
                        qa_ids = tokenizer.tokenize(clean_text(qa))
                        qa_ids = qa_ids[:max_qa_length]
                        ids, types, paddings = (
                            build_tokens_types_paddings_from_ids(
                                qa_ids, context_ids, max_seq_length,
                                tokenizer.cls, tokenizer.sep, tokenizer.pad,
                            )
                        )
                        ids_list.append(ids)
                        types_list.append(types)
                        paddings_list.append(paddings)

                    samples.append(build_sample(ids_list, types_list,
                                                paddings_list, label,
                                                len(samples)))
    print_rank_0(
        f"    > processed {num_docs} documents, {num_questions} questions, "
        f"{len(samples)} samples in {time.time() - start:.2f} seconds"
    )
    return samples
