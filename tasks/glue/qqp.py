"""QQP dataset (reference tasks/glue/qqp.py). Train/dev TSVs have 6 columns
(id, qid1, qid2, question1, question2, is_duplicate); test TSVs have 3
(id, question1, question2) and take `test_label`. Malformed / empty rows are
skipped with a warning, matching the reference's tolerance."""

from __future__ import annotations

from megatron_amd.utils import print_rank_0

from tasks.data_utils import clean_text
from tasks.glue.data import GLUEAbstractDataset

LABELS = [0, 1]


class QQPDataset(GLUEAbstractDataset):
    def __init__(self, name, datapaths, tokenizer, max_seq_length,
                 test_label=0):
        self.test_label = test_label
        super().__init__("QQP", name, datapaths, tokenizer, max_seq_length)

    def process_samples_from_single_path(self, filename):
        print_rank_0(f" > Processing {filename} ...")
        samples = []
        is_test = False
        with open(filename) as f:
            for lineno, line in enumerate(f):
                row = line.strip().split("\t")
                if lineno == 0:
                    is_test = len(row) == 3
                    continue
                if is_test:
                    assert len(row) == 3, f"expected length 3: {row}"
                    uid = int(row[0].strip())
                    text_a = clean_text(row[1].strip())
                    text_b = clean_text(row[2].strip())
                    label = self.test_label
                else:
                    if len(row) != 6:
                        print_rank_0(f"***WARNING*** index error, "
                                     f"skipping: {row}")
                        continue
                    uid = int(row[0].strip())
                    text_a = clean_text(row[3].strip())
                    text_b = clean_text(row[4].strip())
                    label = int(row[5].strip())
                    if not text_a or not text_b:
                        print_rank_0(f"***WARNING*** zero length, "
                                     f"skipping: {row}")
                        continue
                assert label in LABELS and uid >= 0
                samples.append({
                    "text_a": text_a, "text_b": text_b,
                    "label": label, "uid": uid,
                })
        print_rank_0(f" >> processed {len(samples)} samples.")
        return samples
