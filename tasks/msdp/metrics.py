"""Dialogue metrics (reference tasks/msdp/metrics.py, in turn following
ParlAI's token-level F1): normalized unigram precision/recall/F1 between
generated and gold sentences."""

from __future__ import annotations

import re
from collections import Counter

import numpy as np

_RE_ARTICLE = re.compile(r"\b(a|an|the)\b")
_RE_PUNCT = re.compile(r"[!\"#$%&()*+,-./:;<=>?@\[\]\\^`{|}~_']")


def normalize_answer(s: str) -> str:
    """Lowercase; strip punctuation, articles and extra whitespace."""
    s = s.lower()
    s = _RE_PUNCT.sub(" ", s)
    s = _RE_ARTICLE.sub(" ", s)
    return " ".join(s.split())


class F1Metric:
    @staticmethod
    def _prec_recall_f1_score(pred_items, gold_items):
        common = Counter(gold_items) & Counter(pred_items)
        num_same = sum(common.values())
        if num_same == 0:
            return 0, 0, 0
        precision = num_same / len(pred_items)
        recall = num_same / len(gold_items)
        return precision, recall, 2 * precision * recall / (precision + recall)

    @staticmethod
    def compute_each_pair(guess: str, answer: str):
        if answer == "":
            return None, None, None
        if guess == "":
            return 0, 0, 0
        return F1Metric._prec_recall_f1_score(
            normalize_answer(guess).split(), normalize_answer(answer).split()
        )

    @staticmethod
    def compute_all_pairs(guesses, answers):
        assert len(guesses) == len(answers)
        ps, rs, f1s = [], [], []
        for guess, answer in zip(guesses, answers):
            p, r, f1 = F1Metric.compute_each_pair(guess, answer)
            if p is None:
                continue
            ps.append(p)
            rs.append(r)
            f1s.append(f1)
        return np.mean(ps), np.mean(rs), np.mean(f1s)
