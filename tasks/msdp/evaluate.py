"""MSDP F1 evaluation (reference tasks/msdp/evaluate.py): token-level F1
between a file of generated lines and a file of gold lines."""

from __future__ import annotations

from megatron_amd.config import get_config
from megatron_amd.utils import print_rank_0

from tasks.msdp.metrics import F1Metric


def evaluate_f1(guess_file, answer_file):
    guesses = []
    print_rank_0(f"reading {guess_file}")
    with open(guess_file) as f:
        for line in f:
            line = line.strip().replace("<|endoftext|>", "")
            guesses.append(line)

    answers = []
    print_rank_0(f"reading {answer_file}")
    with open(answer_file) as f:
        for line in f:
            line = line.strip()
            if line == "no_passages_used":
                line = ""
            answers.append(line)

    assert len(guesses) == len(answers), (
        f"{len(guesses)} guesses vs {len(answers)} answers"
    )
    precision, recall, f1 = F1Metric.compute_all_pairs(guesses, answers)
    print_rank_0(f"Precision: {precision:.4f}; recall: {recall:.4f}; "
                 f"f1: {f1:.4f}")
    return precision, recall, f1


def main():
    cfg = get_config()
    evaluate_f1(cfg.guess_file, cfg.answer_file)
