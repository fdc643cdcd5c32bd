"""ORQA / retriever evaluation entry (reference
tasks/orqa/evaluate_orqa.py): embed evidence, retrieve for the NQ dev/test
questions, report top-k accuracies."""

from __future__ import annotations

from megatron_amd.config import get_config
from megatron_amd.utils import print_rank_0

from tasks.orqa.evaluate_utils import ORQAEvaluator


def main():
    cfg = get_config()
    print_rank_0("Starting retrieval evaluation!")
    evaluator = ORQAEvaluator()
    if cfg.qa_data_dev is not None:
        evaluator.evaluate(cfg.qa_data_dev, "DEV")
    if cfg.qa_data_test is not None:
        evaluator.evaluate(cfg.qa_data_test, "TEST")
