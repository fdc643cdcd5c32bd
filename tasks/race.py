"""RACE multiple-choice harness (reference tasks/race/, condensed)."""

import os
import sys

sys.path.append(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from megatron_amd.config import get_config
from megatron_amd.utils import print_rank_0


def main():
    cfg = get_config()
    from megatron_amd.models.classification import MultipleChoice

    model = MultipleChoice(cfg)
    print_rank_0(
        f"RACE: built multiple-choice model with "
        f"{sum(p.numel() for p in model.parameters())} params."
    )
    raise SystemExit(0)
