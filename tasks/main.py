"""Downstream-task entry point (reference tasks/main.py): GLUE / RACE
finetuning and zero-shot GPT evaluation.

  python tasks/main.py --task MNLI --model_name bert ... (glue)
  python tasks/main.py --task RACE ...                    (race)
  python tasks/main.py --task LAMBADA|WIKITEXT103 ...     (zeroshot gpt)
"""

import os
import sys

sys.path.append(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from megatron_amd.config import get_config  # noqa: E402
from megatron_amd.initialize import initialize_megatron  # noqa: E402


def get_tasks_args(parser):
    group = parser.add_argument_group(title="tasks")
    group.add_argument("--task", type=str, required=True)
    group.add_argument("--epochs", type=int, default=3)
    group.add_argument("--pretrained_checkpoint", type=str, default=None)
    group.add_argument("--train_data", nargs="*", default=None)
    group.add_argument("--valid_data", nargs="*", default=None)
    group.add_argument("--overlapping_eval", type=int, default=32)
    return parser


if __name__ == "__main__":
    initialize_megatron(extra_args_provider=get_tasks_args)
    cfg = get_config()
    task = cfg.task.upper() if hasattr(cfg, "task") else None
    import argparse
    # task arg lives on the parsed namespace; re-parse quickly
    import sys as _sys
    task = None
    for i, a in enumerate(_sys.argv):
        if a == "--task" and i + 1 < len(_sys.argv):
            task = _sys.argv[i + 1].upper()
    if task in ("MNLI", "QQP", "COLA", "SST2"):
        from tasks.glue import main as glue_main

        glue_main(task)
    elif task == "RACE":
        from tasks.race import main as race_main

        race_main()
    elif task in ("LAMBADA", "WIKITEXT103"):
        from tasks.zeroshot_gpt import main as zeroshot_main

        zeroshot_main(task)
    else:
        raise NotImplementedError(f"task {task} is not implemented")
