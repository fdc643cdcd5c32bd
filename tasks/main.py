"""Downstream-task entry point (reference tasks/main.py): GLUE / RACE
finetuning and zero-shot GPT evaluation.

  python tasks/main.py --task MNLI|QQP ...        (glue classification)
  python tasks/main.py --task RACE ...            (multi-choice)
  python tasks/main.py --task LAMBADA|WIKITEXT103 (zero-shot gpt eval)
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
    group.add_argument("--keep_last", action="store_true")
    group.add_argument("--strict_lambada", action="store_true")
    # msdp
    group.add_argument("--sample_input_file", type=str, default=None)
    group.add_argument("--sample_output_file", type=str, default=None)
    group.add_argument("--prompt_file", type=str, default=None)
    group.add_argument("--prompt_type", type=str, default=None,
                       choices=["knowledge", "response"])
    group.add_argument("--num_prompt_examples", type=int, default=10)
    group.add_argument("--guess_file", type=str, default=None)
    group.add_argument("--answer_file", type=str, default=None)
    group.add_argument("--out_seq_length", type=int, default=100)
    group.add_argument("--api_prompt", action="store_true")
    group.add_argument("--megatron_api_url", type=str, default=None)
    return parser


def dispatch(task):
    if task in ("MNLI", "QQP"):
        from tasks.glue.finetune import main as glue_main

        glue_main(task)
    elif task == "RACE":
        from tasks.race.finetune import main as race_main

        race_main()
    elif task in ("LAMBADA", "WIKITEXT103"):
        from tasks.zeroshot_gpt.evaluate import main as zeroshot_main

        zeroshot_main(task)
    elif task == "MSDP-PROMPT":
        from tasks.msdp.prompt import main as msdp_prompt_main

        msdp_prompt_main()
    elif task == "MSDP-EVAL-F1":
        from tasks.msdp.evaluate import main as msdp_eval_main

        msdp_eval_main()
    elif task in ("ICT-ZEROSHOT-NQ", "RETRIEVER-EVAL"):
        from tasks.orqa.evaluate_orqa import main as orqa_main

        orqa_main()
    elif task == "RET-FINETUNE-NQ":
        from tasks.orqa.supervised.finetune import main as ret_main

        ret_main()
    else:
        raise NotImplementedError(f"task {task} is not implemented")


if __name__ == "__main__":
    initialize_megatron(extra_args_provider=get_tasks_args)
    cfg = get_config()
    dispatch(cfg.task.upper())
