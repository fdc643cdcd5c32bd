"""GLUE finetuning entry (reference tasks/glue/finetune.py): builds the
Classification head over the BERT backbone and drives the shared finetune
loop with the accuracy callback."""

from __future__ import annotations

from megatron_amd.config import get_config
from megatron_amd.global_state import get_tokenizer
from megatron_amd.utils import print_rank_0

from tasks.eval_utils import accuracy_func_provider
from tasks.finetune_utils import finetune


def _glue_classification(num_classes, Dataset, name_from_datapath_func):
    def train_valid_datasets_provider():
        cfg = get_config()
        tokenizer = get_tokenizer()
        train_dataset = Dataset("training", cfg.train_data, tokenizer,
                                cfg.seq_length)
        valid_dataset = Dataset("validation", cfg.valid_data, tokenizer,
                                cfg.seq_length)
        return train_dataset, valid_dataset

    def model_provider(pre_process=True, post_process=True):
        cfg = get_config()
        from megatron_amd.models.classification import Classification

        print_rank_0("building classification model for {} ...".format(
            cfg.task))
        return Classification(cfg, num_classes=num_classes, num_tokentypes=2,
                              pre_process=pre_process,
                              post_process=post_process)

    def single_dataset_provider(datapath):
        cfg = get_config()
        tokenizer = get_tokenizer()
        return Dataset(name_from_datapath_func(datapath), [datapath],
                       tokenizer, cfg.seq_length)

    def metrics_func_provider():
        return accuracy_func_provider(single_dataset_provider)

    finetune(train_valid_datasets_provider, model_provider,
             end_of_epoch_callback_provider=metrics_func_provider)


def main(task):
    if task == "MNLI":
        from tasks.glue.mnli import MNLIDataset as Dataset

        def name_from_datapath(p):
            return p.split("MNLI")[-1].strip("/").replace("_", "-").replace(
                "/", "-").strip(".tsv") or "mnli"

        num_classes = 3
    elif task == "QQP":
        from tasks.glue.qqp import QQPDataset as Dataset

        def name_from_datapath(p):
            return p.split("QQP")[-1].strip("/").replace("_", "-").replace(
                "/", "-").strip(".tsv") or "qqp"

        num_classes = 2
    else:
        raise NotImplementedError(f"GLUE task {task} is not implemented")
    _glue_classification(num_classes, Dataset, name_from_datapath)
