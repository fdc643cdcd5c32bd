"""RACE finetuning entry (reference tasks/race/finetune.py): MultipleChoice
head, CE loss over the 4 collapsed choices, accuracy callback."""

from __future__ import annotations

from megatron_amd.config import get_config
from megatron_amd.global_state import get_tokenizer
from megatron_amd.utils import print_rank_0

from tasks.eval_utils import accuracy_func_provider
from tasks.finetune_utils import finetune
from tasks.race.data import RaceDataset


def train_valid_datasets_provider():
    cfg = get_config()
    tokenizer = get_tokenizer()
    train_dataset = RaceDataset("training", cfg.train_data, tokenizer,
                                cfg.seq_length)
    valid_dataset = RaceDataset("validation", cfg.valid_data, tokenizer,
                                cfg.seq_length)
    return train_dataset, valid_dataset


def model_provider(pre_process=True, post_process=True):
    cfg = get_config()
    from megatron_amd.models.classification import MultipleChoice

    print_rank_0("building multichoice model for RACE ...")
    return MultipleChoice(cfg, num_tokentypes=2, pre_process=pre_process,
                          post_process=post_process)


def single_dataset_provider(datapath):
    cfg = get_config()
    tokenizer = get_tokenizer()
    name = datapath.rstrip("/").split("/")[-1]
    return RaceDataset(name, [datapath], tokenizer, cfg.seq_length)


def metrics_func_provider():
    return accuracy_func_provider(single_dataset_provider)


def main():
    finetune(train_valid_datasets_provider, model_provider,
             end_of_epoch_callback_provider=metrics_func_provider)
