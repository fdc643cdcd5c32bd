"""Finetune / pretrain CLI for Llama / Llama-2 / Code-Llama / Falcon / Mistral / GPT.

Same flag surface as the reference entry point (reference finetune.py:1-270):

  torchrun --nproc_per_node 8 finetune.py --model_name llama2 \
      --tensor_model_parallel_size 4 --pipeline_model_parallel_size 1 \
      --data_path /path/tokenized --tokenizer_type SentencePieceTokenizer ...
"""

from __future__ import annotations

import torch

from megatron_amd import global_state
from megatron_amd.config import get_config
from megatron_amd.data.gpt_dataset import (
    build_train_valid_test_datasets as gpt_datasets,
)
from megatron_amd.data.instruction_dataset import (
    build_train_valid_test_datasets as instruct_datasets,
)
from megatron_amd.models import MODEL_CLASSES, ModelType
from megatron_amd.parallel import broadcast_data
from megatron_amd.training import pretrain
from megatron_amd.utils import (
    average_losses_across_data_parallel_group,
    get_ltor_masks_and_position_ids,
)


def model_provider(pre_process=True, post_process=True):
    cfg = get_config()
    model_cls = MODEL_CLASSES[cfg.model_name or "gpt"]
    model = model_cls(
        cfg, parallel_output=True, pre_process=pre_process,
        post_process=post_process,
    )
    return model


def get_batch(data_iterator):
    """(reference finetune.py:103-166). For instruction data the loss covers
    only assistant tokens, with --scalar_loss_mask weighting the rest."""
    cfg = get_config()
    instruction = cfg.model_type == "instruction"
    keys = ["text"] if not instruction else ["text", "assistant_mask",
                                             "pad_mask"]
    datatype = torch.int64
    if data_iterator is not None:
        data = next(data_iterator)
        if instruction and data is not None:
            data = {k: v.long() for k, v in data.items() if k in keys}
    else:
        data = None
    data_b = broadcast_data(keys, data, datatype)
    tokens_ = data_b["text"].long()
    labels = tokens_[:, 1:].contiguous()
    tokens = tokens_[:, :-1].contiguous()

    tokenizer_eod = 0
    try:
        tokenizer_eod = global_state.get_tokenizer().eod
    except AssertionError:
        pass
    attention_mask, loss_mask, position_ids = get_ltor_masks_and_position_ids(
        tokens, tokenizer_eod, cfg.reset_position_ids,
        cfg.reset_attention_mask, cfg.eod_mask_loss,
    )
    if instruction:
        # label at position i is token i+1 -> use its role/pad bits
        assistant = data_b["assistant_mask"][:, 1:].contiguous().float()
        pad = data_b["pad_mask"][:, 1:].contiguous().float()
        loss_mask = assistant + cfg.scalar_loss_mask * (pad - assistant)
    return tokens, labels, loss_mask, attention_mask, position_ids


def loss_func(loss_mask, output_tensor):
    losses = output_tensor.float()
    loss_mask = loss_mask.view(-1).float()
    loss = torch.sum(losses.view(-1) * loss_mask) / loss_mask.sum()
    averaged_loss = average_losses_across_data_parallel_group([loss])
    return loss, {"lm loss": averaged_loss[0]}


def forward_step(data_iterator, model):
    """(reference finetune.py:221-239)"""
    timers = global_state.get_timers()
    timers("batch-generator", log_level=2).start()
    tokens, labels, loss_mask, attention_mask, position_ids = get_batch(
        data_iterator
    )
    timers("batch-generator").stop()

    output_tensor = model(tokens, position_ids, attention_mask, labels=labels)
    import functools

    return output_tensor, functools.partial(loss_func, loss_mask)


def train_valid_test_datasets_provider(train_val_test_num_samples):
    cfg = get_config()
    if cfg.data_path is None and not (cfg.train_data_path
                                      or cfg.valid_data_path
                                      or cfg.test_data_path):
        from megatron_amd.data.synthetic import build_synthetic_datasets

        return build_synthetic_datasets(cfg, train_val_test_num_samples)
    builder = instruct_datasets if cfg.model_type == "instruction" else gpt_datasets
    kwargs = {}
    if builder is gpt_datasets:
        kwargs = dict(train_data_prefix=cfg.train_data_path,
                      valid_data_prefix=cfg.valid_data_path,
                      test_data_prefix=cfg.test_data_path)
    return builder(
        data_prefix=cfg.data_path,
        data_impl=cfg.data_impl,
        splits_string=cfg.split,
        train_valid_test_num_samples=train_val_test_num_samples,
        seq_length=cfg.seq_length + 1,
        seed=cfg.seed,
        skip_warmup=(not cfg.mmap_warmup),
        **kwargs,
    )


def extra_args(parser):
    return parser


if __name__ == "__main__":
    pretrain(
        train_valid_test_datasets_provider,
        model_provider,
        ModelType.encoder_or_decoder,
        forward_step,
        extra_args_provider=extra_args,
        args_defaults={"tokenizer_type": None},
    )
