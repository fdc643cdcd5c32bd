"""Zero-shot GPT evaluation (reference tasks/zeroshot_gpt/evaluate.py):
WikiText-103 perplexity (summed masked NLL, adjusted for the
original/tokenized token ratio) and LAMBADA last-word accuracy (a sample
counts only if EVERY masked target token is predicted greedily)."""

from __future__ import annotations

import math

import torch

from megatron_amd import parallel as mpu
from megatron_amd.checkpointing import load_checkpoint
from megatron_amd.config import get_config
from megatron_amd.global_state import get_tokenizer
from megatron_amd.models import MODEL_CLASSES, ModelType
from megatron_amd.training import get_model
from megatron_amd.utils import (
    get_ltor_masks_and_position_ids,
    print_rank_0,
)

from tasks.finetune_utils import build_data_loader
from tasks.zeroshot_gpt.datasets import build_dataset


def _get_model_provider(eval_metric):
    def model_provider(pre_process=True, post_process=True):
        cfg = get_config()
        parallel_output = eval_metric == "loss"
        model_cls = MODEL_CLASSES[cfg.model_name or "gpt"]
        print_rank_0("building GPT model ...")
        return model_cls(cfg, parallel_output=parallel_output,
                         pre_process=pre_process, post_process=post_process)

    return model_provider


def process_batch(batch):
    cfg = get_config()
    tokenizer = get_tokenizer()
    device = "cuda" if torch.cuda.is_available() else "cpu"
    loss_mask = batch["pad_mask"].long().to(device).contiguous().byte()
    tokens_ = batch["text"].long().to(device).contiguous()
    labels = tokens_[:, 1:].contiguous()
    tokens = tokens_[:, :-1].contiguous()
    attention_mask, _, position_ids = get_ltor_masks_and_position_ids(
        tokens, tokenizer.eod, cfg.reset_position_ids,
        cfg.reset_attention_mask, cfg.eod_mask_loss,
    )
    return tokens, labels, attention_mask, position_ids, loss_mask


def forward_step(batch, model, eval_metric):
    cfg = get_config()
    tokens, labels, attention_mask, position_ids, loss_mask = process_batch(
        batch
    )
    cfg.micro_batch_size = len(labels)
    output = model(tokens, position_ids, attention_mask)

    if mpu.is_pipeline_last_stage():
        if eval_metric == "loss":
            # model output without labels is [b, s, v]; CE wants [s, b, v]
            losses = mpu.vocab_parallel_cross_entropy(
                output.transpose(0, 1).contiguous().float(),
                labels.transpose(0, 1).contiguous(),
            ).transpose(0, 1)
            return torch.sum(
                losses.reshape(-1) * loss_mask.reshape(-1).float()
            )
        if eval_metric == "accuracy":
            outputs = torch.argmax(output, -1)
            correct = (outputs == labels).float()
            correct[(1 - loss_mask).bool()] = 1
            return correct.prod(-1).sum()
        raise NotImplementedError(eval_metric)
    return None


def evaluate(data_loader, model, eval_metric):
    cfg = get_config()
    model.eval()
    total = 0.0
    with torch.no_grad():
        for it, batch in enumerate(data_loader):
            if it % cfg.log_interval == 0:
                print_rank_0(f"> working on iteration: {it}")
            output = forward_step(batch, model, eval_metric)
            if mpu.is_pipeline_last_stage():
                torch.distributed.all_reduce(
                    output, group=mpu.get_data_parallel_group()
                )
                total += output.item()
    return total


def evaluate_and_print_results(task, data_loader, model, eval_metric):
    output = evaluate(data_loader, model, eval_metric)
    string = f" validation results on {task} | "
    if eval_metric == "loss":
        n_tok = data_loader.dataset.num_tokenized_tokens
        n_orig = data_loader.dataset.num_original_tokens
        val_loss = output / (n_tok - 1)
        ppl = math.exp(min(20, val_loss))
        token_ratio = (n_tok - 1) / (n_orig - 1)
        adjusted_ppl = math.exp(min(20, val_loss * token_ratio))
        string += f"avg loss: {val_loss:.4E} | ppl: {ppl:.4E} | "
        string += f"adjusted ppl: {adjusted_ppl:.4E} | "
        string += f"token ratio: {token_ratio} |"
    else:
        n = len(data_loader.dataset)
        string += f"number correct: {output:.4E} | total examples: {n:.4E} | "
        string += f"avg accuracy: {output / n:.4E}"
    print_rank_0("-" * (len(string) + 1))
    print_rank_0(string)
    print_rank_0("-" * (len(string) + 1))
    return output


def main(task):
    cfg = get_config()
    eval_metric = "accuracy" if task == "LAMBADA" else "loss"
    model = get_model(_get_model_provider(eval_metric),
                      ModelType.encoder_or_decoder, wrap_with_ddp=False,
                      cfg=cfg)
    if cfg.load is not None:
        load_checkpoint(model, None, None, cfg)
    assert len(model) == 1
    model = model[0]

    dataset = build_dataset(task)
    dataloader = build_data_loader(dataset, cfg.micro_batch_size,
                                   cfg.num_workers, drop_last=False)
    evaluate_and_print_results(task, dataloader, model, eval_metric)
    print_rank_0("done :-)")
