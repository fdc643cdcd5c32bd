"""Accuracy evaluation for classification finetuning (reference
tasks/eval_utils.py): per-dataset correct/total with a DP all-reduce,
wrapped as an end-of-epoch callback."""

from __future__ import annotations

import functools
import os
import time

import torch

from megatron_amd import parallel as mpu
from megatron_amd.config import get_config
from megatron_amd.parallel.schedules import get_forward_backward_func
from megatron_amd.utils import print_rank_0

from tasks import finetune_utils


def accuracy_func_provider(single_dataset_provider):
    """Build the end-of-epoch accuracy callback over cfg.valid_data."""
    cfg = get_config()
    dataloaders = []
    for datapath in cfg.valid_data:
        dataset = single_dataset_provider(datapath)
        dataloader = finetune_utils.build_data_loader(
            dataset,
            getattr(cfg, "orig_micro_batch_size", cfg.micro_batch_size),
            num_workers=cfg.num_workers,
            drop_last=(mpu.get_data_parallel_world_size() > 1),
        )
        dataloaders.append((dataset.dataset_name, dataloader))

    def metrics_func(model, epoch, output_predictions=False):
        print_rank_0("calculating metrics ...")
        correct = total = 0
        named_predictions = []
        names = "predictions"
        for name, dataloader in dataloaders:
            out = calculate_correct_answers(name, model, dataloader, epoch,
                                            output_predictions)
            if output_predictions:
                correct_ans, total_count, preds = out
                named_predictions.append((name, preds))
                names += "_" + name
            else:
                correct_ans, total_count = out
            correct += correct_ans
            total += total_count
        if total > 0:
            print_rank_0(
                f" >> |epoch: {epoch}| overall: correct / total = {correct} /"
                f" {total} = {100.0 * correct / total:.4f} %"
            )
        if output_predictions and cfg.load is not None:
            torch.save(named_predictions,
                       os.path.join(cfg.load, names + ".pt"))

    return metrics_func


def calculate_correct_answers(name, model, dataloader, epoch,
                              output_predictions=False):
    """correct/total over one dataloader; batch sizes adapt to the actual
    (possibly partial) last batch and the dataset's sample_multiplier."""
    cfg = get_config()
    start_time = time.time()
    for m in model:
        m.eval()
    saved_mbs = cfg.micro_batch_size
    saved_gbs = cfg.global_batch_size
    multiplier = getattr(dataloader.dataset, "sample_multiplier", 1)

    def loss_func(labels, batch, output_tensor):
        logits = output_tensor
        loss_dict = {}
        if output_predictions:
            loss_dict["softmaxes"] = torch.softmax(
                logits.float(), dim=-1
            ).cpu().numpy().tolist()
            loss_dict["labels"] = labels.cpu().numpy().tolist()
            loss_dict["ids"] = batch["uid"].cpu().numpy().tolist()
        predicted = torch.argmax(logits, dim=-1)
        loss_dict["total"] = labels.size(0)
        loss_dict["correct"] = (predicted == labels).sum().item()
        return torch.zeros(
            1, device=logits.device
        ).squeeze(), loss_dict

    def fwd(batch, model):
        try:
            batch_ = next(batch)
        except TypeError:
            batch_ = batch
        tokens, types, labels, attention_mask = finetune_utils.process_batch(
            batch_, cfg.fp16
        )
        output_tensor = model(tokens, attention_mask, tokentype_ids=types)
        return output_tensor, functools.partial(loss_func, labels, batch_)

    total = correct = 0
    softmaxes, labels_all, ids = [], [], []
    timers = None
    from megatron_amd import global_state
    timers = global_state.get_timers()
    with torch.no_grad():
        for batch in dataloader:
            actual = len(batch["label"])
            cfg.micro_batch_size = actual * multiplier
            cfg.global_batch_size = (
                actual * multiplier * mpu.get_data_parallel_world_size()
            )
            fb = get_forward_backward_func(cfg)
            loss_dicts = fb(fwd, batch, model, None, cfg, timers,
                            forward_only=True)
            for d in loss_dicts:
                if output_predictions:
                    softmaxes.extend(d["softmaxes"])
                    labels_all.extend(d["labels"])
                    ids.extend(d["ids"])
                total += d["total"]
                correct += d["correct"]
    for m in model:
        m.train()
    cfg.micro_batch_size = saved_mbs
    cfg.global_batch_size = saved_gbs

    if mpu.is_pipeline_last_stage():
        device = "cuda" if torch.cuda.is_available() else "cpu"
        unreduced = torch.tensor([correct, total], dtype=torch.long,
                                 device=device)
        torch.distributed.all_reduce(unreduced,
                                     group=mpu.get_data_parallel_group())
        correct_ans, total_count = unreduced[0].item(), unreduced[1].item()
        print_rank_0(
            f" > |epoch: {epoch}| metrics for {name}: correct / total = "
            f"{correct_ans} / {total_count} = "
            f"{100.0 * correct_ans / max(1, total_count):.4f} %, elapsed "
            f"time (sec): {time.time() - start_time:.3f}"
        )
        if output_predictions:
            return correct_ans, total_count, (softmaxes, labels_all, ids)
        return correct_ans, total_count
    return (0, 0, ()) if output_predictions else (0, 0)
