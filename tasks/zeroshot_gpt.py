"""Zero-shot GPT evaluation: perplexity on WikiText-103 / accuracy on
LAMBADA (reference tasks/zeroshot_gpt/evaluate.py, condensed)."""

import math
import os
import sys

import torch

sys.path.append(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from megatron_amd.checkpointing import load_checkpoint
from megatron_amd.config import get_config
from megatron_amd.global_state import get_tokenizer
from megatron_amd.models import MODEL_CLASSES, ModelType
from megatron_amd.training import get_model
from megatron_amd.utils import get_ltor_masks_and_position_ids, print_rank_0


def main(task):
    cfg = get_config()

    def model_provider(pre_process=True, post_process=True):
        model_cls = MODEL_CLASSES[cfg.model_name or "gpt"]
        return model_cls(cfg, parallel_output=False,
                         pre_process=pre_process, post_process=post_process)

    model = get_model(model_provider, ModelType.encoder_or_decoder,
                      wrap_with_ddp=False)
    if cfg.load:
        load_checkpoint(model, None, None, cfg)
    model = model[0]
    model.eval()

    tokenizer = get_tokenizer()
    valid = getattr(cfg, "valid_data", None)
    assert valid, "--valid_data required"
    text = open(valid[0], encoding="utf-8").read()
    tokens = tokenizer.tokenize(text)

    device = "cuda" if torch.cuda.is_available() else "cpu"
    seq = cfg.seq_length
    total_loss, total_tokens, total_correct = 0.0, 0, 0
    with torch.no_grad():
        for start in range(0, len(tokens) - 1, seq):
            chunk = tokens[start : start + seq + 1]
            if len(chunk) < 2:
                break
            inp = torch.tensor(chunk[:-1], device=device).unsqueeze(0)
            tgt = torch.tensor(chunk[1:], device=device).unsqueeze(0)
            am, _, pids = get_ltor_masks_and_position_ids(
                inp, tokenizer.eod, False, False, False
            )
            logits = model(inp, pids, am)
            lp = torch.log_softmax(logits.float(), dim=-1)
            nll = -lp.gather(2, tgt.unsqueeze(2)).squeeze(2)
            total_loss += nll.sum().item()
            total_tokens += tgt.numel()
            total_correct += (logits.argmax(-1) == tgt).sum().item()
    ppl = math.exp(total_loss / max(1, total_tokens))
    acc = total_correct / max(1, total_tokens)
    if task == "LAMBADA":
        print_rank_0(f"LAMBADA accuracy: {acc:.4f}")
    else:
        print_rank_0(f"{task} perplexity: {ppl:.4f}")
