"""Instruction-tuning preprocessing: jsonl conversations -> paired
<prefix>-text/<prefix>-role indexed datasets with the chat template
'<|im_start|>role\\n...<|im_end|>\\n' (reference
tools/preprocess_instruct_data.py:34-196, template :99-101).

Input jsonl rows: {"system": "...", "conversations":
[{"from": "human"|"gpt", "value": "..."}]} (or {"text","role"} pairs).
Role codes written per token: system=0, prompter=1, assistant=2.
"""

from __future__ import annotations

import argparse
import json
import multiprocessing
import os
import sys
import time

sys.path.append(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np  # noqa: E402

ROLE_SYSTEM, ROLE_PROMPTER, ROLE_ASSISTANT = 0, 1, 2


class Encoder:
    tokenizer = None

    def __init__(self, args):
        self.args = args

    def initializer(self):
        from megatron_amd.config import TrainingConfig
        from megatron_amd.tokenizer import build_tokenizer

        cfg = TrainingConfig(
            tokenizer_type=self.args.tokenizer_type,
            vocab_file=self.args.vocab_file,
            merge_file=self.args.merge_file,
            vocab_extra_ids_list=self.args.vocab_extra_ids_list,
            new_tokens=not self.args.no_new_tokens,
            rank=0, world_size=1,
            make_vocab_size_divisible_by=128,
        )
        Encoder.tokenizer = build_tokenizer(cfg)

    def encode(self, json_line):
        data = json.loads(json_line)
        tok = Encoder.tokenizer

        tokens = []
        roles = []

        def add_turn(role_name, role_code, content):
            text = f"<|im_start|>{role_name}\n{content}<|im_end|>\n"
            ids = tok.tokenize(text)
            tokens.extend(ids)
            roles.extend([role_code] * len(ids))

        if "system" in data and data["system"]:
            add_turn("system", ROLE_SYSTEM, data["system"])
        for turn in data.get("conversations", []):
            src = turn.get("from", turn.get("role", "human"))
            val = turn.get("value", turn.get("content", ""))
            if src in ("human", "user", "prompter"):
                add_turn("user", ROLE_PROMPTER, val)
            else:
                add_turn("assistant", ROLE_ASSISTANT, val)
        return tokens, roles, len(json_line)


def get_args():
    parser = argparse.ArgumentParser()
    parser.add_argument("--input", type=str, required=True)
    parser.add_argument("--tokenizer_type", type=str, required=True)
    parser.add_argument("--vocab_file", type=str, default=None)
    parser.add_argument("--merge_file", type=str, default=None)
    parser.add_argument(
        "--vocab_extra_ids_list", type=str,
        default="<|im_start|>,<|im_end|>",
    )
    parser.add_argument("--no_new_tokens", action="store_true")
    parser.add_argument("--output_prefix", type=str, required=True)
    parser.add_argument("--dataset_impl", type=str, default="mmap")
    parser.add_argument("--workers", type=int, default=1)
    parser.add_argument("--chunk_size", type=int, default=25)
    parser.add_argument("--log_interval", type=int, default=100)
    return parser.parse_args()


def main():
    args = get_args()
    from megatron_amd.data import indexed_dataset

    encoder = Encoder(args)
    pool = multiprocessing.Pool(args.workers, initializer=encoder.initializer)
    fin = open(args.input, "r", encoding="utf-8")
    encoded = pool.imap(encoder.encode, fin, args.chunk_size)

    text_builder = indexed_dataset.make_builder(
        f"{args.output_prefix}-text.bin", impl=args.dataset_impl,
        dtype=np.int32,
    )
    role_builder = indexed_dataset.make_builder(
        f"{args.output_prefix}-role.bin", impl=args.dataset_impl,
        dtype=np.int8,
    )

    proc_start = time.time()
    total_bytes = 0
    for i, (tokens, roles, nbytes) in enumerate(encoded, start=1):
        total_bytes += nbytes
        if not tokens:
            continue
        text_builder.add_item(np.array(tokens, dtype=np.int32))
        text_builder.end_document()
        role_builder.add_item(np.array(roles, dtype=np.int8))
        role_builder.end_document()
        if i % args.log_interval == 0:
            elapsed = time.time() - proc_start
            print(f"Processed {i} documents ({i / elapsed:.1f} docs/s).",
                  flush=True)

    text_builder.finalize(f"{args.output_prefix}-text.idx")
    role_builder.finalize(f"{args.output_prefix}-role.idx")
    print("done")


if __name__ == "__main__":
    main()
