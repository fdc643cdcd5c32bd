"""jsonl -> tokenized binary indexed dataset (reference tools/preprocess_data.py:42-201).

Usage:
  python tools/preprocess_data.py --input data.jsonl --output_prefix out \
      --tokenizer_type SentencePieceTokenizer --vocab_file tok.model \
      --workers 8 --chunk_size 32 [--append_eod] [--split_sentences]
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
            vocab_extra_ids=self.args.vocab_extra_ids,
            vocab_extra_ids_list=self.args.vocab_extra_ids_list,
            new_tokens=not self.args.no_new_tokens,
            rank=0, world_size=1,
            make_vocab_size_divisible_by=128,
        )
        Encoder.tokenizer = build_tokenizer(cfg)

    def encode(self, json_line):
        data = json.loads(json_line)
        ids = {}
        for key in self.args.json_keys:
            text = data[key]
            doc_ids = Encoder.tokenizer.tokenize(text)
            if self.args.append_eod and doc_ids:
                doc_ids.append(Encoder.tokenizer.eod)
            ids[key] = [doc_ids] if doc_ids else []
        return ids, len(json_line)


def get_args():
    parser = argparse.ArgumentParser()
    group = parser.add_argument_group(title="input data")
    group.add_argument("--input", type=str, required=True)
    group.add_argument("--json_keys", nargs="+", default=["text"])
    group.add_argument("--split_sentences", action="store_true")

    group = parser.add_argument_group(title="tokenizer")
    group.add_argument("--tokenizer_type", type=str, required=True)
    group.add_argument("--vocab_file", type=str, default=None)
    group.add_argument("--merge_file", type=str, default=None)
    group.add_argument("--append_eod", action="store_true")
    group.add_argument("--vocab_extra_ids", type=int, default=0)
    group.add_argument("--vocab_extra_ids_list", type=str, default=None)
    group.add_argument("--no_new_tokens", action="store_true")

    group = parser.add_argument_group(title="output data")
    group.add_argument("--output_prefix", type=str, required=True)
    group.add_argument("--dataset_impl", type=str, default="mmap")

    group = parser.add_argument_group(title="runtime")
    group.add_argument("--workers", type=int, default=1)
    group.add_argument("--chunk_size", type=int, default=25)
    group.add_argument("--log_interval", type=int, default=100)
    return parser.parse_args()


def main():
    args = get_args()
    from megatron_amd.data import indexed_dataset

    encoder = Encoder(args)
    pool = multiprocessing.Pool(args.workers,
                                initializer=encoder.initializer)
    fin = open(args.input, "r", encoding="utf-8")
    encoded_docs = pool.imap(encoder.encode, fin, args.chunk_size)

    builders = {}
    output_bin_files = {}
    output_idx_files = {}
    for key in args.json_keys:
        output_bin_files[key] = f"{args.output_prefix}_{key}_document.bin"
        output_idx_files[key] = f"{args.output_prefix}_{key}_document.idx"
        builders[key] = indexed_dataset.make_builder(
            output_bin_files[key], impl=args.dataset_impl, dtype=np.int32
        )

    startup_end = time.time()
    proc_start = time.time()
    total_bytes_processed = 0
    print("Time to startup:", startup_end - proc_start)

    for i, (doc, bytes_processed) in enumerate(encoded_docs, start=1):
        total_bytes_processed += bytes_processed
        for key, sentences in doc.items():
            for sentence in sentences:
                builders[key].add_item(np.array(sentence, dtype=np.int32))
            builders[key].end_document()
        if i % args.log_interval == 0:
            current = time.time()
            elapsed = current - proc_start
            mbs = total_bytes_processed / elapsed / 1024 / 1024
            print(f"Processed {i} documents ({i / elapsed:.1f} docs/s, "
                  f"{mbs:.3f} MB/s).", flush=True)

    for key in args.json_keys:
        builders[key].finalize(output_idx_files[key])
    print("done")


if __name__ == "__main__":
    main()
