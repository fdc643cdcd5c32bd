"""Merge multiple indexed datasets into one (reference tools/merge_datasets.py)."""

import argparse
import os
import sys

sys.path.append(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from megatron_amd.data import indexed_dataset


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--input", type=str, required=True,
                        help="directory containing .bin/.idx files to merge")
    parser.add_argument("--output_prefix", type=str, required=True)
    args = parser.parse_args()

    prefixes = set()
    for basename in sorted(os.listdir(args.input)):
        prefix, ext = os.path.splitext(basename)
        if prefix in prefixes or ext not in (".bin", ".idx"):
            continue
        if not os.path.isfile(os.path.join(args.input, basename)):
            continue
        other_ext = ".bin" if ext == ".idx" else ".idx"
        assert os.path.isfile(
            os.path.join(args.input, prefix + other_ext)
        ), f"missing pair for {basename}"
        prefixes.add(prefix)

    builder = None
    for prefix in sorted(prefixes):
        ds = indexed_dataset.MMapIndexedDataset(
            os.path.join(args.input, prefix)
        )
        if builder is None:
            builder = indexed_dataset.make_builder(
                args.output_prefix + ".bin", dtype=ds._index.dtype
            )
        del ds
        builder.merge_file_(os.path.join(args.input, prefix))

    builder.finalize(args.output_prefix + ".idx")
    print(f"merged {len(prefixes)} datasets into {args.output_prefix}")


if __name__ == "__main__":
    main()
