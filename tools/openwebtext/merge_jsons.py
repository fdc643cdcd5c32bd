"""Concatenate jsonl shards (reference tools/openwebtext/merge_jsons.py)."""

import argparse


def main():
    p = argparse.ArgumentParser()
    p.add_argument("inputs", nargs="+")
    p.add_argument("-o", "--output", required=True)
    a = p.parse_args()
    n = 0
    with open(a.output, "w") as fout:
        for path in a.inputs:
            with open(path) as fin:
                for line in fin:
                    if line.strip():
                        fout.write(line.rstrip("\n") + "\n")
                        n += 1
    print(f"merged {n} documents from {len(a.inputs)} shards")


if __name__ == "__main__":
    main()
