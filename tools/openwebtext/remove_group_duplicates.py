"""Drop the non-first member of every duplicate group (reference
tools/openwebtext/remove_group_duplicates.py)."""

import json
import sys


def remove_duplicates(group_file, input_file, output_file):
    drop = set()
    with open(group_file) as f:
        for line in f:
            urls = line.rstrip("\n").split("\t")
            drop.update(urls[1:])  # keep the first of each group
    kept = total = 0
    with open(input_file) as fin, open(output_file, "w") as fout:
        for line in fin:
            total += 1
            if json.loads(line)["url"] in drop:
                continue
            fout.write(line)
            kept += 1
    print(f"kept {kept}/{total} documents")
    return kept, total


if __name__ == "__main__":
    remove_duplicates(sys.argv[1], sys.argv[2], sys.argv[3])
