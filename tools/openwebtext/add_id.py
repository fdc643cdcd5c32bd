"""Assign sequential ids to a loose-json corpus (reference
tools/openwebtext/add_id.py)."""

import json
import sys


def add_ids(input_file, output_file, start_id=0):
    i = start_id
    with open(input_file) as fin, open(output_file, "w") as fout:
        for line in fin:
            d = json.loads(line)
            d["id"] = i
            fout.write(json.dumps(d) + "\n")
            i += 1
    return i - start_id


if __name__ == "__main__":
    n = add_ids(sys.argv[1], sys.argv[2],
                int(sys.argv[3]) if len(sys.argv) > 3 else 0)
    print(f"wrote {n} documents")
