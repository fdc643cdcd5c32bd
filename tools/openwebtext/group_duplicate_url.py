"""Group documents that share an identical URL (reference
tools/openwebtext/group_duplicate_url.py): exact-url duplicate groups in
the same `kept \\t dup...` format remove_group_duplicates consumes."""

import json
import sys
from collections import defaultdict


def group_by_url(input_file, output_file):
    seen = defaultdict(list)
    with open(input_file) as f:
        for i, line in enumerate(f):
            d = json.loads(line)
            seen[d["url"]].append(i)
    groups = 0
    with open(output_file, "w") as f:
        for url, rows in seen.items():
            if len(rows) > 1:
                f.write("\t".join([url] * len(rows)) + "\n")
                groups += 1
    print(f"{groups} exact-url duplicate groups")
    return groups


if __name__ == "__main__":
    group_by_url(sys.argv[1], sys.argv[2])
