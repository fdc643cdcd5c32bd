"""Find near-duplicate documents by MinHash LSH (reference
tools/openwebtext/find_duplicates.py, on the in-repo minhash instead of the
external LSH package). Writes one line per duplicate group:
`kept_url \\t dup1 \\t dup2 ...` — candidates from shared LSH buckets are
confirmed by shingle jaccard."""

import argparse
import json
from collections import defaultdict

from minhash import jaccard, lsh_buckets, minhash_signature


def find_duplicate_groups(docs, num_perm=64, bands=8, threshold=0.7):
    """docs: {url: text} -> list of [kept, dup, dup...] groups."""
    buckets = defaultdict(list)
    for url, text in docs.items():
        sig = minhash_signature(text, num_perm)
        for key in lsh_buckets(sig, bands):
            buckets[key].append(url)

    parent = {u: u for u in docs}

    def find(u):
        while parent[u] != u:
            parent[u] = parent[parent[u]]
            u = parent[u]
        return u

    for urls in buckets.values():
        if len(urls) < 2:
            continue
        head = urls[0]
        for other in urls[1:]:
            if find(head) == find(other):
                continue
            if jaccard(docs[head], docs[other]) >= threshold:
                parent[find(other)] = find(head)

    groups = defaultdict(list)
    for u in docs:
        groups[find(u)].append(u)
    return [sorted(g) for g in groups.values() if len(g) > 1]


def main():
    p = argparse.ArgumentParser()
    p.add_argument("input")
    p.add_argument("output")
    p.add_argument("--threshold", type=float, default=0.7)
    a = p.parse_args()
    docs = {}
    with open(a.input) as f:
        for line in f:
            d = json.loads(line)
            docs[d["url"]] = d["text"]
    groups = find_duplicate_groups(docs, threshold=a.threshold)
    with open(a.output, "w") as f:
        for g in groups:
            f.write("\t".join(g) + "\n")
    print(f"{len(groups)} duplicate groups")


if __name__ == "__main__":
    main()
