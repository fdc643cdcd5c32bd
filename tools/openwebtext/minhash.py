"""Tiny MinHash/LSH for near-duplicate detection (replaces the reference's
external mattilyra/LSH dependency, tools/openwebtext/find_duplicates.py):
5-char shingles -> k independent min-hashes -> banded LSH buckets."""

from __future__ import annotations

import hashlib
import struct


def shingles(text: str, char_ngram: int = 5):
    text = " ".join(text.split())
    return {text[i : i + char_ngram]
            for i in range(max(1, len(text) - char_ngram + 1))}


def _hash64(data: bytes, seed: int) -> int:
    h = hashlib.blake2b(data, digest_size=8, salt=seed.to_bytes(8, "little"))
    return struct.unpack("<Q", h.digest())[0]


def minhash_signature(text: str, num_perm: int = 64, char_ngram: int = 5):
    sh = shingles(text, char_ngram)
    if not sh:
        return tuple([0] * num_perm)
    enc = [s.encode() for s in sh]
    return tuple(
        min(_hash64(e, seed) for e in enc) for seed in range(num_perm)
    )


def lsh_buckets(signature, bands: int = 8):
    rows = len(signature) // bands
    for b in range(bands):
        yield b, hash(signature[b * rows : (b + 1) * rows])


def jaccard(text_a: str, text_b: str, char_ngram: int = 5) -> float:
    a, b = shingles(text_a, char_ngram), shingles(text_b, char_ngram)
    if not a or not b:
        return 0.0
    return len(a & b) / len(a | b)
