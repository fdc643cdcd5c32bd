"""Normalize and filter a loose-json corpus (reference
tools/openwebtext/cleanup_dataset.py: ftfy fix + langdetect + minimum
length). Substitutions for this offline image: unicode NFC + a common
mojibake repair pass instead of ftfy, and an ASCII-letter-ratio heuristic
instead of langdetect."""

import argparse
import json
import time
import unicodedata

MIN_DOCUMENT_WORDS = 128

_MOJIBAKE = {
    "â": "'", "â": "'",
    "â": '"', "â": '"',
    "â": "-", "â": "--",
    "â¦": "...", "Â ": " ",
}


def fix_text(text: str) -> str:
    for bad, good in _MOJIBAKE.items():
        text = text.replace(bad, good)
    return unicodedata.normalize("NFC", text)


def looks_english(text: str, threshold: float = 0.8) -> bool:
    if not text:
        return False
    sample = text[:4000]
    letters = sum(ch.isalpha() for ch in sample)
    ascii_letters = sum(ch.isalpha() and ch.isascii() for ch in sample)
    return letters > 0 and ascii_letters / letters >= threshold


def filter_corpus(filename, out_filename, min_words=MIN_DOCUMENT_WORDS):
    stats = {"docs": 0, "fixed": 0, "non_english": 0, "small": 0,
             "written": 0}
    start = time.time()
    with open(filename) as fin, open(out_filename, "w") as fout:
        for line in fin:
            stats["docs"] += 1
            try:
                doc = json.loads(line)
            except json.JSONDecodeError:
                continue
            text = fix_text(doc.get("text", ""))
            if text != doc.get("text"):
                stats["fixed"] += 1
            doc["text"] = text
            if not looks_english(text):
                stats["non_english"] += 1
                continue
            if len(text.split()) < min_words:
                stats["small"] += 1
                continue
            fout.write(json.dumps(doc) + "\n")
            stats["written"] += 1
    stats["elapsed"] = round(time.time() - start, 2)
    print(stats)
    return stats


def main():
    p = argparse.ArgumentParser()
    p.add_argument("input")
    p.add_argument("output")
    p.add_argument("--min_words", type=int, default=MIN_DOCUMENT_WORDS)
    a = p.parse_args()
    filter_corpus(a.input, a.output, a.min_words)


if __name__ == "__main__":
    main()
