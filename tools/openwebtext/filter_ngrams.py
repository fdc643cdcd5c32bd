"""Task decontamination: drop (or trim) training documents that contain
n-grams from downstream eval sets (reference
tools/openwebtext/filter_ngrams.py, condensed to the core mechanism: build
a set of word n-grams from the task files, scan each training document,
remove documents with a match)."""

import argparse
import json
import re

_WORD_RE = re.compile(r"[a-z0-9']+")


def get_words(text):
    return _WORD_RE.findall(text.lower())


def build_task_ngrams(task_files, n):
    ngrams = set()
    for path in task_files:
        with open(path) as f:
            for line in f:
                line = line.strip()
                if not line:
                    continue
                try:
                    text = json.loads(line).get("text", "")
                except json.JSONDecodeError:
                    text = line
                words = get_words(text)
                if len(words) < n:
                    # short eval samples contribute their full text
                    if words:
                        ngrams.add(tuple(words))
                    continue
                for i in range(len(words) - n + 1):
                    ngrams.add(tuple(words[i : i + n]))
    return ngrams


def document_contaminated(text, task_ngrams, n):
    words = get_words(text)
    for i in range(max(0, len(words) - n + 1)):
        if tuple(words[i : i + n]) in task_ngrams:
            return True
    # short-sample full matches
    return tuple(words) in task_ngrams if len(words) < n else False


def filter_corpus(input_file, output_file, task_files, n):
    task_ngrams = build_task_ngrams(task_files, n)
    kept = total = 0
    with open(input_file) as fin, open(output_file, "w") as fout:
        for line in fin:
            total += 1
            doc = json.loads(line)
            if document_contaminated(doc.get("text", ""), task_ngrams, n):
                continue
            fout.write(line)
            kept += 1
    print(f"kept {kept}/{total} documents "
          f"({len(task_ngrams)} task n-grams, n={n})")
    return kept, total


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--tasks", nargs="+", required=True)
    p.add_argument("--input", required=True)
    p.add_argument("--output", required=True)
    p.add_argument("--ngram", type=int, default=13)
    a = p.parse_args()
    filter_corpus(a.input, a.output, a.tasks, a.ngram)


if __name__ == "__main__":
    main()
