"""Filter and clean loose-json documents: drop short docs, short docs that
contain javascript, and non-English docs; normalize mojibake; apply general
regex cleaning (reference tools/openwebtext/cleanup_fix_dataset.py:1-200).

Tasks (choose with --tasks, default all):
  remove_512              drop docs shorter than 512 characters
  remove_256_javascript   drop docs < 256 chars that mention javascript
  remove_512_non_english  drop docs < 512 chars not detected as English
  ftfy_fix_text           fix text encoding (ftfy if importable, else NFC)
  general_cleaning        collapse repeated punctuation/whitespace artifacts

The reference hard-depends on ftfy + langdetect; this environment is
offline, so both are optional: language detection falls back to an
ASCII-letter-ratio heuristic and text fixing to unicodedata NFC
normalization. Writes <out>_cleaned.json and <out>_filtered.json (the docs
that were removed, tagged with the reason).
"""

import argparse
import glob
import json
import re
import sys
import time
import unicodedata

try:
    import ftfy
except ImportError:
    ftfy = None
try:
    from langdetect import detect as _detect_lang
except ImportError:
    _detect_lang = None

ALL_TASKS = ["remove_512", "remove_256_javascript", "remove_512_non_english",
             "ftfy_fix_text", "general_cleaning"]


def looks_english(text: str) -> bool:
    if _detect_lang is not None:
        try:
            return _detect_lang(text[:2000]) == "en"
        except Exception:
            return False
    # offline heuristic: mostly ASCII letters/spaces => call it English
    sample = text[:2000]
    if not sample:
        return False
    ascii_letters = sum(ch.isascii() and (ch.isalpha() or ch.isspace())
                        for ch in sample)
    return ascii_letters / len(sample) > 0.75


def fix_text(text: str) -> str:
    if ftfy is not None:
        return ftfy.fix_text(text)
    return unicodedata.normalize("NFC", text)


_GENERAL_PATTERNS = [
    (re.compile(r"\n{3,}"), "\n\n"),          # >2 blank lines
    (re.compile(r"[ \t]{2,}"), " "),          # runs of spaces/tabs
    (re.compile(r"([!?.]){4,}"), r"\1\1\1"),  # !!!!!!! -> !!!
    (re.compile(r"-{6,}"), "-----"),          # ascii rules
]


def general_cleaning(text: str) -> str:
    for pat, rep in _GENERAL_PATTERNS:
        text = pat.sub(rep, text)
    return text


def process_doc(json_line: str, tasks) -> tuple:
    """Returns (flags, text, document, filtered)."""
    document = json.loads(json_line)
    text = document["text"]
    flags = {t: False for t in ALL_TASKS}

    if "remove_512" in tasks and len(text) < 512:
        flags["remove_512"] = True
        return flags, text, document, True
    if ("remove_256_javascript" in tasks and len(text) < 256
            and "javascript" in text.lower()):
        flags["remove_256_javascript"] = True
        return flags, text, document, True
    if ("remove_512_non_english" in tasks and len(text) < 512
            and not looks_english(text)):
        flags["remove_512_non_english"] = True
        return flags, text, document, True
    if "ftfy_fix_text" in tasks:
        fixed = fix_text(text)
        flags["ftfy_fix_text"] = fixed != text
        text = fixed
    if "general_cleaning" in tasks:
        cleaned = general_cleaning(text)
        flags["general_cleaning"] = cleaned != text
        text = cleaned
    document["text"] = text
    return flags, text, document, False


def process_file(path, tasks, out_cleaned, out_filtered, counts):
    with open(path, "r", encoding="utf-8") as f:
        for line in f:
            line = line.strip()
            if not line:
                continue
            flags, _, document, filtered = process_doc(line, tasks)
            for k, v in flags.items():
                counts[k] += int(v)
            if filtered:
                document["filter_reason"] = [k for k, v in flags.items() if v]
                out_filtered.write(json.dumps(document) + "\n")
            else:
                out_cleaned.write(json.dumps(document) + "\n")
            counts["filtered" if filtered else "kept"] += 1


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--input_glob", required=True,
                        help="glob of loose-json input files")
    parser.add_argument("--output_prefix", required=True)
    parser.add_argument("--tasks", nargs="+", default=ALL_TASKS,
                        choices=ALL_TASKS)
    args = parser.parse_args()

    files = sorted(glob.glob(args.input_glob))
    if not files:
        sys.exit(f"no input files match {args.input_glob}")

    counts = {t: 0 for t in ALL_TASKS}
    counts.update(kept=0, filtered=0)
    start = time.time()
    with open(args.output_prefix + "_cleaned.json", "w") as oc, \
            open(args.output_prefix + "_filtered.json", "w") as of:
        for path in files:
            process_file(path, set(args.tasks), oc, of, counts)
    print(f"done in {time.time() - start:.1f}s: {counts}", flush=True)


if __name__ == "__main__":
    main()
