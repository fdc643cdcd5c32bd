"""Dataset-specific detokenizers (reference
tasks/zeroshot_gpt/detokenizer.py): undo the whitespace tokenization of the
published eval corpora before re-tokenizing with the model's tokenizer."""

from __future__ import annotations

import re


def ptb_detokenizer(s):
    s = s.replace(" '", "'")
    s = s.replace(" \n", "\n").replace("\n ", "\n")
    s = s.replace(" n't", "n't")
    s = s.replace(" N ", "1 ")
    s = s.replace("$ 1", "$1").replace("# 1", "#1")
    return s


def wikitext_detokenizer(s):
    s = s.replace("s '", "s'")
    s = re.sub(r"/' [0-9]/", r"/'[0-9]/", s)
    # wikitext escapes - , . inside numbers as @x@
    s = s.replace(" @-@ ", "-").replace(" @,@ ", ",").replace(" @.@ ", ".")
    for p in [":", ";", ".", "!", "?", ","]:
        s = s.replace(f" {p} ", f"{p} ")
    s = re.sub(r"\(\s*([^\)]*?)\s*\)", r"(\1)", s)
    s = re.sub(r"\[\s*([^\]]*?)\s*\]", r"[\1]", s)
    s = re.sub(r"{\s*([^}]*?)\s*}", r"{\1}", s)
    s = re.sub(r"\"\s*([^\"]*?)\s*\"", r'"\1"', s)
    s = re.sub(r"'\s*([^']*?)\s*'", r"'\1'", s)
    s = s.replace("= = = =", "====").replace("= = =", "===").replace(
        "= =", "==")
    s = s.replace(" " + chr(176) + " ", chr(176))
    s = s.replace(" \n", "\n").replace("\n ", "\n")
    s = s.replace(" N ", " 1 ").replace(" 's", "'s")
    return s


def lambada_detokenizer(s):
    return s


_DETOKENIZERS = {
    "ptb": ptb_detokenizer,
    "wiki": wikitext_detokenizer,
    "lambada": lambada_detokenizer,
}


def get_detokenizer(path):
    for key, fn in _DETOKENIZERS.items():
        if key in path:
            return fn
    return lambda s: s
