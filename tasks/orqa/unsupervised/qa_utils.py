"""Open-retrieval QA matching utilities (reference
tasks/orqa/unsupervised/qa_utils.py, which follows facebookresearch/DPR —
CC-BY-NC licensed there, re-implemented here): does a retrieved passage
contain one of the reference answers, and top-k hit statistics.

The reference relies on DPR's regex SimpleTokenizer; here word tokens are
`\\w+` matches lowercased, which is what that tokenizer reduces to for
`words(uncased=True)`."""

from __future__ import annotations

import collections
import re
import string
import unicodedata


QAMatchStats = collections.namedtuple(
    "QAMatchStats", ["top_k_hits", "questions_doc_hits"]
)


def _normalize(text):
    return unicodedata.normalize("NFD", text)


def _words(text):
    return re.findall(r"\w+", text.lower())


def regex_match(text, pattern):
    try:
        compiled = re.compile(
            pattern, flags=re.IGNORECASE + re.UNICODE + re.MULTILINE
        )
    except re.error:
        return False
    return compiled.search(text) is not None


def has_answer(answers, text, match_type="string"):
    """True if the passage contains any reference answer — token-subsequence
    match for 'string', whole-text regex for 'regex'."""
    text = _normalize(text)
    if match_type == "string":
        words = _words(text)
        for answer in answers:
            answer_words = _words(_normalize(answer))
            if not answer_words:
                continue
            for i in range(len(words) - len(answer_words) + 1):
                if words[i : i + len(answer_words)] == answer_words:
                    return True
        return False
    if match_type == "regex":
        return any(regex_match(text, _normalize(a)) for a in answers)
    raise ValueError(match_type)


def check_answer(answers, doc_ids, all_docs, match_type="string"):
    """Per-retrieved-doc hit list for one question."""
    hits = []
    for doc_id in doc_ids:
        doc = all_docs.get(doc_id)
        if doc is None or doc[0] is None:
            hits.append(False)
            continue
        hits.append(has_answer(answers, doc[0], match_type))
    return hits


def calculate_matches(all_docs, answers, closest_docs, workers_num=1,
                      match_type="string"):
    """For each question, which of its top-k retrieved docs contain an
    answer; top_k_hits[i] counts questions answered within the top i+1.

    all_docs: {doc_id: (text, title)}; closest_docs: list of
    (doc_ids, scores) per question, aligned with `answers`."""
    n_docs = max((len(d[0]) for d in closest_docs), default=0)
    top_k_hits = [0] * n_docs
    questions_doc_hits = []
    for q_answers, (doc_ids, _scores) in zip(answers, closest_docs):
        hits = check_answer(q_answers, doc_ids, all_docs, match_type)
        questions_doc_hits.append(hits)
        best = next((i for i, h in enumerate(hits) if h), None)
        if best is not None:
            for k in range(best, n_docs):
                top_k_hits[k] += 1
    return QAMatchStats(top_k_hits, questions_doc_hits)


# -- reader-side answer validation (reference qa_utils.py:155-180) ----------

def _normalize_answer(s):
    s = s.lower()
    s = "".join(ch for ch in s if ch not in string.punctuation)
    s = re.sub(r"\b(a|an|the)\b", " ", s)
    return " ".join(s.split())


def exact_match_score(prediction, ground_truth):
    return _normalize_answer(prediction) == _normalize_answer(ground_truth)


def metric_max_over_ground_truths(metric_fn, prediction, ground_truths):
    return max(metric_fn(prediction, gt) for gt in ground_truths)
