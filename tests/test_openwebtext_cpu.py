"""Corpus-curation tools (tools/openwebtext/): URL blacklist, cleanup,
MinHash dedup, group removal, n-gram decontamination."""

import json
import os
import sys

sys.path.insert(0, os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    "tools", "openwebtext",
))


def test_blacklist_urls():
    from blacklist_urls import url_is_clean

    assert url_is_clean("https://example.com/article")
    assert not url_is_clean("https://www.youtube.com/watch?v=x")
    assert not url_is_clean("https://example.com/file.jpg")
    assert not url_is_clean("ftp://example.com/x")
    assert not url_is_clean("https://bad url.com/")


def test_cleanup_dataset(tmp_path):
    from cleanup_dataset import filter_corpus, looks_english

    assert looks_english("The quick brown fox " * 10)
    assert not looks_english("это полностью русский текст " * 10)

    inp = tmp_path / "in.jsonl"
    rows = [
        {"text": "word " * 200, "url": "u1"},            # kept
        {"text": "short doc", "url": "u2"},              # too small
        {"text": "текст на русском языке " * 80, "url": "u3"},  # non-english
    ]
    inp.write_text("\n".join(json.dumps(r) for r in rows) + "\n")
    out = tmp_path / "out.jsonl"
    stats = filter_corpus(str(inp), str(out))
    assert stats["written"] == 1
    assert stats["small"] == 1
    assert stats["non_english"] == 1


def test_minhash_dedup(tmp_path):
    from find_duplicates import find_duplicate_groups
    from minhash import jaccard
    from remove_group_duplicates import remove_duplicates

    base = ("the quick brown fox jumps over the lazy dog and runs far "
            "away into the deep dark forest tonight ") * 5
    docs = {
        "a": base,
        "b": base + " tiny tail difference",
        "c": "completely different content about graphics processors "
             "and matrix cores on modern accelerators " * 5,
    }
    assert jaccard(docs["a"], docs["b"]) > 0.8
    groups = find_duplicate_groups(docs, threshold=0.7)
    assert groups == [["a", "b"]]

    inp = tmp_path / "docs.jsonl"
    inp.write_text("\n".join(
        json.dumps({"url": u, "text": t}) for u, t in docs.items()
    ) + "\n")
    gf = tmp_path / "groups.txt"
    gf.write_text("a\tb\n")
    out = tmp_path / "dedup.jsonl"
    kept, total = remove_duplicates(str(gf), str(inp), str(out))
    assert (kept, total) == (2, 3)
    urls = [json.loads(l)["url"] for l in out.read_text().splitlines()]
    assert urls == ["a", "c"]


def test_filter_ngrams(tmp_path):
    from filter_ngrams import filter_corpus

    task = tmp_path / "task.jsonl"
    task.write_text(json.dumps(
        {"text": "the secret evaluation passage about unicorns"}
    ) + "\n")
    inp = tmp_path / "train.jsonl"
    rows = [
        {"text": "clean document " + "filler words " * 30, "url": "u1"},
        {"text": "contains the secret evaluation passage about unicorns "
                 "inside " + "pad " * 30, "url": "u2"},
    ]
    inp.write_text("\n".join(json.dumps(r) for r in rows) + "\n")
    out = tmp_path / "out.jsonl"
    kept, total = filter_corpus(str(inp), str(out), [str(task)], n=5)
    assert (kept, total) == (1, 2)
    assert json.loads(out.read_text())["url"] == "u1"


def test_cleanup_fix_dataset(tmp_path):
    """Short docs dropped, short javascript docs dropped, mojibake fixed,
    repeated punctuation collapsed; filtered docs carry their reason."""
    import json
    import subprocess
    import sys as _sys

    long_text = "All work and no play makes for dull text. " * 20  # > 512
    docs = [
        {"text": "too short"},
        {"text": "enable javascript to view this page please " * 3},
        {"text": long_text + " wow!!!!!!!!"},
    ]
    inp = tmp_path / "in.json"
    inp.write_text("\n".join(json.dumps(d) for d in docs) + "\n")
    out = str(tmp_path / "owt")
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [_sys.executable,
         os.path.join(repo, "tools", "openwebtext",
                      "cleanup_fix_dataset.py"),
         "--input_glob", str(inp), "--output_prefix", out],
        capture_output=True, text=True, timeout=120,
    )
    assert r.returncode == 0, r.stderr[-500:]
    kept = [json.loads(l) for l in
            open(out + "_cleaned.json").read().splitlines()]
    dropped = [json.loads(l) for l in
               open(out + "_filtered.json").read().splitlines()]
    assert len(kept) == 1 and len(dropped) == 2
    assert "!!!!" not in kept[0]["text"] and "!!!" in kept[0]["text"]
    # remove_512 is checked first, so both short docs drop with that reason
    assert all(d["filter_reason"] == ["remove_512"] for d in dropped)

    # with remove_512 disabled the javascript filter fires
    from tools.openwebtext.cleanup_fix_dataset import process_doc

    flags, _, _, filtered = process_doc(
        json.dumps({"text": "please enable javascript to continue"}),
        {"remove_256_javascript"},
    )
    assert filtered and flags["remove_256_javascript"]
