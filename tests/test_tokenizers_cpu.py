"""Tokenizers over local files (no hub access): GPT-2 BPE merges, BERT
WordPiece greedy matching, FakeTokenizer special tokens, build_tokenizer
vocab padding."""

import json
import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


@pytest.fixture()
def gpt2_files(tmp_path):
    # a tiny BPE: characters + a couple of merges
    vocab = {"<|endoftext|>": 0}
    for i, ch in enumerate("helo wrd", start=1):
        vocab[ch] = i
    vocab["Ġ"] = 9  # byte-level BPE maps the space byte to this symbol
    vocab["he"] = 10
    vocab["ll"] = 11
    vocab["hell"] = 12
    vf = tmp_path / "vocab.json"
    vf.write_text(json.dumps(vocab))
    mf = tmp_path / "merges.txt"
    mf.write_text("#version\nh e\nl l\nhe ll\n")
    return str(vf), str(mf)


def test_gpt2_bpe(gpt2_files):
    from megatron_amd.tokenizer.tokenizers import GPT2BPETokenizer

    tok = GPT2BPETokenizer(*gpt2_files)
    ids = tok.tokenize("hello")
    assert ids, ids
    # merges applied: "hell" becomes one token
    assert tok.vocab["hell"] in ids
    assert "hell" in tok.detokenize(ids)
    assert tok.eod == tok.vocab["<|endoftext|>"]


def test_bert_wordpiece(tmp_path):
    from megatron_amd.tokenizer.tokenizers import BertWordPieceTokenizer

    vf = tmp_path / "vocab.txt"
    vf.write_text("\n".join([
        "[PAD]", "[UNK]", "[CLS]", "[SEP]", "[MASK]",
        "play", "##ing", "##ed", "the", "game",
    ]) + "\n")
    tok = BertWordPieceTokenizer(str(vf), lower_case=True, vocab_extra_ids=2)
    ids = tok.tokenize("playing the game")
    assert ids[:2] == [tok.vocab["play"], tok.vocab["##ing"]]
    assert tok.vocab["the"] in ids and tok.vocab["game"] in ids
    # greedy longest-match: "played" -> play + ##ed
    assert tok.tokenize("played") == [tok.vocab["play"], tok.vocab["##ed"]]
    # unknown word -> [UNK]
    assert tok.tokenize("zzz") == [tok._unk_id]
    assert tok.cls == tok.vocab["[CLS]"]
    assert tok.mask == tok.vocab["[MASK]"]
    assert len(tok.additional_special_tokens_ids) == 2


def test_build_tokenizer_pads_vocab(tmp_path):
    from megatron_amd.config import TrainingConfig
    from megatron_amd.tokenizer import build_tokenizer

    vf = tmp_path / "vocab.txt"
    vf.write_text("\n".join(f"tok{i}" for i in range(10)) + "\n")
    cfg = TrainingConfig(
        tokenizer_type="BertWordPieceLowerCase", vocab_file=str(vf),
        make_vocab_size_divisible_by=16,
    )
    cfg.finalize()
    tok = build_tokenizer(cfg)
    assert cfg.padded_vocab_size % 16 == 0
    assert cfg.padded_vocab_size >= tok.vocab_size


def test_preprocess_data_cli(tmp_path, gpt2_files):
    """tools/preprocess_data.py: jsonl -> .bin/.idx with the GPT-2 BPE
    tokenizer, tokens round-trip through the indexed dataset."""
    import subprocess

    import numpy as np

    from megatron_amd.data import indexed_dataset

    vf, mf = gpt2_files
    inp = tmp_path / "docs.jsonl"
    inp.write_text("\n".join(
        json.dumps({"text": t}) for t in ["hello world", "hell he herd"]
    ) + "\n")
    prefix = str(tmp_path / "corpus")
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, os.path.join(repo, "tools", "preprocess_data.py"),
         "--input", str(inp), "--output_prefix", prefix,
         "--tokenizer_type", "GPT2BPETokenizer",
         "--vocab_file", vf, "--merge_file", mf,
         "--append_eod", "--workers", "1", "--chunk_size", "2"],
        capture_output=True, text=True, timeout=300,
    )
    assert r.returncode == 0, r.stderr[-600:]

    from megatron_amd.tokenizer.tokenizers import GPT2BPETokenizer

    tok = GPT2BPETokenizer(vf, mf)
    ds = indexed_dataset.make_dataset(prefix + "_text_document", "infer")
    assert len(ds) == 2
    expected = tok.tokenize("hello world") + [tok.eod]
    assert list(np.asarray(ds[0])) == expected
