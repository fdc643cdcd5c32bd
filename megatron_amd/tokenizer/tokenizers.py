"""Tokenizers: SentencePiece (Llama/Mistral), Falcon (HF), GPT-2 BPE.

Reference: megatron/tokenizer/tokenizer.py:12-400+. Vocab is padded to a
multiple of make_vocab_size_divisible_by * tensor_parallel_size (:49-62).
"""

from __future__ import annotations

from abc import ABC, abstractmethod



def build_tokenizer(cfg):
    if cfg.rank == 0:
        print(f"> building {cfg.tokenizer_type} tokenizer ...", flush=True)

    if cfg.tokenizer_type == "SentencePieceTokenizer":
        assert cfg.vocab_file is not None
        tokenizer = SentencePieceTokenizer(
            cfg.vocab_file,
            vocab_extra_ids=cfg.vocab_extra_ids,
            vocab_extra_ids_list=cfg.vocab_extra_ids_list,
            new_tokens=cfg.new_tokens,
        )
    elif cfg.tokenizer_type == "FalconTokenizer":
        tokenizer = FalconTokenizer(
            vocab_extra_ids_list=cfg.vocab_extra_ids_list,
            new_tokens=cfg.new_tokens,
        )
    elif cfg.tokenizer_type == "GPT2BPETokenizer":
        assert cfg.vocab_file is not None and cfg.merge_file is not None
        tokenizer = GPT2BPETokenizer(cfg.vocab_file, cfg.merge_file)
    elif cfg.tokenizer_type in ("BertWordPieceLowerCase",
                                "BertWordPieceCase"):
        assert cfg.vocab_file is not None
        tokenizer = BertWordPieceTokenizer(
            cfg.vocab_file,
            lower_case=(cfg.tokenizer_type == "BertWordPieceLowerCase"),
            vocab_extra_ids=cfg.vocab_extra_ids,
        )
    elif cfg.tokenizer_type == "FakeTokenizer":
        tokenizer = FakeTokenizer(cfg.padded_vocab_size or 32000)
    else:
        raise NotImplementedError(
            f"{cfg.tokenizer_type} tokenizer is not implemented."
        )

    if cfg.padded_vocab_size is None:
        cfg.padded_vocab_size = _vocab_size_with_padding(
            tokenizer.vocab_size, cfg
        )
    return tokenizer


def _vocab_size_with_padding(orig_vocab_size, cfg):
    """(reference tokenizer.py:49-62)"""
    after = orig_vocab_size
    multiple = (
        cfg.make_vocab_size_divisible_by * cfg.tensor_model_parallel_size
    )
    while (after % multiple) != 0:
        after += 1
    if cfg.rank == 0:
        print(
            f" > padded vocab (size: {orig_vocab_size}) with "
            f"{after - orig_vocab_size} dummy tokens (new size: {after})",
            flush=True,
        )
    return after


class AbstractTokenizer(ABC):
    def __init__(self, name):
        self.name = name
        super().__init__()

    @property
    @abstractmethod
    def vocab_size(self):
        ...

    @property
    @abstractmethod
    def vocab(self):
        ...

    @property
    @abstractmethod
    def inv_vocab(self):
        ...

    @abstractmethod
    def tokenize(self, text):
        ...

    def detokenize(self, token_ids):
        raise NotImplementedError(f"detokenizer is not implemented for {self.name}")

    @property
    def cls(self):
        raise NotImplementedError(f"CLS is not provided for {self.name}")

    @property
    def sep(self):
        raise NotImplementedError(f"SEP is not provided for {self.name}")

    @property
    def pad(self):
        raise NotImplementedError(f"PAD is not provided for {self.name}")

    @property
    def eod(self):
        raise NotImplementedError(f"EOD is not provided for {self.name}")

    @property
    def mask(self):
        raise NotImplementedError(f"MASK is not provided for {self.name}")


class FakeTokenizer(AbstractTokenizer):
    """Synthetic-data tokenizer for benchmarks/tests (no vocab files in the
    offline environment). vocab_extra_ids reserves T5 sentinel ids at the
    top of the vocab, mirroring SentencePieceTokenizer."""

    def __init__(self, vocab_size, vocab_extra_ids=0):
        super().__init__("FakeTokenizer")
        self._vocab_size = vocab_size
        self._extra_ids = vocab_extra_ids

    @property
    def vocab_size(self):
        return self._vocab_size

    @property
    def vocab(self):
        return {}

    @property
    def inv_vocab(self):
        return {}

    def tokenize(self, text):
        return [1] * max(1, len(text.split()))

    def detokenize(self, token_ids):
        return " ".join(str(t) for t in token_ids)

    @property
    def eod(self):
        return 0

    @property
    def pad(self):
        return 0

    @property
    def bos(self):
        return 1

    @property
    def eos(self):
        return 2

    @property
    def bos_token_id(self):
        return self.bos

    @property
    def eos_token_id(self):
        return self.eos

    @property
    def cls(self):
        return 3

    @property
    def sep(self):
        return 4

    @property
    def mask(self):
        return 5

    @property
    def additional_special_tokens_ids(self):
        return list(range(self._vocab_size - self._extra_ids,
                          self._vocab_size))


class SentencePieceTokenizer(AbstractTokenizer):
    """Llama's sentencepiece tokenizer with the reference's special tokens
    (reference tokenizer.py:326-500)."""

    def __init__(self, vocab_file, vocab_extra_ids=0, vocab_extra_ids_list=None,
                 new_tokens=True):
        super().__init__("SentencePieceTokenizer")
        import sentencepiece

        self._tokenizer = sentencepiece.SentencePieceProcessor(
            model_file=vocab_file
        )
        self._initalize(vocab_extra_ids, vocab_extra_ids_list, new_tokens)

    def _initalize(self, vocab_extra_ids, vocab_extra_ids_list, new_tokens):
        self._vocab = {}
        self._inv_vocab = {}
        self._special_tokens = {}
        self._inv_special_tokens = {}
        self._t5_tokens = []

        for i in range(len(self._tokenizer)):
            t = self._tokenizer.id_to_piece(i)
            self._inv_vocab[i] = t
            self._vocab[t] = i

        def _add_special_token(t, force=True):
            if t not in self._vocab:
                if not (force or new_tokens):
                    return
                next_id = len(self._vocab)
                self._vocab[t] = next_id
                self._inv_vocab[next_id] = t
            self._special_tokens[t] = self._vocab[t]
            self._inv_special_tokens[self._vocab[t]] = t

        _add_special_token("<CLS>", force=False)
        self._cls_id = self._vocab.get("<CLS>")
        _add_special_token("<SEP>", force=False)
        self._sep_id = self._vocab.get("<SEP>")
        _add_special_token("<EOD>", force=False)
        self._eod_id = self._vocab.get("<EOD>")
        _add_special_token("<MASK>", force=False)
        self._mask_id = self._vocab.get("<MASK>")

        pad_id = self._tokenizer.pad_id()
        try:
            pad_token = self._tokenizer.id_to_piece(pad_id)
        except IndexError:
            pad_token = "<PAD>"
        if new_tokens:
            _add_special_token(pad_token)
            self._pad_id = self._vocab[pad_token]
        else:
            self._pad_id = self._tokenizer.eos_id()

        bos_id = self._tokenizer.bos_id()
        try:
            bos_token = self._tokenizer.id_to_piece(bos_id)
        except IndexError:
            bos_token = "<BOS>"
        _add_special_token(bos_token)
        self._bos_id = self._vocab[bos_token]

        eos_id = self._tokenizer.eos_id()
        try:
            eos_token = self._tokenizer.id_to_piece(eos_id)
        except IndexError:
            eos_token = "<EOS>"
        _add_special_token(eos_token)
        self._eos_id = self._vocab[eos_token]

        for i in range(vocab_extra_ids):
            t = f"<extra_id_{i}>"
            _add_special_token(t)
            self._t5_tokens += [t]

        if vocab_extra_ids_list:
            for t in vocab_extra_ids_list.split(","):
                _add_special_token(t)

    @property
    def vocab_size(self):
        return len(self._vocab)

    @property
    def vocab(self):
        return self._vocab

    @property
    def inv_vocab(self):
        return self._inv_vocab

    def tokenize(self, text):
        ids = []
        idx = 0
        while 1:
            indices = {}
            for token in self._special_tokens:
                try:
                    indices[token] = text[idx:].index(token)
                except ValueError:
                    continue
            if len(indices) == 0:
                break
            next_token = min(indices, key=indices.get)
            next_idx = idx + indices[next_token]
            ids.extend(self._tokenizer.encode_as_ids(text[idx:next_idx]))
            ids.append(self._special_tokens[next_token])
            idx = next_idx + len(next_token)
        ids.extend(self._tokenizer.encode_as_ids(text[idx:]))
        return ids

    def detokenize(self, ids):
        text = ""
        last_i = 0
        for i, id in enumerate(ids):
            if id in self._inv_special_tokens:
                text += self._tokenizer.decode_ids(ids[last_i:i]) + " "
                text += self._inv_special_tokens[id] + " "
                last_i = i + 1
        text += self._tokenizer.decode_ids(ids[last_i:])
        return text

    @property
    def cls(self):
        return self._cls_id

    @property
    def sep(self):
        return self._sep_id

    @property
    def pad(self):
        return self._pad_id

    @property
    def bos_token_id(self):
        return self._bos_id

    @property
    def bos(self):
        return self._bos_id

    @property
    def eod(self):
        return self._eod_id if self._eod_id is not None else self._eos_id

    @property
    def eos_token_id(self):
        return self._eos_id

    @property
    def eos(self):
        return self._eos_id

    @property
    def mask(self):
        return self._mask_id

    @property
    def additional_special_tokens_ids(self):
        return [self.vocab[k] for k in self._t5_tokens]


class FalconTokenizer(AbstractTokenizer):
    """HF tiiuae/falcon tokenizer wrapper (reference tokenizer.py:288-323)."""

    def __init__(self, vocab_extra_ids_list=None, new_tokens=True):
        super().__init__("FalconTokenizer")
        from transformers import AutoTokenizer

        self._tokenizer = AutoTokenizer.from_pretrained("tiiuae/falcon-40b")
        if vocab_extra_ids_list and new_tokens:
            self._tokenizer.add_special_tokens(
                {
                    "additional_special_tokens": (
                        self._tokenizer.additional_special_tokens
                        + vocab_extra_ids_list.split(",")
                    )
                }
            )
        self._eod = self._tokenizer.vocab["<|endoftext|>"]

    @property
    def vocab_size(self):
        return len(self._tokenizer)

    @property
    def vocab(self):
        return self._tokenizer.vocab

    @property
    def inv_vocab(self):
        return {v: k for k, v in self._tokenizer.vocab.items()}

    def tokenize(self, text):
        return self._tokenizer.encode(text)

    def detokenize(self, token_ids):
        return self._tokenizer.decode(token_ids)

    @property
    def eod(self):
        return self._eod

    @property
    def eos_token_id(self):
        return self._eod

    @property
    def eos(self):
        return self._eod


class GPT2BPETokenizer(AbstractTokenizer):
    """GPT-2 BPE from local vocab/merges files (reference tokenizer.py:254-285)."""

    def __init__(self, vocab_file, merge_file):
        super().__init__("GPT2 BPE")
        from .gpt2_bpe import GPT2BPE

        self.tokenizer = GPT2BPE(vocab_file, merge_file)
        self.eod_id = self.tokenizer.encoder.get("<|endoftext|>", 0)

    @property
    def vocab_size(self):
        return len(self.tokenizer.encoder)

    @property
    def vocab(self):
        return self.tokenizer.encoder

    @property
    def inv_vocab(self):
        return self.tokenizer.decoder

    def tokenize(self, text):
        return self.tokenizer.encode(text)

    def detokenize(self, token_ids):
        return self.tokenizer.decode(token_ids)

    @property
    def eod(self):
        return self.eod_id


class BertWordPieceTokenizer(AbstractTokenizer):
    """Greedy longest-match-first WordPiece over a local vocab.txt
    (reference tokenizer.py:123-252 wrapping the original BERT tokenizer)."""

    def __init__(self, vocab_file, lower_case=True, vocab_extra_ids=0):
        name = "BERT Lower Case" if lower_case else "BERT Upper Case"
        super().__init__(name)
        self.lower_case = lower_case
        self._vocab = {}
        with open(vocab_file, encoding="utf-8") as f:
            for i, line in enumerate(f):
                self._vocab[line.rstrip("\n")] = i
        self._inv_vocab = {v: k for k, v in self._vocab.items()}
        for tok in ("[PAD]", "[UNK]", "[CLS]", "[SEP]", "[MASK]"):
            if tok not in self._vocab:
                self._vocab[tok] = len(self._vocab)
                self._inv_vocab[self._vocab[tok]] = tok
        self._cls_id = self._vocab["[CLS]"]
        self._sep_id = self._vocab["[SEP]"]
        self._pad_id = self._vocab["[PAD]"]
        self._mask_id = self._vocab["[MASK]"]
        self._unk_id = self._vocab["[UNK]"]
        self._additional = []
        for i in range(vocab_extra_ids):
            tok = f"<extra_id_{i}>"
            self._vocab[tok] = len(self._vocab)
            self._inv_vocab[self._vocab[tok]] = tok
            self._additional.append(self._vocab[tok])

    @property
    def vocab_size(self):
        return len(self._vocab)

    @property
    def vocab(self):
        return self._vocab

    @property
    def inv_vocab(self):
        return self._inv_vocab

    def _wordpiece(self, word):
        if word in self._vocab:
            return [self._vocab[word]]
        tokens = []
        start = 0
        while start < len(word):
            end = len(word)
            piece = None
            while start < end:
                sub = word[start:end]
                if start > 0:
                    sub = "##" + sub
                if sub in self._vocab:
                    piece = self._vocab[sub]
                    break
                end -= 1
            if piece is None:
                return [self._unk_id]
            tokens.append(piece)
            start = end
        return tokens

    def tokenize(self, text):
        import re

        if self.lower_case:
            text = text.lower()
        words = re.findall(r"\w+|[^\w\s]", text)
        ids = []
        for w in words:
            ids.extend(self._wordpiece(w))
        return ids

    def detokenize(self, token_ids):
        out = []
        for t in token_ids:
            piece = self._inv_vocab.get(int(t), "[UNK]")
            if piece.startswith("##") and out:
                out[-1] += piece[2:]
            else:
                out.append(piece)
        return " ".join(out)

    @property
    def cls(self):
        return self._cls_id

    @property
    def sep(self):
        return self._sep_id

    @property
    def pad(self):
        return self._pad_id

    @property
    def mask(self):
        return self._mask_id

    @property
    def eod(self):
        return self._sep_id

    @property
    def additional_special_tokens_ids(self):
        return list(self._additional)
