"""T5 span-corruption dataset (reference megatron/data/t5_dataset.py,
condensed): mask contiguous spans, replace with sentinel tokens, decoder
reconstructs the spans."""

from __future__ import annotations

import numpy as np
import torch

from ..global_state import get_tokenizer


class T5Dataset(torch.utils.data.Dataset):
    def __init__(self, name, indexed_dataset, documents, num_samples,
                 max_seq_length, max_seq_length_dec, masked_lm_prob, seed):
        self.name = name
        self.indexed_dataset = indexed_dataset
        self.documents = documents
        self.num_samples = max(1, num_samples)
        self.max_seq_length = max_seq_length
        self.max_seq_length_dec = max_seq_length_dec
        self.masked_lm_prob = masked_lm_prob
        self.seed = seed

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx):
        rng = np.random.RandomState((self.seed + idx) % 2 ** 31)
        tok = get_tokenizer()
        sentinels = tok.additional_special_tokens_ids
        assert sentinels, "T5 needs --vocab_extra_ids sentinel tokens"
        bos, eos, pad = tok.bos_token_id, tok.eos_token_id, tok.pad

        doc = np.asarray(
            self.indexed_dataset[
                self.documents[rng.randint(len(self.documents))]
            ]
        )
        max_enc = self.max_seq_length - 1
        tokens = doc[:max_enc]
        n = len(tokens)
        n_mask = max(1, int(n * self.masked_lm_prob))

        # sample span starts (mean span length 3)
        spans = []
        masked = np.zeros(n, dtype=bool)
        budget = n_mask
        tries = 0
        while budget > 0 and tries < 100:
            tries += 1
            ln = min(budget, max(1, int(rng.poisson(3))))
            start = rng.randint(0, max(1, n - ln))
            if masked[max(0, start - 1):min(n, start + ln + 1)].any():
                continue
            masked[start:start + ln] = True
            spans.append((start, ln))
            budget -= ln
        spans.sort()

        enc_tokens = []
        dec_tokens = [bos]
        dec_labels = []
        prev = 0
        for si, (start, ln) in enumerate(spans[: len(sentinels)]):
            sent = sentinels[si]
            enc_tokens.extend(tokens[prev:start])
            enc_tokens.append(sent)
            dec_tokens.append(sent)
            dec_labels.append(sent)
            dec_tokens.extend(tokens[start:start + ln])
            dec_labels.extend(tokens[start:start + ln])
            prev = start + ln
        enc_tokens.extend(tokens[prev:])
        dec_labels.append(eos)

        enc = np.full(self.max_seq_length, pad, dtype=np.int64)
        enc[: len(enc_tokens)] = enc_tokens[: self.max_seq_length]
        dec_in = np.full(self.max_seq_length_dec, pad, dtype=np.int64)
        dec_in[: len(dec_tokens)] = dec_tokens[: self.max_seq_length_dec]
        labels = np.full(self.max_seq_length_dec, -1, dtype=np.int64)
        labels[: len(dec_labels)] = dec_labels[: self.max_seq_length_dec]

        enc_mask = (enc != pad).astype(np.int64)
        dec_mask = (dec_in != pad).astype(np.int64)
        loss_mask = (labels != -1).astype(np.int64)
        labels[labels == -1] = pad
        return {
            "text_enc": enc,
            "text_dec": dec_in,
            "labels": labels,
            "loss_mask": loss_mask,
            "enc_mask": enc_mask,
            "dec_mask": dec_mask,
        }
