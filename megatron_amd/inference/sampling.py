"""Token sampling: greedy / top-k / top-p with temperature
(reference megatron/text_generation/sampling.py:18-98)."""

from __future__ import annotations

import torch


def modify_logits_for_top_k_filtering(logits, top_k):
    filter_ = logits < torch.topk(logits, top_k)[0][..., -1, None]
    logits.masked_fill_(filter_, float("-inf"))


def modify_logits_for_top_p_filtering(logits, top_p):
    sorted_logits, sorted_indices = torch.sort(logits, descending=True)
    cumulative_probs = sorted_logits.softmax(dim=-1).cumsum(dim=-1)
    filter_ = cumulative_probs > top_p
    # shift so the first token above the threshold is kept
    filter_[:, 1:] = filter_[:, :-1].clone()
    filter_[..., 0] = 0
    filter_ = filter_.scatter(1, sorted_indices, filter_)
    logits.masked_fill_(filter_, float("-inf"))


def sample(logits, top_k=0, top_p=0.0, temperature=1.0, vocab_size=None):
    """Sample one token per row of logits [b, v]
    (reference sampling.py:45-98)."""
    assert logits.ndim == 2
    assert top_k == 0 or top_p == 0.0, "cannot set both top-k and top-p"

    if top_k == 1:  # greedy
        samples = torch.argmax(logits, dim=-1)
    else:
        logits = logits.clone()
        if temperature != 1.0:
            logits.div_(temperature)
        if top_k > 1:
            assert top_k <= logits.size(1)
            modify_logits_for_top_k_filtering(logits, top_k)
        elif top_p > 0.0:
            assert top_p <= 1.0
            modify_logits_for_top_p_filtering(logits, top_p)
        probs = logits.softmax(dim=-1)
        samples = torch.multinomial(probs, num_samples=1).view(-1)

    if vocab_size:
        samples = torch.clamp(samples, min=0, max=(vocab_size - 1))
    return samples
