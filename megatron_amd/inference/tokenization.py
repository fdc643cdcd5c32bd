"""Prompt tokenization + detokenization with cross-rank broadcast
(reference megatron/text_generation/tokenization.py)."""

from __future__ import annotations

import torch

from ..global_state import get_tokenizer
from .communication import broadcast_int_list, broadcast_tensor


def detokenize_generations(tokens_gpu_tensor, lengths_gpu_tensor,
                           return_segments):
    tokenizer = get_tokenizer()
    prompts_plus_generations = []
    prompts_plus_generations_segments = []

    tokens = tokens_gpu_tensor.cpu().numpy().tolist()
    lengths = lengths_gpu_tensor.cpu().numpy().tolist()
    for sequence_tokens, length in zip(tokens, lengths):
        sequence_tokens = sequence_tokens[:length]
        prompts_plus_generations.append(tokenizer.detokenize(sequence_tokens))
        if return_segments:
            words = [tokenizer.detokenize([token])
                     for token in sequence_tokens]
            prompts_plus_generations_segments.append(words)

    if return_segments:
        return tokens, prompts_plus_generations, (
            prompts_plus_generations_segments
        )
    return tokens, prompts_plus_generations


def tokenize_prompts(prompts=None, tokens_to_generate=None, rank=0):
    """Tokenize on rank 0, broadcast sizes then the padded token tensor."""
    sizes_list = None
    prompts_tokens_cuda_long_tensor = None
    prompts_length_cuda_long_tensor = None
    if torch.distributed.get_rank() == rank:
        assert prompts is not None and tokens_to_generate is not None
        (
            prompts_tokens_cuda_long_tensor,
            prompts_length_cuda_long_tensor,
        ) = _tokenize_prompts_and_batch(prompts, tokens_to_generate)
        sizes_list = [
            prompts_tokens_cuda_long_tensor.size(0),
            prompts_tokens_cuda_long_tensor.size(1),
        ]

    sizes_tensor = broadcast_int_list(2, int_list=sizes_list, rank=rank)
    sizes = sizes_tensor.tolist()
    prompts_tokens_cuda_long_tensor = broadcast_tensor(
        sizes, torch.int64, tensor=prompts_tokens_cuda_long_tensor, rank=rank
    )
    prompts_length_cuda_long_tensor = broadcast_tensor(
        sizes[0], torch.int64, tensor=prompts_length_cuda_long_tensor,
        rank=rank,
    )
    return prompts_tokens_cuda_long_tensor, prompts_length_cuda_long_tensor


def _tokenize_prompts_and_batch(prompts, tokens_to_generate):
    tokenizer = get_tokenizer()
    prompts_tokens = [tokenizer.tokenize(prompt) for prompt in prompts]
    prompts_length = [len(prompt_tokens) for prompt_tokens in prompts_tokens]
    max_prompt_len = max(prompts_length)
    samples_length = max_prompt_len + tokens_to_generate
    pad_id = getattr(tokenizer, "pad", 0) or 0
    for prompt_tokens, prompt_length in zip(prompts_tokens, prompts_length):
        padding_size = samples_length - prompt_length
        prompt_tokens.extend([pad_id] * padding_size)

    device = torch.cuda.current_device() if torch.cuda.is_available() else "cpu"
    prompts_tokens_tensor = torch.tensor(prompts_tokens, dtype=torch.int64,
                                         device=device)
    prompts_length_tensor = torch.tensor(prompts_length, dtype=torch.int64,
                                         device=device)
    return prompts_tokens_tensor, prompts_length_tensor
