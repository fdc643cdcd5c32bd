"""Generation API: broadcast sampling params across ranks, run the loop,
detokenize on the first stage (reference megatron/text_generation/api.py:19-201)."""

from __future__ import annotations

import torch

from .. import parallel as mpu
from .communication import broadcast_float_list
from .generation import (
    beam_search_and_return_on_first_stage,
    generate_tokens_probs_and_return_on_first_stage,
    score_and_return_on_first_stage,
)
from .tokenization import detokenize_generations, tokenize_prompts


def generate_and_post_process(
    model, prompts=None, tokens_to_generate=0, return_output_log_probs=False,
    top_k_sampling=0, top_p_sampling=0.0, temperature=1.0, add_BOS=False,
    use_eod_token_for_early_termination=True, stop_on_double_eol=False,
    stop_on_eol=False, prevent_newline_after_colon=False, random_seed=-1,
):
    """(reference api.py:19-66)"""
    tokens, lengths, output_log_probs = generate(
        model, prompts=prompts, tokens_to_generate=tokens_to_generate,
        return_output_log_probs=return_output_log_probs,
        top_k_sampling=top_k_sampling, top_p_sampling=top_p_sampling,
        temperature=temperature, add_BOS=add_BOS,
        use_eod_token_for_early_termination=use_eod_token_for_early_termination,
        stop_on_double_eol=stop_on_double_eol, stop_on_eol=stop_on_eol,
        prevent_newline_after_colon=prevent_newline_after_colon,
        random_seed=random_seed,
    )
    if mpu.is_pipeline_first_stage() and mpu.get_tensor_model_parallel_rank() == 0:
        tokens, prompts_plus_generations, segments = detokenize_generations(
            tokens, lengths, True
        )
        if return_output_log_probs:
            output_log_probs = output_log_probs.cpu().numpy().tolist()
            for i, (prob, seg) in enumerate(zip(output_log_probs, segments)):
                output_log_probs[i] = prob[: len(seg) - 1]
        return prompts_plus_generations, segments, output_log_probs, tokens
    return None


def generate(model, prompts=None, tokens_to_generate=0,
             return_output_log_probs=False, top_k_sampling=0,
             top_p_sampling=0.0, temperature=1.0, add_BOS=False,
             use_eod_token_for_early_termination=True,
             stop_on_double_eol=False, stop_on_eol=False,
             prevent_newline_after_colon=False, random_seed=-1):
    """(reference api.py:70-135)"""
    values = [
        tokens_to_generate, return_output_log_probs, top_k_sampling,
        top_p_sampling, temperature, add_BOS,
        use_eod_token_for_early_termination, stop_on_double_eol, stop_on_eol,
        prevent_newline_after_colon, random_seed,
    ]
    values_float_tensor = broadcast_float_list(
        len(values), float_list=values
    )
    tokens_to_generate = int(values_float_tensor[0].item())
    return_output_log_probs = bool(values_float_tensor[1].item())
    top_k_sampling = int(values_float_tensor[2].item())
    top_p_sampling = values_float_tensor[3].item()
    temperature = values_float_tensor[4].item()
    add_BOS = bool(values_float_tensor[5].item())
    use_eod_token_for_early_termination = bool(values_float_tensor[6].item())
    stop_on_double_eol = bool(values_float_tensor[7].item())
    stop_on_eol = bool(values_float_tensor[8].item())
    prevent_newline_after_colon = bool(values_float_tensor[9].item())
    random_seed = int(values_float_tensor[10].item())

    if random_seed != -1:
        torch.random.manual_seed(random_seed)

    context_tokens_tensor, context_length_tensor = tokenize_prompts(
        prompts=prompts, tokens_to_generate=tokens_to_generate
    )

    if tokens_to_generate == 0:
        return score_and_return_on_first_stage(
            model, context_tokens_tensor, context_length_tensor
        )

    return generate_tokens_probs_and_return_on_first_stage(
        model, context_tokens_tensor, context_length_tensor,
        return_output_log_probs=return_output_log_probs,
        top_k=top_k_sampling, top_p=top_p_sampling, temperature=temperature,
        use_eod_token_for_early_termination=use_eod_token_for_early_termination,
        stop_on_double_eol=stop_on_double_eol, stop_on_eol=stop_on_eol,
        prevent_newline_after_colon=prevent_newline_after_colon,
    )


def beam_search_and_post_process(model, prompts=None, tokens_to_generate=0,
                                 beam_size=0, add_BOS=False, stop_token=0,
                                 num_return_gen=1, length_penalty=1.0,
                                 prevent_newline_after_colon=False):
    """(reference api.py:139-201)"""
    tokens, scores = beam_search(
        model, prompts=prompts, tokens_to_generate=tokens_to_generate,
        beam_size=beam_size, add_BOS=add_BOS, stop_token=stop_token,
        num_return_gen=num_return_gen, length_penalty=length_penalty,
    )
    if mpu.is_pipeline_first_stage() and mpu.get_tensor_model_parallel_rank() == 0:
        lengths = tokens.new_tensor(
            [tokens.size(1)] * tokens.size(0)
        )
        tokens, prompts_plus_generations, segments = detokenize_generations(
            tokens, lengths, True
        )
        scores = scores.cpu().numpy().tolist()
        return prompts_plus_generations, segments, scores
    return None


def beam_search(model, prompts=None, tokens_to_generate=0, beam_size=0,
                add_BOS=False, stop_token=0, num_return_gen=1,
                length_penalty=1.0):
    values = [tokens_to_generate, beam_size, add_BOS, stop_token,
              num_return_gen, length_penalty]
    values_float_tensor = broadcast_float_list(len(values), float_list=values)
    tokens_to_generate = int(values_float_tensor[0].item())
    beam_size = int(values_float_tensor[1].item())
    add_BOS = bool(values_float_tensor[2].item())
    stop_token = int(values_float_tensor[3].item())
    num_return_gen = int(values_float_tensor[4].item())
    length_penalty = values_float_tensor[5].item()

    context_tokens_tensor, context_length_tensor = tokenize_prompts(
        prompts=prompts, tokens_to_generate=tokens_to_generate
    )
    return beam_search_and_return_on_first_stage(
        model, context_tokens_tensor, context_length_tensor, beam_size,
        stop_token=stop_token, num_return_gen=num_return_gen,
        length_penalty=length_penalty,
    )
