"""Token generation loop with KV cache, scoring and beam search
(reference megatron/text_generation/generation.py:20-429)."""

from __future__ import annotations

import torch
import torch.nn.functional as F

from .. import parallel as mpu
from ..config import get_config
from ..global_state import get_tokenizer
from ..utils import get_ltor_masks_and_position_ids
from .beam_utils import BeamHypotheses
from .communication import (
    broadcast_from_last_pipeline_stage,
    broadcast_from_last_to_first_pipeline_stage,
    copy_from_last_to_first_pipeline_stage,
)
from .forward_step import ForwardStep
from .sampling import sample


def _device():
    return (
        torch.cuda.current_device() if torch.cuda.is_available() else "cpu"
    )


def score_and_return_on_first_stage(model, tokens, lengths):
    """Forward the prompts and return log-probs (reference generation.py:20-86)."""
    cfg = get_config()
    batch_size = tokens.size(0)
    max_prompt_length = lengths.max().item()
    assert max_prompt_length == tokens.size(1)

    if max_prompt_length * batch_size > cfg.max_tokens_to_oom:
        raise ValueError(
            f"Too many tokens: {max_prompt_length * batch_size} > "
            f"{cfg.max_tokens_to_oom}"
        )

    forward_step = ForwardStep(model, batch_size, max_prompt_length)
    attention_mask, _, position_ids = get_ltor_masks_and_position_ids(
        tokens, 0, False, False, False
    )

    output_log_probs = None
    output_log_probs_size = (batch_size, max_prompt_length - 1)

    logits = forward_step(tokens, position_ids, attention_mask)
    if mpu.is_pipeline_last_stage():
        log_probs = F.log_softmax(logits, dim=2)
        output_log_probs = torch.gather(
            log_probs[:, :-1], 2, tokens[:, 1:].unsqueeze(2)
        ).squeeze(2)

    output_log_probs = broadcast_from_last_to_first_pipeline_stage(
        output_log_probs_size, torch.float32, output_log_probs
    )
    return tokens, lengths, output_log_probs


def generate_tokens_probs_and_return_on_first_stage(
    model, tokens, lengths, return_output_log_probs=False,
    top_k=0, top_p=0.0, temperature=1.0, use_eod_token_for_early_termination=True,
    stop_on_double_eol=False, stop_on_eol=False, prevent_newline_after_colon=False,
    echo_prompts=False,
):
    """Main generation loop (reference generation.py:89-285)."""
    cfg = get_config()
    tokenizer = get_tokenizer()

    batch_size = tokens.size(0)
    min_prompt_length = lengths.min().item()
    max_sequence_length = tokens.size(1)

    if max_sequence_length * batch_size > cfg.max_tokens_to_oom:
        raise ValueError(
            f"Too many tokens: {max_sequence_length * batch_size}"
        )

    forward_step = ForwardStep(model, batch_size, max_sequence_length)

    pad_id = getattr(tokenizer, "pad", None) or getattr(tokenizer, "eod", 0)
    termination_id = getattr(tokenizer, "eod", 0)

    # log probs of the generated sequence
    output_log_probs = None
    output_log_probs_size = (batch_size, max_sequence_length - 1)
    generated_sequence_lengths = None
    if mpu.is_pipeline_last_stage():
        if return_output_log_probs:
            output_log_probs = torch.empty(
                output_log_probs_size, dtype=torch.float32, device=_device()
            )
        generated_sequence_lengths = torch.ones(
            batch_size, dtype=torch.int64, device=_device()
        ) * max_sequence_length

    is_generation_done = torch.zeros(
        batch_size, dtype=torch.uint8, device=_device()
    )

    attention_mask, _, position_ids = get_ltor_masks_and_position_ids(
        tokens, 0, False, False, False
    )

    with torch.no_grad():
        prev_context_length = 0
        for context_length in range(min_prompt_length, max_sequence_length):
            tokens2use = tokens[:, prev_context_length:context_length]
            positions2use = position_ids[:, prev_context_length:context_length]
            attention_mask2use = attention_mask[
                ..., prev_context_length:context_length, :context_length
            ]

            logits = forward_step(tokens2use, positions2use,
                                  attention_mask2use)

            if mpu.is_pipeline_last_stage():
                if prevent_newline_after_colon:
                    # disable "\n" right after ":" (reference generation.py
                    # :191; token ids resolved through the live tokenizer)
                    colon = tokenizer.tokenize(":")[0]
                    newline = tokenizer.tokenize("\n")[0]
                    logits[tokens2use[:, -1] == colon, -1, newline] = -1e10
                last_token_logits = logits[:, -1, :]
                new_sample = sample(
                    last_token_logits.float(), top_k=top_k, top_p=top_p,
                    temperature=temperature,
                    vocab_size=getattr(tokenizer, "vocab_size", None),
                )
                # only update tokens past each prompt
                started = lengths <= context_length
                tokens[started, context_length] = new_sample[started]

                if return_output_log_probs:
                    log_probs = F.log_softmax(logits.float(), dim=2)
                    indices = torch.unsqueeze(
                        tokens[:, (prev_context_length + 1):(context_length + 1)],
                        2,
                    )
                    output_log_probs[
                        :, prev_context_length:context_length
                    ] = torch.gather(log_probs, 2, indices).squeeze(2)

            # send new tokens from last stage to first stage
            copy_from_last_to_first_pipeline_stage(
                batch_size, torch.int64, tokens[:, context_length]
            )
            prev_context_length = context_length

            # termination check. The eol-based stops use the GPT-2 BPE ids
            # the reference hard-codes (628 = "\n\n", 198 = "\n";
            # generation.py:244-251) — tokenizer-dependent by design.
            if mpu.is_pipeline_last_stage():
                if use_eod_token_for_early_termination:
                    new_tok = tokens[:, context_length]
                    if stop_on_double_eol:
                        done_token = ((new_tok == 628) | (
                            (new_tok == 198)
                            & (tokens[:, context_length - 1] == 198)
                        )) & started
                    elif stop_on_eol:
                        done_token = (
                            (new_tok == 628) | (new_tok == 198)
                        ) & started
                    else:
                        done_token = (new_tok == termination_id) & started
                    just_finished = done_token & (is_generation_done == 0)
                    generated_sequence_lengths[just_finished.bool()] = (
                        context_length + 1
                    )
                    is_generation_done = is_generation_done | (
                        just_finished.to(torch.uint8)
                    )
                    done = is_generation_done.all() & (
                        context_length >= lengths.max() - 1
                    )
                else:
                    done = torch.tensor(False, device=_device())
                done = done.to(torch.uint8).reshape(1)
            else:
                done = torch.zeros(1, dtype=torch.uint8, device=_device())
            done = broadcast_from_last_pipeline_stage(
                1, torch.uint8, done
            )
            if use_eod_token_for_early_termination and done is not None and (
                bool(done[0].item())
            ):
                break

    tokens = tokens[:, : (context_length + 1)]
    if mpu.is_pipeline_last_stage():
        if return_output_log_probs:
            output_log_probs = output_log_probs[:, :context_length]

    generated_sequence_lengths = broadcast_from_last_to_first_pipeline_stage(
        (batch_size,), torch.int64, generated_sequence_lengths
    )
    if return_output_log_probs:
        output_log_probs = broadcast_from_last_to_first_pipeline_stage(
            (batch_size, context_length), torch.float32, output_log_probs
        )
    return tokens, generated_sequence_lengths, output_log_probs


def beam_search_and_return_on_first_stage(model, tokens, lengths, beam_size,
                                          stop_token, num_return_gen,
                                          length_penalty):
    """Beam search (reference generation.py:288-429)."""
    cfg = get_config()
    tokenizer = get_tokenizer()
    batch_size = tokens.size(0)
    assert batch_size == 1
    prompt_length = lengths.item()
    final_sequence_length = tokens.size(1)
    final_sequence_length = min(final_sequence_length,
                                cfg.max_position_embeddings)

    if final_sequence_length * beam_size > cfg.max_tokens_to_oom:
        raise ValueError("Too many tokens for beam search")

    # expand to beam
    tokens = tokens.repeat(beam_size, 1)
    attention_mask, _, position_ids = get_ltor_masks_and_position_ids(
        tokens, 0, False, False, False
    )

    forward_step = ForwardStep(model, beam_size, final_sequence_length)

    beam_hyp = BeamHypotheses(beam_size, length_penalty)
    done = False
    scores = torch.zeros(beam_size, dtype=torch.float32,
                         device=_device()).unsqueeze(1)
    scores_size_tensor, tokens_size_tensor = None, None

    with torch.no_grad():
        prev_context_length = 0
        for context_length in range(prompt_length, final_sequence_length):
            tokens2use = tokens[:, prev_context_length:context_length]
            positions2use = position_ids[:, prev_context_length:context_length]
            attention_mask2use = attention_mask[
                ..., prev_context_length:context_length, :context_length
            ]

            logits = forward_step(tokens2use, positions2use,
                                  attention_mask2use)

            if mpu.is_pipeline_last_stage():
                vocab_size = logits.size(2)
                log_probs = F.log_softmax(logits.float(), dim=2)
                new_scores = log_probs[:, -1, :] + scores

                if context_length == prompt_length:
                    sorted_scores, indices = torch.sort(
                        new_scores[0, :], descending=True
                    )
                else:
                    sorted_scores, indices = torch.sort(
                        new_scores.view(-1), descending=True
                    )

                best_beam_ids = torch.div(
                    indices[:2 * beam_size], vocab_size
                ).trunc().long()
                best_words = indices[:2 * beam_size] % vocab_size
                best_scores = sorted_scores[:2 * beam_size]

                next_beams = []
                for beam_token_rank, (token_id, beam_score, beam_id) in (
                    enumerate(zip(best_words, best_scores, best_beam_ids))
                ):
                    if token_id.item() == stop_token:
                        is_beam_token_worse_than_top_num_beams = (
                            beam_token_rank >= beam_size
                        )
                        if is_beam_token_worse_than_top_num_beams:
                            continue
                        beam_hyp.add(
                            tokens[beam_id].clone(), beam_score,
                            context_length + 1 - prompt_length,
                        )
                    else:
                        next_beams.append((token_id, beam_score, beam_id))
                    if len(next_beams) == beam_size:
                        break

                if beam_hyp.is_done(best_scores.max().item(),
                                    context_length + 1 - prompt_length):
                    done = True

                best_batches = tokens.new_tensor(
                    [item[2] for item in next_beams]
                )
                tokens = tokens[best_batches, :]
                tokens[:, context_length] = tokens.new_tensor(
                    [item[0] for item in next_beams]
                )
                scores = scores.new_tensor(
                    [item[1] for item in next_beams]
                ).unsqueeze(1)

                if hasattr(forward_step.inference_params,
                           "swap_key_value_dict"):
                    forward_step.inference_params.swap_key_value_dict(
                        best_batches
                    )

            done_tensor = torch.tensor(
                [int(done)], dtype=torch.uint8, device=_device()
            )
            done_tensor = broadcast_from_last_pipeline_stage(
                1, torch.uint8, done_tensor
            )
            if bool(done_tensor[0].item()):
                break

            tokens = broadcast_from_last_pipeline_stage(
                tokens.size() if mpu.is_pipeline_last_stage() else
                (beam_size, final_sequence_length),
                torch.int64, tokens if mpu.is_pipeline_last_stage() else None,
            )
            prev_context_length = context_length

        if mpu.is_pipeline_last_stage():
            num_hyp = len(beam_hyp)
            if num_hyp < beam_size:
                for beam_id in range(beam_size - num_hyp):
                    beam_hyp.add(
                        tokens[beam_id].clone(), scores[beam_id].squeeze(),
                        context_length + 1 - prompt_length,
                    )
            sorted_hyps = sorted(beam_hyp.beams, key=lambda x: x[0],
                                 reverse=True)
            num_return_gen = min(num_return_gen, len(sorted_hyps))
            scores = [sorted_hyps[i][0] for i in range(num_return_gen)]
            tokens = [sorted_hyps[i][1] for i in range(num_return_gen)]
            scores = torch.stack(scores, dim=0)
            tokens = torch.stack(tokens, dim=0)
            scores_size_tensor = torch.tensor(
                scores.shape, dtype=torch.int64, device=_device()
            )
            tokens_size_tensor = torch.tensor(
                tokens.shape, dtype=torch.int64, device=_device()
            )

        scores_size_tensor = broadcast_from_last_pipeline_stage(
            1, torch.int64, scores_size_tensor
        )
        tokens_size_tensor = broadcast_from_last_pipeline_stage(
            2, torch.int64, tokens_size_tensor
        )
        scores = broadcast_from_last_to_first_pipeline_stage(
            tuple(scores_size_tensor.tolist()), torch.float32, scores
        )
        tokens = broadcast_from_last_to_first_pipeline_stage(
            tuple(tokens_size_tensor.tolist()), torch.int64, tokens
        )

    return tokens, scores
