"""Causal LM wrapper + model-family subclasses (Llama/Falcon/Mistral/Code-Llama).

Reference: megatron/model/gpt_model.py:18-123, llama_model.py:10-43,
falcon_model.py:10-41, mistral_model.py:10-46.
"""

from __future__ import annotations

import torch

from .. import parallel as mpu
from .enums import AttnMaskType
from .language_model import (
    TransformerLanguageModel,
    init_method_normal,
    parallel_lm_logits,
    scaled_init_method_normal,
)
from .module import MegatronModule


def post_language_model_processing(lm_output, labels, logit_weights,
                                   parallel_output, fp16_lm_cross_entropy, cfg):
    """(reference gpt_model.py:18-42)."""
    if labels is None:
        output = parallel_lm_logits(lm_output, logit_weights,
                                    parallel_output, cfg)
        # [s b v] -> [b s v]
        return output.transpose(0, 1).contiguous()

    # [b s] -> [s b]
    labels = labels.transpose(0, 1).contiguous()

    chunk = getattr(cfg, "loss_chunk_size", 0) or 0
    if chunk > 0 and lm_output.shape[0] > chunk:
        # chunk the head GEMM + CE over the sequence so the fp32 logits
        # transient is chunk-sized (a [32k, 32k-vocab] fp32 logits buffer
        # plus CE internals is tens of GB at long sequence); each chunk is
        # activation-checkpointed, so only lm_output stays resident
        import torch.utils.checkpoint as _ckpt

        def _chunk_loss(out_c, lab_c):
            logits_c = parallel_lm_logits(out_c, logit_weights,
                                          parallel_output, cfg)
            if fp16_lm_cross_entropy:
                return mpu.vocab_parallel_cross_entropy(logits_c, lab_c)
            return mpu.vocab_parallel_cross_entropy(logits_c.float(), lab_c)

        parts = []
        for s0 in range(0, lm_output.shape[0], chunk):
            parts.append(_ckpt.checkpoint(
                _chunk_loss, lm_output[s0 : s0 + chunk],
                labels[s0 : s0 + chunk], use_reentrant=False,
            ))
        loss = torch.cat(parts, dim=0)
        return loss.transpose(0, 1).contiguous()

    output = parallel_lm_logits(lm_output, logit_weights, parallel_output,
                                cfg)
    if fp16_lm_cross_entropy:
        assert output.dtype == torch.half
        loss = mpu.vocab_parallel_cross_entropy(output, labels)
    else:
        loss = mpu.vocab_parallel_cross_entropy(output.float(), labels)
    # [s b] -> [s b] ... [s b] -> [b s]
    return loss.transpose(0, 1).contiguous()


class GPTModel(MegatronModule):
    """Causal language model (reference gpt_model.py:45-123)."""

    def __init__(self, cfg, num_tokentypes=0, parallel_output=True,
                 pre_process=True, post_process=True,
                 model_type=None):
        super().__init__(
            share_embeddings_and_output_weights=cfg.tie_embed_logits
        )
        self.cfg = cfg
        self.parallel_output = parallel_output
        self.pre_process = pre_process
        self.post_process = post_process
        self.fp16_lm_cross_entropy = False

        self.language_model = TransformerLanguageModel(
            cfg,
            init_method_normal(cfg.init_method_std),
            scaled_init_method_normal(cfg.init_method_std, cfg.num_layers),
            encoder_attn_mask_type=AttnMaskType.causal,
            pre_process=pre_process,
            post_process=post_process,
        )
        self._language_model_key = "language_model"

        if cfg.tie_embed_logits:
            self.initialize_word_embeddings(init_method_normal, cfg)

    def set_input_tensor(self, input_tensor):
        self.language_model.set_input_tensor(input_tensor)

    def forward(self, input_ids, position_ids, attention_mask, labels=None,
                inference_params=None):
        lm_output = self.language_model(
            input_ids, position_ids, attention_mask,
            inference_params=inference_params,
        )

        if self.post_process:
            if self.cfg.tie_embed_logits:
                logit_weights = self.shared_embedding_or_output_weight()
            else:
                logit_weights = self.language_model.lm_head
            return post_language_model_processing(
                lm_output, labels, logit_weights, self.parallel_output,
                self.fp16_lm_cross_entropy, self.cfg,
            )
        return lm_output

    def state_dict_for_save_checkpoint(self, prefix="", keep_vars=False):
        state_dict_ = {}
        state_dict_[self._language_model_key] = (
            self.language_model.state_dict(prefix=prefix, keep_vars=keep_vars)
        )
        if (
            self.post_process
            and not self.pre_process
            and self.share_embeddings_and_output_weights
        ):
            state_dict_["word_embeddings_for_head"] = self.word_embeddings.state_dict(
                prefix=prefix, keep_vars=keep_vars
            )
        return state_dict_

    def load_state_dict(self, state_dict, strict=True):
        if (
            self.post_process
            and not self.pre_process
            and self.share_embeddings_and_output_weights
        ):
            self.word_embeddings.load_state_dict(
                state_dict["word_embeddings_for_head"], strict=strict
            )
        if self._language_model_key in state_dict:
            state_dict = state_dict[self._language_model_key]
        elif any(k.startswith("language_model.") for k in state_dict):
            # plain nn.Module state dict (flat keys)
            torch.nn.Module.load_state_dict(self, state_dict, strict=strict)
            return
        state_dict = _flatten_nested_language_model(state_dict)
        self.language_model.load_state_dict(state_dict, strict=strict)


def _flatten_nested_language_model(sd):
    """Accept reference-style nested checkpoints (megatron/language_model.py
    :556-580 stores {'embedding': {'word_embeddings': {'weight': ...}},
    'encoder': {...layer-local keys...}, 'lm_head'/'output_layer': ...}) in
    addition to this repo's flat key scheme."""
    import torch as _torch

    if not any(isinstance(v, dict) for v in sd.values()):
        return sd
    flat = {}

    def rec(prefix, d):
        for k, v in d.items():
            key = f"{prefix}.{k}" if prefix else k
            if isinstance(v, dict):
                rec(key, v)
            elif _torch.is_tensor(v):
                flat[key] = v

    rec("", sd)
    return flat


def _force(cfg, **kwargs):
    for k, v in kwargs.items():
        setattr(cfg, k, v)


class LlamaModel(GPTModel):
    """Llama / Llama-2 / Code-Llama architecture flags
    (reference llama_model.py:10-43): RoPE, RMSNorm, SwiGLU, no bias,
    untied embeddings."""

    def __init__(self, cfg, version=2, **kwargs):
        _force(
            cfg,
            use_rms_norm=True,
            use_bias=False,
            glu_activation="swiglu",
            position_embedding_type="rotary",
            tie_embed_logits=False,
            parallel_attn=False,
            parallel_layernorm=False,
        )
        cfg.finalize()
        super().__init__(cfg, **kwargs)


class CodeLlamaModel(LlamaModel):
    def __init__(self, cfg, **kwargs):
        cfg.rope_theta = 1e6
        super().__init__(cfg, version=2, **kwargs)


class FalconModel(GPTModel):
    """Falcon: parallel attention (+ parallel LN for 40B), GQA, rotary, LN
    (reference falcon_model.py:10-41)."""

    def __init__(self, cfg, **kwargs):
        _force(
            cfg,
            use_rms_norm=False,
            use_bias=False,
            glu_activation=None,
            position_embedding_type="rotary",
            tie_embed_logits=True,
            parallel_attn=True,
        )
        cfg.finalize()
        super().__init__(cfg, **kwargs)


class MistralModel(GPTModel):
    """Mistral-7B: Llama architecture + sliding-window attention
    (reference mistral_model.py:10-46)."""

    def __init__(self, cfg, **kwargs):
        _force(
            cfg,
            use_rms_norm=True,
            use_bias=False,
            glu_activation="swiglu",
            position_embedding_type="rotary",
            tie_embed_logits=False,
            parallel_attn=False,
            parallel_layernorm=False,
        )
        if cfg.sliding_window_size is None:
            cfg.sliding_window_size = 4096
        cfg.finalize()
        super().__init__(cfg, **kwargs)
