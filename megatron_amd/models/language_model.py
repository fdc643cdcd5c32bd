"""Embedding + transformer stack + LM head plumbing.

Reference: megatron/model/language_model.py (parallel_lm_logits :24-53,
Embedding :133-326, TransformerLanguageModel :329-638).
"""

from __future__ import annotations

import torch

from .. import parallel as mpu
from ..parallel import mappings
from .enums import AttnMaskType
from .module import MegatronModule
from .transformer import ParallelTransformer


def parallel_lm_logits(input_, word_embeddings_weight, parallel_output, cfg,
                       bias=None):
    """Logits = X @ E^T with E vocab-sharded (reference language_model.py:24-53)."""
    async_grad_allreduce = (
        not cfg.no_async_tensor_model_parallel_allreduce
        and mpu.get_tensor_model_parallel_world_size() > 1
        and not cfg.sequence_parallel
    )
    # When the async all-reduce path is taken, the linear's backward already
    # all-reduces grad_input across TP ranks; inserting the copy region too
    # would reduce twice (reference language_model.py:32-40).
    if async_grad_allreduce or cfg.sequence_parallel:
        input_parallel = input_
    else:
        input_parallel = mappings.copy_to_tensor_model_parallel_region(input_)

    logits_parallel = mpu.linear_with_grad_accumulation_and_async_allreduce(
        input_parallel, word_embeddings_weight, bias,
        cfg.gradient_accumulation_fusion, async_grad_allreduce,
        cfg.sequence_parallel,
    )
    if parallel_output:
        return logits_parallel
    return mappings.gather_from_tensor_model_parallel_region(logits_parallel)


def init_method_normal(sigma):
    def init_(tensor):
        return torch.nn.init.normal_(tensor, mean=0.0, std=sigma)

    return init_


def scaled_init_method_normal(sigma, num_layers):
    import math

    std = sigma / math.sqrt(2.0 * num_layers)

    def init_(tensor):
        return torch.nn.init.normal_(tensor, mean=0.0, std=std)

    return init_


class Embedding(MegatronModule):
    """Vocab-parallel word embeddings + optional absolute position embeddings
    (reference language_model.py:133-326)."""

    def __init__(self, cfg, init_method):
        super().__init__()
        self.cfg = cfg
        self.hidden_size = cfg.hidden_size
        self.init_method = init_method

        self.word_embeddings = mpu.VocabParallelEmbedding(
            cfg.padded_vocab_size, self.hidden_size, init_method=init_method,
            params_dtype=cfg.params_dtype,
            use_cpu_initialization=cfg.use_cpu_initialization,
            perform_initialization=cfg.perform_initialization,
        )
        self._word_embeddings_key = "word_embeddings"

        self.use_position_embeddings = cfg.position_embedding_type == "absolute"
        if self.use_position_embeddings:
            self.position_embeddings = torch.nn.Embedding(
                cfg.max_position_embeddings, self.hidden_size,
                dtype=cfg.params_dtype,
            )
            self._position_embeddings_key = "position_embeddings"
            if cfg.perform_initialization:
                self.init_method(self.position_embeddings.weight)
        self.fp32_residual_connection = cfg.fp32_residual_connection
        self.sequence_parallel = cfg.sequence_parallel
        self.embedding_dropout = torch.nn.Dropout(cfg.hidden_dropout)

    def zero_parameters(self):
        self.word_embeddings.weight.data.fill_(0)
        self.word_embeddings.weight.shared = True
        if self.use_position_embeddings:
            self.position_embeddings.weight.data.fill_(0)
            self.position_embeddings.weight.shared = True

    def forward(self, input_ids, position_ids):
        embeddings = self.word_embeddings(input_ids)
        if self.use_position_embeddings:
            embeddings = embeddings + self.position_embeddings(position_ids)

        # [b, s, h] -> [s, b, h]
        embeddings = embeddings.transpose(0, 1).contiguous()
        if self.fp32_residual_connection:
            embeddings = embeddings.float()

        if self.sequence_parallel:
            embeddings = mappings.scatter_to_sequence_parallel_region(embeddings)
            with mpu.get_cuda_rng_tracker().fork():
                embeddings = self.embedding_dropout(embeddings)
        else:
            embeddings = self.embedding_dropout(embeddings)
        return embeddings


class TransformerLanguageModel(MegatronModule):
    """Embedding + encoder (+ untied lm_head param on the last stage)
    (reference language_model.py:329-638)."""

    def __init__(self, cfg, init_method, output_layer_init_method,
                 encoder_attn_mask_type=AttnMaskType.causal,
                 pre_process=True, post_process=True):
        super().__init__()
        self.cfg = cfg
        self.pre_process = pre_process
        self.post_process = post_process
        self.encoder_attn_mask_type = encoder_attn_mask_type
        self.encoder_hidden_state = None

        if self.pre_process:
            self.embedding = Embedding(cfg, init_method)
            self._embedding_key = "embedding"

        self.encoder = ParallelTransformer(
            cfg, init_method, output_layer_init_method,
            self_attn_mask_type=encoder_attn_mask_type,
            pre_process=pre_process, post_process=post_process,
        )
        self._encoder_key = "encoder"

        # untied LM head parameter (reference language_model.py:437-457)
        if self.post_process and not cfg.tie_embed_logits:
            vocab_local = mpu.divide(
                cfg.padded_vocab_size, mpu.get_tensor_model_parallel_world_size()
            )
            device = (
                torch.cuda.current_device() if torch.cuda.is_available() else None
            )
            self.lm_head = torch.nn.Parameter(
                torch.empty(vocab_local, cfg.hidden_size, dtype=cfg.params_dtype,
                            device=device)
            )
            self.lm_head.model_parallel = True
            self.lm_head.partition_dim = 0
            self.lm_head.partition_stride = 1
            if cfg.perform_initialization:
                with mpu.get_cuda_rng_tracker().fork():
                    init_method(self.lm_head)

    def set_input_tensor(self, input_tensor):
        if not isinstance(input_tensor, list):
            input_tensor = [input_tensor]
        self.encoder.set_input_tensor(input_tensor[0])

    def forward(self, enc_input_ids, enc_position_ids, enc_attn_mask,
                inference_params=None):
        if self.pre_process:
            encoder_input = self.embedding(enc_input_ids, enc_position_ids)
        else:
            encoder_input = None

        encoder_output = self.encoder(
            encoder_input, enc_attn_mask, position_ids=enc_position_ids,
            inference_params=inference_params,
        )
        return encoder_output
