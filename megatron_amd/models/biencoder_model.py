"""Biencoder (ICT / REALM retriever) model: twin BERT towers embedding
queries and context blocks (reference megatron/model/biencoder_model.py,
345 LoC, condensed to the trained surface)."""

from __future__ import annotations

import torch

from .bert_model import bert_extended_attention_mask, bert_position_ids
from .enums import AttnMaskType
from .language_model import (
    TransformerLanguageModel,
    init_method_normal,
    scaled_init_method_normal,
)
from .module import MegatronModule
from .utils_heads import get_linear_layer


class PretrainedBertModel(MegatronModule):
    """BERT tower + linear projection to the retrieval embedding space."""

    def __init__(self, cfg, num_tokentypes=2, projection_dim=128):
        super().__init__(share_embeddings_and_output_weights=False)
        self.cfg = cfg
        init_method = init_method_normal(cfg.init_method_std)
        self.language_model = TransformerLanguageModel(
            cfg, init_method,
            scaled_init_method_normal(cfg.init_method_std, cfg.num_layers),
            encoder_attn_mask_type=AttnMaskType.padding,
        )
        self.projection = get_linear_layer(
            cfg.hidden_size, projection_dim, init_method
        )

    def forward(self, input_ids, attention_mask, tokentype_ids=None):
        extended = bert_extended_attention_mask(attention_mask)
        position_ids = bert_position_ids(input_ids)
        lm_output = self.language_model(input_ids, position_ids, extended)
        # CLS pooling -> projection
        pooled = lm_output[0, :, :]
        return self.projection(pooled)


class BiEncoderModel(MegatronModule):
    """Query tower + context tower with optionally shared weights
    (reference biencoder_model.py:30-120)."""

    def __init__(self, cfg, num_tokentypes=2, projection_dim=128,
                 shared_query_context_model=False):
        super().__init__(share_embeddings_and_output_weights=False)
        self.shared = shared_query_context_model
        self.query_model = PretrainedBertModel(cfg, num_tokentypes,
                                               projection_dim)
        if shared_query_context_model:
            self.context_model = self.query_model
        else:
            import copy

            self.context_model = PretrainedBertModel(
                copy.deepcopy(cfg), num_tokentypes, projection_dim
            )

    def set_input_tensor(self, input_tensor):
        pass

    def embed_query(self, query_tokens, query_mask):
        return self.query_model(query_tokens, query_mask)

    def embed_context(self, context_tokens, context_mask):
        return self.context_model(context_tokens, context_mask)

    def forward(self, query_tokens, query_mask, context_tokens, context_mask):
        q = self.embed_query(query_tokens, query_mask)
        c = self.embed_context(context_tokens, context_mask)
        # in-batch negatives retrieval scores
        return torch.matmul(q, c.t())
