"""BERT model (reference megatron/model/bert_model.py, 242 LoC): bidirectional
encoder + LM head (masked LM) + optional binary head (NSP)."""

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
from .norms import LayerNorm
from .utils_heads import get_linear_layer


def bert_extended_attention_mask(attention_mask):
    """[b, s] padding mask -> [b, 1, s, s] boolean 'masked' tensor
    (reference bert_model.py:23-38)."""
    attention_mask_b1s = attention_mask.unsqueeze(1)
    attention_mask_bs1 = attention_mask.unsqueeze(2)
    attention_mask_bss = attention_mask_b1s * attention_mask_bs1
    extended_attention_mask = attention_mask_bss.unsqueeze(1)
    return extended_attention_mask < 0.5


def bert_position_ids(token_ids):
    seq_length = token_ids.size(1)
    position_ids = torch.arange(seq_length, dtype=torch.long,
                                device=token_ids.device)
    return position_ids.unsqueeze(0).expand_as(token_ids)


class BertLMHead(MegatronModule):
    """Masked-LM head: dense + LN + vocab logits (reference bert_model.py:41-87)."""

    def __init__(self, mpu_vocab_size, hidden_size, cfg, parallel_output):
        super().__init__()
        self.bias = torch.nn.Parameter(torch.zeros(mpu_vocab_size))
        self.bias.model_parallel = True
        self.bias.partition_dim = 0
        self.bias.partition_stride = 1
        self.parallel_output = parallel_output

        self.dense = get_linear_layer(
            hidden_size, hidden_size, init_method_normal(cfg.init_method_std)
        )
        self.layernorm = LayerNorm(hidden_size, eps=cfg.layernorm_epsilon,
                                   params_dtype=cfg.params_dtype)
        self.gelu = torch.nn.functional.gelu

    def forward(self, hidden_states, word_embeddings_weight, cfg):
        hidden_states = self.dense(hidden_states)
        hidden_states = self.gelu(hidden_states)
        hidden_states = self.layernorm(hidden_states)
        output = parallel_lm_logits(
            hidden_states, word_embeddings_weight, self.parallel_output, cfg,
            bias=self.bias,
        )
        return output


def post_language_model_processing(lm_output, pooled_output, lm_head,
                                   binary_head, lm_labels, logit_weights, cfg):
    lm_logits = lm_head(lm_output, logit_weights, cfg)
    binary_logits = None
    if binary_head is not None and pooled_output is not None:
        binary_logits = binary_head(pooled_output)
    if lm_labels is None:
        return lm_logits.transpose(0, 1).contiguous(), binary_logits
    lm_labels = lm_labels.transpose(0, 1).contiguous()
    lm_loss = mpu.vocab_parallel_cross_entropy(lm_logits.float(), lm_labels)
    return lm_loss.transpose(0, 1).contiguous(), binary_logits


class Pooler(MegatronModule):
    """First-token pooler (reference language_model.py:56-130)."""

    def __init__(self, hidden_size, init_method):
        super().__init__()
        self.dense = get_linear_layer(hidden_size, hidden_size, init_method)

    def forward(self, hidden_states, sequence_index=0):
        # hidden_states [s, b, h]
        pooled = hidden_states[sequence_index, :, :]
        pooled = self.dense(pooled)
        pooled = torch.tanh(pooled)
        return pooled


class BertModel(MegatronModule):
    def __init__(self, cfg, num_tokentypes=0, add_binary_head=True,
                 parallel_output=True, pre_process=True, post_process=True):
        super().__init__(share_embeddings_and_output_weights=True)
        cfg.use_flash_attn = False  # bidirectional padding mask path
        cfg.finalize()
        self.cfg = cfg
        self.add_binary_head = add_binary_head
        self.parallel_output = parallel_output
        self.pre_process = pre_process
        self.post_process = post_process

        init_method = init_method_normal(cfg.init_method_std)
        scaled_init = scaled_init_method_normal(cfg.init_method_std,
                                                cfg.num_layers)

        self.language_model = TransformerLanguageModel(
            cfg, init_method, scaled_init,
            encoder_attn_mask_type=AttnMaskType.padding,
            pre_process=pre_process, post_process=post_process,
        )
        self._language_model_key = "language_model"

        if self.post_process:
            # reference language_model.py:448: the pooler optionally uses
            # xavier-uniform init
            pooler_init = (
                torch.nn.init.xavier_uniform_
                if cfg.init_method_xavier_uniform else init_method
            )
            self.pooler = Pooler(cfg.hidden_size, pooler_init)
            self.lm_head = BertLMHead(
                self.language_model.embedding.word_embeddings.weight.size(0)
                if pre_process else mpu.divide(
                    cfg.padded_vocab_size,
                    mpu.get_tensor_model_parallel_world_size(),
                ),
                cfg.hidden_size, cfg, parallel_output,
            )
            self._lm_head_key = "lm_head"
            if self.add_binary_head:
                self.binary_head = get_linear_layer(
                    cfg.hidden_size, 2, init_method
                )
                self._binary_head_key = "binary_head"

        if cfg.tie_embed_logits:
            self.initialize_word_embeddings(init_method_normal, cfg)

    def set_input_tensor(self, input_tensor):
        self.language_model.set_input_tensor(input_tensor)

    def forward(self, bert_model_input, attention_mask, tokentype_ids=None,
                lm_labels=None):
        extended_attention_mask = bert_extended_attention_mask(attention_mask)
        input_ids = bert_model_input
        position_ids = bert_position_ids(input_ids)

        lm_output = self.language_model(
            input_ids, position_ids, extended_attention_mask
        )

        if self.post_process:
            pooled_output = (
                self.pooler(lm_output) if self.add_binary_head else None
            )
            return post_language_model_processing(
                lm_output, pooled_output, self.lm_head,
                self.binary_head if self.add_binary_head else None,
                lm_labels, self.shared_embedding_or_output_weight(), self.cfg,
            )
        return lm_output
