"""T5 encoder-decoder model (reference megatron/model/t5_model.py, 198 LoC)."""

from __future__ import annotations

import torch

from .. import parallel as mpu
from .enums import AttnMaskType, LayerType
from .language_model import (
    Embedding,
    init_method_normal,
    parallel_lm_logits,
    scaled_init_method_normal,
)
from .module import MegatronModule
from .transformer import ParallelTransformer


def t5_extended_attention_mask(attention_mask_list):
    def attn_mask_postprocess(attn_mask):
        extended = attn_mask.unsqueeze(1)
        return extended < 0.5

    return [attn_mask_postprocess(m) for m in attention_mask_list]


def t5_position_ids(token_ids):
    seq_length = token_ids.size(1)
    position_ids = torch.arange(seq_length, dtype=torch.long,
                                device=token_ids.device)
    return position_ids.unsqueeze(0).expand_as(token_ids)


class T5LMHead(MegatronModule):
    """(reference t5_model.py:30-55)"""

    def __init__(self, mpu_vocab_size, parallel_output):
        super().__init__()
        self.bias = torch.nn.Parameter(torch.zeros(mpu_vocab_size))
        self.bias.model_parallel = True
        self.bias.partition_dim = 0
        self.bias.partition_stride = 1
        self.parallel_output = parallel_output

    def forward(self, hidden_states, word_embeddings_weight, cfg):
        return parallel_lm_logits(
            hidden_states, word_embeddings_weight, self.parallel_output, cfg,
            bias=self.bias,
        )


class T5Model(MegatronModule):
    def __init__(self, cfg, num_tokentypes=0, parallel_output=True,
                 pre_process=True, post_process=True,
                 add_encoder=True, add_decoder=True):
        super().__init__(share_embeddings_and_output_weights=True)
        cfg.use_flash_attn = False
        cfg.finalize()
        self.cfg = cfg
        self.parallel_output = parallel_output
        self.pre_process = pre_process
        self.post_process = post_process
        self.add_encoder = add_encoder
        self.add_decoder = add_decoder

        init_method = init_method_normal(cfg.init_method_std)
        scaled_init = scaled_init_method_normal(cfg.init_method_std,
                                                cfg.num_layers)

        self.embedding = Embedding(cfg, init_method)
        if self.add_encoder:
            self.encoder = ParallelTransformer(
                cfg, init_method, scaled_init,
                self_attn_mask_type=AttnMaskType.padding,
                pre_process=pre_process, post_process=True,
            )
        if self.add_decoder:
            self.decoder = ParallelTransformer(
                cfg, init_method, scaled_init,
                layer_type=LayerType.decoder,
                self_attn_mask_type=AttnMaskType.causal,
                pre_process=pre_process, post_process=post_process,
            )
        if self.post_process:
            self.lm_head = T5LMHead(
                self.embedding.word_embeddings.weight.size(0),
                parallel_output,
            )

    def set_input_tensor(self, input_tensor):
        if self.add_encoder:
            self.encoder.set_input_tensor(input_tensor)
        elif self.add_decoder:
            self.decoder.set_input_tensor(input_tensor)

    def forward(self, encoder_input_ids, decoder_input_ids, encoder_attn_mask,
                decoder_attn_mask, encoder_decoder_attn_mask,
                tokentype_ids=None, lm_labels=None, enc_hidden_states=None):
        (
            encoder_attn_mask, decoder_attn_mask, encoder_decoder_attn_mask,
        ) = t5_extended_attention_mask(
            [encoder_attn_mask, decoder_attn_mask, encoder_decoder_attn_mask]
        )

        if enc_hidden_states is None:
            enc_position_ids = t5_position_ids(encoder_input_ids)
            enc_emb = self.embedding(encoder_input_ids, enc_position_ids)
            encoder_output = self.encoder(enc_emb, encoder_attn_mask)
        else:
            encoder_output = enc_hidden_states

        dec_position_ids = t5_position_ids(decoder_input_ids)
        dec_emb = self.embedding(decoder_input_ids, dec_position_ids)
        decoder_output = self.decoder(
            dec_emb, decoder_attn_mask,
            encoder_output=encoder_output,
            enc_dec_attn_mask=encoder_decoder_attn_mask,
        )

        if self.post_process:
            lm_logits = self.lm_head(
                decoder_output, self.embedding.word_embeddings.weight,
                self.cfg,
            )
            if lm_labels is None:
                return lm_logits.transpose(0, 1).contiguous()
            lm_labels = lm_labels.transpose(0, 1).contiguous()
            lm_loss = mpu.vocab_parallel_cross_entropy(
                lm_logits.float(), lm_labels
            )
            return lm_loss.transpose(0, 1).contiguous()
        return decoder_output
