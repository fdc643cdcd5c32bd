"""Classification / multiple-choice heads over the BERT encoder
(reference megatron/model/classification.py and multiple_choice.py)."""

from __future__ import annotations

import torch

from .bert_model import bert_extended_attention_mask, bert_position_ids
from .language_model import TransformerLanguageModel, init_method_normal, scaled_init_method_normal
from .bert_model import Pooler
from .enums import AttnMaskType
from .module import MegatronModule
from .utils_heads import get_linear_layer


class Classification(MegatronModule):
    def __init__(self, cfg, num_classes, num_tokentypes=2, pre_process=True,
                 post_process=True):
        super().__init__(share_embeddings_and_output_weights=False)
        cfg.use_flash_attn = False
        cfg.finalize()
        self.cfg = cfg
        self.num_classes = num_classes
        self.pre_process = pre_process
        self.post_process = post_process
        init_method = init_method_normal(cfg.init_method_std)

        self.language_model = TransformerLanguageModel(
            cfg, init_method,
            scaled_init_method_normal(cfg.init_method_std, cfg.num_layers),
            encoder_attn_mask_type=AttnMaskType.padding,
            pre_process=pre_process, post_process=post_process,
        )
        if self.post_process:
            self.pooler = Pooler(cfg.hidden_size, init_method)
            self.classification_dropout = torch.nn.Dropout(cfg.hidden_dropout)
            self.classification_head = get_linear_layer(
                cfg.hidden_size, num_classes, init_method
            )

    def set_input_tensor(self, input_tensor):
        self.language_model.set_input_tensor(input_tensor)

    def forward(self, model_input, attention_mask, tokentype_ids=None):
        extended_attention_mask = bert_extended_attention_mask(attention_mask)
        input_ids = model_input
        position_ids = bert_position_ids(input_ids)
        lm_output = self.language_model(
            input_ids, position_ids, extended_attention_mask
        )
        if self.post_process:
            pooled = self.pooler(lm_output)
            pooled = self.classification_dropout(pooled)
            return self.classification_head(pooled)
        return lm_output


class MultipleChoice(Classification):
    def __init__(self, cfg, num_tokentypes=2, pre_process=True,
                 post_process=True):
        super().__init__(cfg, num_classes=1, num_tokentypes=num_tokentypes,
                         pre_process=pre_process, post_process=post_process)

    def forward(self, model_input, attention_mask, tokentype_ids=None):
        # [b, num_choices, s] -> flatten, score each choice
        b, c, s = model_input.shape
        logits = super().forward(
            model_input.view(-1, s), attention_mask.view(-1, s),
            tokentype_ids.view(-1, s) if tokentype_ids is not None else None,
        )
        return logits.view(b, c)
