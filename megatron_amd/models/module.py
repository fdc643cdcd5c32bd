"""Base module: checkpoint-shaped state dicts, tied embeddings across pipeline
stages, fp16/bf16 IO conversion wrapper.

Reference: megatron/model/module.py:24-202.
"""

from __future__ import annotations

import torch
from torch.autograd import Variable
from torch.nn.parameter import Parameter

from .. import parallel as mpu

_FLOAT_TYPES = (torch.FloatTensor,)
_HALF_TYPES = (torch.HalfTensor,)
_BF16_TYPES = (torch.BFloat16Tensor,)


def param_is_not_shared(param):
    return not hasattr(param, "shared") or not param.shared


class MegatronModule(torch.nn.Module):
    """Adds word-embedding sharing between first and last pipeline stages
    (module.py:52-121)."""

    def __init__(self, config=None, share_embeddings_and_output_weights=True):
        super().__init__()
        self.config = config
        self.share_embeddings_and_output_weights = share_embeddings_and_output_weights

    def state_dict_for_save_checkpoint(self, prefix="", keep_vars=False):
        return self.state_dict(prefix=prefix, keep_vars=keep_vars)

    def shared_embedding_or_output_weight(self):
        if self.pre_process:
            return self.language_model.embedding.word_embeddings.weight
        else:
            if not self.share_embeddings_and_output_weights:
                raise Exception(
                    "shared_embedding_or_output_weight() called for last stage "
                    "but share_embeddings_and_output_weights is false"
                )
            return self.word_embeddings.weight

    def initialize_word_embeddings(self, init_method_normal, cfg):
        if not self.share_embeddings_and_output_weights:
            raise Exception("initialize_word_embeddings() requires tied weights")
        if cfg.pipeline_model_parallel_size == 1:
            return
        # last stage gets a copy of the word embeddings (module.py:73-106)
        if mpu.is_pipeline_last_stage() and not self.pre_process:
            assert not mpu.is_pipeline_first_stage()
            self._word_embeddings_for_head_key = "word_embeddings_for_head"
            from ..parallel import VocabParallelEmbedding

            self.word_embeddings = VocabParallelEmbedding(
                cfg.padded_vocab_size, cfg.hidden_size,
                init_method=init_method_normal(cfg.init_method_std),
                params_dtype=cfg.params_dtype,
                use_cpu_initialization=cfg.use_cpu_initialization,
                perform_initialization=cfg.perform_initialization,
            )
            self.word_embeddings.weight.data.fill_(0)
            self.word_embeddings.weight.shared = True

        if torch.distributed.is_initialized():
            if mpu.is_rank_in_embedding_group():
                torch.distributed.all_reduce(
                    self.shared_embedding_or_output_weight().data,
                    group=mpu.get_embedding_group(),
                )


def conversion_helper(val, conversion):
    if not isinstance(val, (tuple, list)):
        return conversion(val)
    rtn = [conversion_helper(v, conversion) for v in val]
    if isinstance(val, tuple):
        rtn = tuple(rtn)
    return rtn


def fp32_to_float16(val, float16_convertor):
    def half_conversion(val):
        if val is None or not torch.is_tensor(val):
            return val
        val_typecheck = val
        if isinstance(val_typecheck, (Parameter, Variable)):
            val_typecheck = val.data
        if val_typecheck.dtype == torch.float32 and val_typecheck.is_floating_point():
            val = float16_convertor(val)
        return val

    return conversion_helper(val, half_conversion)


def float16_to_fp32(val):
    def float_conversion(val):
        if val is None or not torch.is_tensor(val):
            return val
        val_typecheck = val
        if isinstance(val_typecheck, (Parameter, Variable)):
            val_typecheck = val.data
        if val_typecheck.dtype in (torch.float16, torch.bfloat16):
            val = val.float()
        return val

    return conversion_helper(val, float_conversion)


class Float16Module(MegatronModule):
    """Converts inputs to fp16/bf16 at the first pipeline stage and outputs
    back to fp32 at the last (module.py:160-202)."""

    def __init__(self, module, cfg):
        super().__init__()
        self.cfg = cfg
        if cfg.fp16:
            self.add_module("module", module.half())

            def float16_convertor(val):
                return val.half()

        elif cfg.bf16:
            self.add_module("module", module.bfloat16())

            def float16_convertor(val):
                return val.bfloat16()

        else:
            raise Exception("should not be called when fp16 and bf16 are both False")
        self.float16_convertor = float16_convertor

    def set_input_tensor(self, input_tensor):
        return self.module.set_input_tensor(input_tensor)

    def forward(self, *inputs, **kwargs):
        if mpu.is_pipeline_first_stage():
            inputs = fp32_to_float16(inputs, self.float16_convertor)
        outputs = self.module(*inputs, **kwargs)
        if mpu.is_pipeline_last_stage():
            outputs = float16_to_fp32(outputs)
        return outputs

    def state_dict(self, prefix="", keep_vars=False):
        return self.module.state_dict(prefix=prefix, keep_vars=keep_vars)

    def state_dict_for_save_checkpoint(self, prefix="", keep_vars=False):
        return self.module.state_dict_for_save_checkpoint(
            prefix=prefix, keep_vars=keep_vars
        )

    def load_state_dict(self, state_dict, strict=True):
        self.module.load_state_dict(state_dict, strict=strict)
