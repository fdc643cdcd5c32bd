"""Mixed-precision optimizer stack.

Reference: megatron/optimizer/optimizer.py:100-783. Responsibilities:
 - grad reduction entry points called from train_step (reduce_model_grads):
   SP layernorm-grad all-reduce across TP, DP grad all-reduce (or
   reduce-scatter in the distributed optimizer), tied-embedding grad
   all-reduce across the embedding group;
 - fp16/bf16: fp32 master params, unscale + inf/nan check, grad clip, Adam,
   master -> model copy;
 - fp32 passthrough variant.
"""

from __future__ import annotations

from abc import ABC, abstractmethod

import torch

from .. import parallel as mpu
from ..models.module import param_is_not_shared
from ..utils import param_is_not_tensor_parallel_duplicate, unwrap_model
from .clip_grads import clip_grad_norm_fp32, count_zeros_fp32


def _zero_grad_group_helper(group, set_to_none):
    for param in group:
        if param.grad is not None:
            if set_to_none:
                param.grad = None
            else:
                if param.grad.grad_fn is not None:
                    param.grad.detach_()
                else:
                    param.grad.requires_grad_(False)
                param.grad.zero_()


def _multi_tensor_copy(src_list, dst_list):
    if not src_list:
        return
    with torch.no_grad():
        torch._foreach_copy_(dst_list, src_list)


class MegatronOptimizer(ABC):
    def __init__(self, optimizer, clip_grad, log_num_zeros_in_grad,
                 params_have_main_grad, use_contiguous_buffers_in_local_ddp,
                 models, cfg):
        self.optimizer = optimizer
        assert self.optimizer is not None
        self.clip_grad = clip_grad
        self.log_num_zeros_in_grad = log_num_zeros_in_grad
        self.params_have_main_grad = params_have_main_grad
        self.use_contiguous_buffers_in_local_ddp = use_contiguous_buffers_in_local_ddp
        if self.use_contiguous_buffers_in_local_ddp:
            assert self.params_have_main_grad
        self.models = models
        self.cfg = cfg

    def get_parameters(self):
        params = []
        for param_group in self.optimizer.param_groups:
            for param in param_group["params"]:
                params.append(param)
        return params

    def get_main_grads_for_grad_norm(self):
        grads_for_norm = []
        for param in self.get_parameters():
            grad = param.grad
            if (
                grad is not None
                and param_is_not_shared(param)
                and param_is_not_tensor_parallel_duplicate(param)
            ):
                grads_for_norm.append(grad)
        return grads_for_norm

    def get_model_parallel_group(self):
        return mpu.get_model_parallel_group()

    def _flat_grad_buffers(self):
        """The whole-model fp32 grad buffers, when they tile exactly the
        grads the norm covers: TP=PP=1 (no duplicate exclusions), contiguous
        local-DDP buffers, fp32 accumulation. Dead pad regions stay zero."""
        if (
            mpu.get_tensor_model_parallel_world_size() != 1
            or mpu.get_pipeline_model_parallel_world_size() != 1
            or not self.use_contiguous_buffers_in_local_ddp
        ):
            return None
        bufs = []
        for model in self.models:
            gb = getattr(model, "_grad_buffers", None)
            if not gb:
                return None
            for dtype, buf in gb.items():
                if dtype != torch.float:
                    return None
                bufs.append(buf.data)
        return bufs or None

    def clip_grad_norm(self, clip_grad, defer_scale=False):
        params = self.get_parameters()
        grads_for_norm = self.get_main_grads_for_grad_norm()
        return clip_grad_norm_fp32(
            params, grads_for_norm, clip_grad,
            model_parallel_group=self.get_model_parallel_group(),
            flat_buffers=self._flat_grad_buffers(),
            defer_scale=defer_scale,
        )

    def count_zeros(self):
        return count_zeros_fp32(
            self.get_parameters(),
            model_parallel_group=self.get_model_parallel_group(),
        )

    @abstractmethod
    def zero_grad(self, set_to_none=True):
        ...

    @abstractmethod
    def get_loss_scale(self):
        ...

    def scale_loss(self, loss):
        return self.get_loss_scale() * loss

    @abstractmethod
    def reload_model_params(self):
        ...

    @abstractmethod
    def state_dict(self):
        ...

    @abstractmethod
    def load_state_dict(self, state_dict):
        ...

    @property
    def state(self):
        return self.optimizer.state

    @state.setter
    def state(self, value):
        self.optimizer.state = value

    @property
    def param_groups(self):
        return self.optimizer.param_groups

    @param_groups.setter
    def param_groups(self, value):
        self.optimizer.param_groups = value

    @abstractmethod
    def step(self):
        ...

    # --- grad reduction (reference optimizer.py:203-301) ------------------

    def allreduce_word_embedding_grads(self):
        """Tied word-embedding grads must match on first and last PP stage."""
        from ..parallel.ddp import DistributedDataParallel as LocalDDP
        from ..models.module import Float16Module

        if (
            mpu.is_rank_in_embedding_group(ignore_virtual=True)
            and mpu.get_pipeline_model_parallel_world_size() > 1
        ):
            if mpu.is_pipeline_first_stage(ignore_virtual=True):
                unwrapped_model = self.models[0]
            elif mpu.is_pipeline_last_stage(ignore_virtual=True):
                unwrapped_model = self.models[-1]
            else:  # pipeline split (T5)
                unwrapped_model = self.models[0]
            unwrapped_model = unwrap_model(unwrapped_model, (LocalDDP, Float16Module))
            if unwrapped_model.share_embeddings_and_output_weights:
                word_embeddings_weight = (
                    unwrapped_model.shared_embedding_or_output_weight()
                )
                if self.params_have_main_grad:
                    grad = word_embeddings_weight.main_grad
                else:
                    grad = word_embeddings_weight.grad
                torch.distributed.all_reduce(grad, group=mpu.get_embedding_group())

    def allreduce_position_embedding_grads(self):
        from ..parallel.ddp import DistributedDataParallel as LocalDDP
        from ..models.module import Float16Module

        if (
            mpu.is_rank_in_position_embedding_group()
            and mpu.get_pipeline_model_parallel_world_size() > 1
            and self.cfg.pipeline_model_parallel_split_rank is not None
        ):
            unwrapped_model = unwrap_model(self.models[0], (LocalDDP, Float16Module))
            assert self.cfg.DDP_impl == "local"
            grad = unwrapped_model.language_model.embedding.position_embeddings.weight.main_grad
            torch.distributed.all_reduce(grad, group=mpu.get_position_embedding_group())

    def allreduce_embedding_grads(self):
        self.allreduce_word_embedding_grads()
        self.allreduce_position_embedding_grads()

    def allreduce_layernorm_grads(self):
        """SP: norm params are replicated over TP but see different sequence
        shards — all-reduce their grads across TP (reference optimizer.py:257-277)."""
        if (
            mpu.get_tensor_model_parallel_world_size() > 1
            and self.cfg.sequence_parallel
        ):
            grads = []
            for model_module in self.models:
                unwrapped_model = unwrap_model(model_module)
                for param in unwrapped_model.parameters():
                    if getattr(param, "sequence_parallel", False):
                        grad = (
                            param.main_grad
                            if self.params_have_main_grad
                            else param.grad
                        )
                        grads.append(grad.data)
            if grads:
                coalesced = torch._utils._flatten_dense_tensors(grads)
                torch.distributed.all_reduce(
                    coalesced, group=mpu.get_tensor_model_parallel_group()
                )
                for buf, synced in zip(
                    grads, torch._utils._unflatten_dense_tensors(coalesced, grads)
                ):
                    buf.copy_(synced)

    def reduce_model_grads(self, timers=None):
        """All grad reduction: SP-LN allreduce, DP allreduce, embedding
        allreduce (reference optimizer.py:280-301)."""
        if timers:
            timers("layernorm-grads-all-reduce", log_level=1).start()
        self.allreduce_layernorm_grads()
        if timers:
            timers("layernorm-grads-all-reduce").stop()
            timers("grads-all-reduce", log_level=1).start()
        for model in self.models:
            model.allreduce_gradients()
        if timers:
            timers("grads-all-reduce").stop()
            timers("embedding-grads-all-reduce", log_level=1).start()
        self.allreduce_embedding_grads()
        if timers:
            timers("embedding-grads-all-reduce").stop()


class MixedPrecisionOptimizer(MegatronOptimizer):
    """(reference optimizer.py:304-466)"""

    def __init__(self, optimizer, clip_grad, log_num_zeros_in_grad,
                 params_have_main_grad, use_contiguous_buffers_in_local_ddp,
                 fp16, bf16, params_dtype, grad_scaler, models, cfg):
        super().__init__(
            optimizer, clip_grad, log_num_zeros_in_grad, params_have_main_grad,
            use_contiguous_buffers_in_local_ddp, models, cfg,
        )
        self.fp16 = fp16
        self.bf16 = bf16
        self.params_dtype = params_dtype
        self.grad_scaler = grad_scaler
        if self.grad_scaler is None:
            assert not self.fp16, "fp16 expects a grad scaler"

        device = "cuda" if torch.cuda.is_available() else "cpu"
        if self.grad_scaler:
            self.found_inf = torch.tensor([0.0], dtype=torch.float, device=device)
        if bf16:
            self._dummy_overflow_buf = None
        else:
            self._dummy_overflow_buf = torch.tensor(
                [0], dtype=torch.int, device=device
            )
        if self.grad_scaler is None:
            self._scale_one = torch.tensor([1.0], dtype=torch.float, device=device)

    def get_loss_scale(self):
        if self.grad_scaler is None:
            return self._scale_one
        return self.grad_scaler.scale

    def reload_model_params(self):
        self._copy_model_params_to_main_params()

    def _unscale_main_grads_and_check_for_nan(self):
        main_grads = self._collect_main_grad_data_for_unscaling()
        self.found_inf.fill_(0.0)
        torch._amp_foreach_non_finite_check_and_unscale_(
            main_grads, self.found_inf, self.grad_scaler.inv_scale
        )
        torch.distributed.all_reduce(
            self.found_inf, op=torch.distributed.ReduceOp.MAX,
            group=self.get_model_parallel_group(),
        )
        return self.found_inf.item() > 0

    @torch.no_grad()
    def step(self, timers=None):
        def _t(name, start):
            if timers is None:
                return
            if start:
                timers(name, log_level=1).start()
            else:
                timers(name).stop()

        _t("optimizer-copy-to-main-grad", True)
        self._copy_model_grads_to_main_grads()
        _t("optimizer-copy-to-main-grad", False)

        if self.grad_scaler:
            _t("optimizer-unscale-and-check-inf", True)
            found_inf_flag = self._unscale_main_grads_and_check_for_nan()
            _t("optimizer-unscale-and-check-inf", False)
            self.grad_scaler.update(found_inf_flag)
            if found_inf_flag:
                return False, None, None

        _t("optimizer-clip-main-grad", True)
        grad_norm = None
        grad_scale = 1.0
        # with a flat grad buffer and a kernel that accepts a grad multiplier,
        # the clip coefficient folds into the Adam pass (no extra 2x-buffer
        # memory sweep); the grad scaler path keeps the eager order
        can_defer = (
            self.grad_scaler is None
            and getattr(self.optimizer, "supports_grad_scale", False)
            and self._flat_grad_buffers() is not None
        )
        if self.clip_grad > 0.0:
            if can_defer:
                grad_norm, grad_scale = self.clip_grad_norm(
                    self.clip_grad, defer_scale=True
                )
            else:
                grad_norm = self.clip_grad_norm(self.clip_grad)
        _t("optimizer-clip-main-grad", False)

        _t("optimizer-count-zeros", True)
        num_zeros_in_grad = self.count_zeros() if self.log_num_zeros_in_grad else None
        _t("optimizer-count-zeros", False)

        _t("optimizer-inner-step", True)
        if grad_scale != 1.0:
            self.optimizer.step(grad_scale=grad_scale)
        else:
            self.optimizer.step()
        _t("optimizer-inner-step", False)

        _t("optimizer-copy-main-to-model-params", True)
        if not getattr(self.optimizer, "wrote_model_params", False):
            self._copy_main_params_to_model_params()
        _t("optimizer-copy-main-to-model-params", False)

        if getattr(self.cfg, "fp8", False):
            from ..fp8 import bump_weight_epoch

            bump_weight_epoch()

        return True, grad_norm, num_zeros_in_grad


class Float16OptimizerWithFloat16Params(MixedPrecisionOptimizer):
    """fp16/bf16 model params with fp32 masters (reference optimizer.py:469-695)."""

    def __init__(self, optimizer, clip_grad, log_num_zeros_in_grad,
                 params_have_main_grad, use_contiguous_buffers_in_local_ddp,
                 fp16, bf16, params_dtype, grad_scaler, models, cfg):
        super().__init__(
            optimizer, clip_grad, log_num_zeros_in_grad, params_have_main_grad,
            use_contiguous_buffers_in_local_ddp, fp16, bf16, params_dtype,
            grad_scaler, models, cfg,
        )

        self.float16_groups = []
        self.fp32_from_float16_groups = []
        self.fp32_from_fp32_groups = []

        for param_group in self.optimizer.param_groups:
            float16_params_this_group = []
            fp32_params_this_group = []
            fp32_from_float16_params_this_group = []
            for i, param in enumerate(param_group["params"]):
                if not param.requires_grad:
                    continue
                if param.type() in (
                    "torch.cuda.HalfTensor", "torch.cuda.BFloat16Tensor",
                    "torch.HalfTensor", "torch.BFloat16Tensor",
                ):
                    float16_params_this_group.append(param)
                    main_param = param.detach().clone().float()
                    # copy tensor-parallel attributes
                    for attr in ("model_parallel", "partition_dim",
                                 "partition_stride", "shared",
                                 "sequence_parallel"):
                        if hasattr(param, attr):
                            setattr(main_param, attr, getattr(param, attr))
                    param.main_param = main_param
                    # lets the fused Adam kernel write the bf16/fp16 model
                    # copy in the same pass (ops/csrc/adam.hip)
                    main_param.model_out = param
                    param_group["params"][i] = main_param
                    fp32_from_float16_params_this_group.append(main_param)
                    if param in self.optimizer.state:
                        self.optimizer.state[main_param] = self.optimizer.state.pop(
                            param
                        )
                elif param.type() in ("torch.cuda.FloatTensor", "torch.FloatTensor"):
                    fp32_params_this_group.append(param)
                    param_group["params"][i] = param
                else:
                    raise TypeError(f"unexpected param type {param.type()}")

            self.float16_groups.append(float16_params_this_group)
            self.fp32_from_float16_groups.append(fp32_from_float16_params_this_group)
            self.fp32_from_fp32_groups.append(fp32_params_this_group)

    def zero_grad(self, set_to_none=True):
        for group in self.float16_groups:
            _zero_grad_group_helper(group, set_to_none)
        for group in self.fp32_from_float16_groups:
            _zero_grad_group_helper(group, set_to_none)
        for group in self.fp32_from_fp32_groups:
            _zero_grad_group_helper(group, set_to_none)

    def _collect_main_grad_data_for_unscaling(self):
        main_grads = []
        for main_group in self.fp32_from_float16_groups:
            for main_param in main_group:
                if main_param.grad is not None:
                    main_grads.append(main_param.grad.data)
        for main_group in self.fp32_from_fp32_groups:
            for main_param in main_group:
                if main_param.grad is not None:
                    main_grads.append(main_param.grad.data)
        return main_grads

    def _get_model_and_main_params_data_float16(self):
        model_data, main_data = [], []
        for model_group, main_group in zip(
            self.float16_groups, self.fp32_from_float16_groups
        ):
            for model_param, main_param in zip(model_group, main_group):
                model_data.append(model_param.data)
                main_data.append(main_param.data)
        return model_data, main_data

    def _copy_model_grads_to_main_grads(self):
        for model_group, main_group in zip(
            self.float16_groups, self.fp32_from_float16_groups
        ):
            for model_param, main_param in zip(model_group, main_group):
                if self.params_have_main_grad and hasattr(model_param, "main_grad"):
                    main_param.grad = model_param.main_grad.float()
                else:
                    if model_param.grad is not None:
                        main_param.grad = model_param.grad.float()
                if not self.use_contiguous_buffers_in_local_ddp:
                    model_param.grad = None
        for model_group in self.fp32_from_fp32_groups:
            for model_param in model_group:
                if self.params_have_main_grad and hasattr(model_param, "main_grad"):
                    model_param.grad = model_param.main_grad
                    if not self.use_contiguous_buffers_in_local_ddp:
                        model_param.main_grad = None

    def _copy_main_params_to_model_params(self):
        model_data, main_data = self._get_model_and_main_params_data_float16()
        _multi_tensor_copy(main_data, model_data)

    def _copy_model_params_to_main_params(self):
        model_data, main_data = self._get_model_and_main_params_data_float16()
        _multi_tensor_copy(model_data, main_data)

    def state_dict(self):
        state_dict = {}
        state_dict["optimizer"] = self.optimizer.state_dict()
        if self.grad_scaler:
            state_dict["grad_scaler"] = self.grad_scaler.state_dict()
        state_dict["fp32_from_fp16_params"] = self.fp32_from_float16_groups
        return state_dict

    def load_state_dict(self, state_dict):
        optimizer_key = "optimizer"
        if optimizer_key not in state_dict:
            optimizer_key = "optimizer_state_dict"
        self.optimizer.load_state_dict(state_dict[optimizer_key])
        if "grad_scaler" in state_dict and self.grad_scaler:
            self.grad_scaler.load_state_dict(state_dict["grad_scaler"])
        fp32_key = "fp32_from_fp16_params"
        if fp32_key not in state_dict:
            fp32_key = "fp32_from_fp16"
        for current_group, saved_group in zip(
            self.fp32_from_float16_groups, state_dict[fp32_key]
        ):
            for current_param, saved_param in zip(current_group, saved_group):
                current_param.data.copy_(saved_param.data)


class FP32Optimizer(MegatronOptimizer):
    """(reference optimizer.py:698-783)"""

    def __init__(self, optimizer, clip_grad, log_num_zeros_in_grad,
                 params_have_main_grad, use_contiguous_buffers_in_local_ddp,
                 models, cfg):
        super().__init__(
            optimizer, clip_grad, log_num_zeros_in_grad, params_have_main_grad,
            use_contiguous_buffers_in_local_ddp, models, cfg,
        )
        device = "cuda" if torch.cuda.is_available() else "cpu"
        self._scale = torch.tensor([1.0], dtype=torch.float, device=device)

    def zero_grad(self, set_to_none=True):
        for group in self.optimizer.param_groups:
            _zero_grad_group_helper(group["params"], set_to_none)

    def get_loss_scale(self):
        return self._scale

    @torch.no_grad()
    def step(self, timers=None):
        if self.params_have_main_grad:
            for param_group in self.optimizer.param_groups:
                for param in param_group["params"]:
                    if hasattr(param, "main_grad"):
                        param.grad = param.main_grad

        grad_norm = None
        if self.clip_grad > 0.0:
            grad_norm = self.clip_grad_norm(self.clip_grad)
        num_zeros_in_grad = self.count_zeros() if self.log_num_zeros_in_grad else None
        self.optimizer.step()
        return True, grad_norm, num_zeros_in_grad

    def reload_model_params(self):
        pass

    def state_dict(self):
        return self.optimizer.state_dict()

    def load_state_dict(self, state_dict):
        self.optimizer.load_state_dict(state_dict)
