"""ZeRO-1-style distributed optimizer.

Reference: megatron/optimizer/distrib_optimizer.py:32-701. The DP group shards
optimizer state by slicing the *contiguous grad buffer* into dp equal ranges,
ignoring parameter boundaries (range maps :62-188); grads arrive by
reduce-scatter over the buffer and updated params return by all-gather.

MI355X shape of the same idea:
 - the DDP grad buffer is already padded to a multiple of dp (parallel/ddp.py),
   so the reduce-scatter and all-gather are single in-place RCCL calls over one
   allocation (recv = send + rank*shard — RCCL's in-place fast path);
 - model params of each dtype are re-pointed into ONE contiguous param buffer
   with the same layout as the grad buffer, so gather_model_params is one
   in-place all-gather that lands directly in param storage (the reference
   aliases its param buffer onto grad-buffer storage to save memory,
   distrib_optimizer.py:376-389 — with 288 GB HBM3E a separate resident buffer
   is cheaper than the aliasing gymnastics).
"""

from __future__ import annotations

import torch

from .. import parallel as mpu
from .optimizer import MixedPrecisionOptimizer, _zero_grad_group_helper


class Range:
    def __init__(self, start, end):
        self.start = start
        self.end = end
        self.size = end - start

    def normalize(self, start=0):
        return Range(start, start + self.size)

    def __str__(self):
        return f"{self.start},{self.end} [{self.size}]"


class DistributedOptimizer(MixedPrecisionOptimizer):
    def _flat_grad_buffers(self):
        # ZeRO-1 owns only a shard: after the in-place reduce-scatter the
        # non-owned buffer regions hold stale local grads, so the whole-buffer
        # norm shortcut is invalid — clip over the shard views instead
        return None

    @classmethod
    def build_model_gbuf_param_range_map(cls, model, dtype, gbuf_world_range):
        param_world_index_map = model._grad_buffer_param_index_map[dtype]
        param_range_map = {}
        for param, (param_world_start, param_world_end) in (
            param_world_index_map.items()
        ):
            # clamp param world range to this rank's gbuf range
            param_world_start_clamped = max(param_world_start, gbuf_world_range.start)
            param_world_end_clamped = min(param_world_end, gbuf_world_range.end)
            if param_world_end_clamped <= param_world_start_clamped:
                continue
            param_world_range = Range(param_world_start_clamped,
                                      param_world_end_clamped)
            param_local_range = Range(
                param_world_range.start - gbuf_world_range.start,
                param_world_range.end - gbuf_world_range.start,
            )
            # sub-range within the param itself
            sub_param_start = param_world_range.start - param_world_start
            sub_param_range = Range(
                sub_param_start, sub_param_start + param_world_range.size
            )
            param_range_map[param] = {
                "gbuf_world": param_world_range,
                "gbuf_local": param_local_range,
                "param": sub_param_range,
            }
        return param_range_map

    @classmethod
    def build_model_gbuf_range(cls, model, dtype):
        data_parallel_rank = mpu.get_data_parallel_rank()
        data_parallel_world_size = mpu.get_data_parallel_world_size()

        grad_buffer = model._grad_buffers[dtype]
        gbuf_size = grad_buffer.numel_padded
        assert gbuf_size % data_parallel_world_size == 0
        max_gbuf_range_size = gbuf_size // data_parallel_world_size

        gbuf_world_range = Range(
            data_parallel_rank * max_gbuf_range_size,
            (data_parallel_rank + 1) * max_gbuf_range_size,
        )
        param_range_map = cls.build_model_gbuf_param_range_map(
            model, dtype, gbuf_world_range
        )
        return {
            "world": gbuf_world_range,
            "param_map": param_range_map,
            "max_range_size": max_gbuf_range_size,
        }

    @classmethod
    def build_model_gbuf_range_map(cls, model):
        return {
            dtype: cls.build_model_gbuf_range(model, dtype)
            for dtype in model._grad_buffers
        }

    @classmethod
    def build_model_param_gbuf_map(cls, model_gbuf_ranges):
        param_gbuf_map = {}
        for model_index, model_gbuf_range_map in enumerate(model_gbuf_ranges):
            for dtype, gbuf_range_map in model_gbuf_range_map.items():
                for param in gbuf_range_map["param_map"]:
                    param_gbuf_map[param] = (model_index, dtype)
        return param_gbuf_map

    @classmethod
    def build_optimizer_group_ranges(cls, param_groups, model_gbuf_ranges):
        world_param_group_map = {}
        for group_index, group in enumerate(param_groups):
            for param in group["params"]:
                assert param.requires_grad
                world_param_group_map[param] = group_index

        group_ranges = [{"params": []} for _ in param_groups]
        for model_gbuf_range_map in model_gbuf_ranges:
            for dtype, gbuf_range_map in model_gbuf_range_map.items():
                for param in gbuf_range_map["param_map"]:
                    group_index = world_param_group_map[param]
                    group_ranges[group_index]["params"].append(param)
        for group_index, group_range in enumerate(group_ranges):
            group_range["orig_group"] = param_groups[group_index]
        return group_ranges

    @classmethod
    def build_model_and_main_param_groups(cls, model_gbuf_ranges,
                                          param_gbuf_map, opt_group_ranges):
        model_float16_groups = []
        model_fp32_groups = []
        shard_float16_groups = []
        shard_fp32_groups = []
        shard_fp32_from_float16_groups = []

        for group_index, group_range in enumerate(opt_group_ranges):
            model_float16_params_this_group = []
            model_fp32_params_this_group = []
            shard_float16_params_this_group = []
            shard_fp32_params_this_group = []
            shard_fp32_from_float16_params_this_group = []
            model_float16_groups.append(model_float16_params_this_group)
            model_fp32_groups.append(model_fp32_params_this_group)
            shard_float16_groups.append(shard_float16_params_this_group)
            shard_fp32_groups.append(shard_fp32_params_this_group)
            shard_fp32_from_float16_groups.append(
                shard_fp32_from_float16_params_this_group
            )

            for param in group_range["params"]:
                assert param.requires_grad
                model_index, dtype = param_gbuf_map[param]
                gbuf_range = model_gbuf_ranges[model_index][dtype]
                param_range = gbuf_range["param_map"][param]["param"]

                if param.type() in (
                    "torch.cuda.HalfTensor", "torch.cuda.BFloat16Tensor",
                    "torch.HalfTensor", "torch.BFloat16Tensor",
                ):
                    shard_model_param = param.detach().view(-1)[
                        param_range.start : param_range.end
                    ]
                    shard_main_param = shard_model_param.clone().float()
                    for attr in ("model_parallel", "partition_dim",
                                 "partition_stride", "shared",
                                 "sequence_parallel"):
                        if hasattr(param, attr):
                            setattr(shard_model_param, attr, getattr(param, attr))
                            setattr(shard_main_param, attr, getattr(param, attr))
                    shard_main_param.model_out = shard_model_param
                    model_float16_params_this_group.append(param)
                    shard_float16_params_this_group.append(shard_model_param)
                    shard_fp32_from_float16_params_this_group.append(shard_main_param)
                elif param.type() in ("torch.cuda.FloatTensor", "torch.FloatTensor"):
                    shard_model_param = param.view(-1)[
                        param_range.start : param_range.end
                    ]
                    model_fp32_params_this_group.append(param)
                    shard_fp32_params_this_group.append(shard_model_param)
                    for attr in ("model_parallel", "partition_dim",
                                 "partition_stride", "shared",
                                 "sequence_parallel"):
                        if hasattr(param, attr):
                            setattr(shard_model_param, attr, getattr(param, attr))
                else:
                    raise TypeError(f"unexpected param type {param.type()}")

            group_range["orig_group"]["params"] = [
                *shard_fp32_params_this_group,
                *shard_fp32_from_float16_params_this_group,
            ]

        return (
            model_float16_groups, model_fp32_groups,
            shard_float16_groups, shard_fp32_groups,
            shard_fp32_from_float16_groups,
        )

    def __init__(self, optimizer, clip_grad, log_num_zeros_in_grad,
                 params_have_main_grad, use_contiguous_buffers_in_local_ddp,
                 fp16, bf16, params_dtype, grad_scaler, models, cfg):
        super().__init__(
            optimizer, clip_grad, log_num_zeros_in_grad, params_have_main_grad,
            use_contiguous_buffers_in_local_ddp, fp16, bf16, params_dtype,
            grad_scaler, models, cfg,
        )
        assert use_contiguous_buffers_in_local_ddp

        self.model_gbuf_ranges = []
        for model_index, model in enumerate(self.models):
            self.model_gbuf_ranges.append(self.build_model_gbuf_range_map(model))
        self.model_param_gbuf_map = self.build_model_param_gbuf_map(
            self.model_gbuf_ranges
        )

        # Contiguous param buffers mirroring the grad buffer layout; params
        # are re-pointed into them FIRST so every shard view built below
        # references the final storage (the DP all-gather then lands directly
        # in param storage).
        self.param_buffers = []
        for model_index, model in enumerate(self.models):
            current_param_buffers = {}
            for dtype, grad_buffer in model._grad_buffers.items():
                buf = torch.empty(
                    grad_buffer.numel_padded, dtype=params_dtype,
                    device=grad_buffer.data.device,
                )
                for param, (start, end) in (
                    model._grad_buffer_param_index_map[dtype].items()
                ):
                    view = buf[start:end].view(param.data.shape)
                    view.detach().copy_(param.data)
                    param.data = view
                current_param_buffers[dtype] = buf
            self.param_buffers.append(current_param_buffers)

        self.opt_group_ranges = self.build_optimizer_group_ranges(
            self.optimizer.param_groups, self.model_gbuf_ranges
        )

        (
            self.model_float16_groups, self.model_fp32_groups,
            self.shard_float16_groups, self.shard_fp32_groups,
            self.shard_fp32_from_float16_groups,
        ) = self.build_model_and_main_param_groups(
            self.model_gbuf_ranges, self.model_param_gbuf_map,
            self.opt_group_ranges,
        )

        self.optimizer.param_groups = [
            g["orig_group"] for g in self.opt_group_ranges
        ]

    def get_model_param_range_map(self, param):
        model_index, dtype = self.model_param_gbuf_map[param]
        gbuf_range_map = self.model_gbuf_ranges[model_index][dtype]
        return gbuf_range_map["param_map"][param]

    def get_model_parallel_group(self):
        return None  # grads are fully sharded; norm uses world all-reduce

    def zero_grad(self, set_to_none=True):
        for groups in (
            self.model_float16_groups, self.model_fp32_groups,
            self.shard_float16_groups, self.shard_fp32_groups,
            self.shard_fp32_from_float16_groups,
        ):
            for group in groups:
                _zero_grad_group_helper(group, set_to_none)

    def reduce_model_grads(self, timers=None):
        """SP-LN + embedding all-reduce, then grad-buffer reduce-scatter
        (reference distrib_optimizer.py:522-569)."""
        if timers:
            timers("layernorm-grads-all-reduce", log_level=1).start()
        self.allreduce_layernorm_grads()
        if timers:
            timers("layernorm-grads-all-reduce").stop()
            timers("embedding-grads-all-reduce", log_level=1).start()
        self.allreduce_embedding_grads()
        if timers:
            timers("embedding-grads-all-reduce").stop()
            timers("grads-reduce-scatter", log_level=1).start()
        data_parallel_world_size = mpu.get_data_parallel_world_size()
        data_parallel_rank = mpu.get_data_parallel_rank()
        data_parallel_group = mpu.get_data_parallel_group()
        for model in self.models:
            for dtype, gbuf in model._grad_buffers.items():
                gbuf.data /= data_parallel_world_size
                shard_size = gbuf.numel_padded // data_parallel_world_size
                local_shard = gbuf.data[
                    data_parallel_rank * shard_size :
                    (data_parallel_rank + 1) * shard_size
                ]
                torch.distributed.reduce_scatter_tensor(
                    local_shard, gbuf.data, group=data_parallel_group
                )
        if timers:
            timers("grads-reduce-scatter").stop()

    def gather_model_params(self, timers=None):
        """In-place all-gather of the param buffers
        (reference distrib_optimizer.py:571-610)."""
        if timers:
            timers("params-all-gather", log_level=1).start()
        data_parallel_world_size = mpu.get_data_parallel_world_size()
        data_parallel_rank = mpu.get_data_parallel_rank()
        data_parallel_group = mpu.get_data_parallel_group()
        for model_index, param_buffers in enumerate(self.param_buffers):
            for dtype, buf in param_buffers.items():
                shard_size = buf.numel() // data_parallel_world_size
                local_shard = buf[
                    data_parallel_rank * shard_size :
                    (data_parallel_rank + 1) * shard_size
                ]
                torch.distributed.all_gather_into_tensor(
                    buf, local_shard, group=data_parallel_group
                )
        if timers:
            timers("params-all-gather").stop()

    def _collect_main_grad_data_for_unscaling(self):
        return [
            param.grad.data
            for group in self.optimizer.param_groups
            for param in group["params"]
            if param.grad is not None
        ]

    def _get_model_and_main_params_data_float16(self):
        model_data, main_data = [], []
        for model_group, main_group in zip(
            self.shard_float16_groups, self.shard_fp32_from_float16_groups
        ):
            for model_param, main_param in zip(model_group, main_group):
                model_data.append(model_param.data)
                main_data.append(main_param.data)
        return model_data, main_data

    def _copy_model_grads_to_main_grads(self):
        def copy_group_grads(model_groups, shard_main_groups):
            for model_group, shard_main_group in zip(model_groups,
                                                     shard_main_groups):
                for model_param, shard_main_param in zip(model_group,
                                                         shard_main_group):
                    param_range_map = self.get_model_param_range_map(model_param)
                    gbuf_local = param_range_map["gbuf_local"]
                    model_index, dtype = self.model_param_gbuf_map[model_param]
                    gbuf = self.models[model_index]._grad_buffers[dtype]
                    dp_rank = mpu.get_data_parallel_rank()
                    shard_size = gbuf.numel_padded // (
                        mpu.get_data_parallel_world_size()
                    )
                    shard = gbuf.data[
                        dp_rank * shard_size : (dp_rank + 1) * shard_size
                    ]
                    model_grad_shard = shard[gbuf_local.start : gbuf_local.end]
                    shard_main_param.grad = model_grad_shard.float()

        copy_group_grads(self.model_float16_groups,
                         self.shard_fp32_from_float16_groups)
        copy_group_grads(self.model_fp32_groups, self.shard_fp32_groups)

    def _copy_main_params_to_model_params(self):
        def copy_group_params(shard_main_groups, model_groups):
            for shard_main_group, model_group in zip(shard_main_groups,
                                                     model_groups):
                for shard_main_param, model_param in zip(shard_main_group,
                                                         model_group):
                    param_range_map = self.get_model_param_range_map(model_param)
                    world_range = param_range_map["param"]
                    model_param.data.view(-1)[
                        world_range.start : world_range.end
                    ].copy_(shard_main_param.data)

        copy_group_params(self.shard_fp32_from_float16_groups,
                          self.model_float16_groups)

    def _copy_model_params_to_main_params(self):
        def copy_group_params(model_groups, shard_main_groups):
            for model_group, shard_main_group in zip(model_groups,
                                                     shard_main_groups):
                for model_param, shard_main_param in zip(model_group,
                                                         shard_main_group):
                    param_range_map = self.get_model_param_range_map(model_param)
                    world_range = param_range_map["param"]
                    shard_main_param.data.copy_(
                        model_param.data.view(-1)[
                            world_range.start : world_range.end
                        ]
                    )

        copy_group_params(self.model_float16_groups,
                          self.shard_fp32_from_float16_groups)

    def reload_model_params(self):
        self._copy_model_params_to_main_params()

    @torch.no_grad()
    def step(self, timers=None):
        update_successful, grad_norm, num_zeros_in_grad = super().step(timers)
        if update_successful:
            self.gather_model_params(timers)
        return update_successful, grad_norm, num_zeros_in_grad

    def state_dict(self):
        state_dict = {}
        state_dict["optimizer"] = self.optimizer.state_dict()
        if self.grad_scaler:
            state_dict["grad_scaler"] = self.grad_scaler.state_dict()
        state_dict["shard_fp32_from_float16_groups"] = (
            self.shard_fp32_from_float16_groups
        )
        return state_dict

    def load_state_dict(self, state_dict):
        self.optimizer.load_state_dict(state_dict["optimizer"])
        if "grad_scaler" in state_dict and self.grad_scaler:
            self.grad_scaler.load_state_dict(state_dict["grad_scaler"])
        for current_group, saved_group in zip(
            self.shard_fp32_from_float16_groups,
            state_dict["shard_fp32_from_float16_groups"],
        ):
            for current_param, saved_param in zip(current_group, saved_group):
                current_param.data.copy_(saved_param.data)
