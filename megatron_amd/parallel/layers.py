"""Tensor-parallel layers: vocab-parallel embedding, column/row-parallel linear.

Reference semantics: megatron/core/tensor_parallel/layers.py:128-701. The GEMMs
run on hipBLASLt through torch.matmul; the fused weight-gradient GEMM with fp32
accumulation into the DDP main_grad buffer is provided by the in-tree HIP ops
extension (megatron_amd/ops). Communication/compute overlap uses async RCCL
handles: PyTorch's ProcessGroupNCCL launches each collective on its own HIP
stream ordered by events, so issuing the dgrad all-reduce (or the SP
reduce-scatter) before the wgrad GEMM overlaps them without the reference's
CUDA_DEVICE_MAX_CONNECTIONS=1 serialization trick (layers.py:344-351).
"""

from __future__ import annotations

from typing import Callable, Optional

import torch
import torch.nn.functional as F
import torch.nn.init as init
from torch.nn.parameter import Parameter

from . import state as ps
from .mappings import (
    copy_to_tensor_model_parallel_region,
    gather_from_tensor_model_parallel_region,
    reduce_from_tensor_model_parallel_region,
    reduce_scatter_to_sequence_parallel_region,
    scatter_to_tensor_model_parallel_region,
)
from .random import get_cuda_rng_tracker
from .utils import VocabUtility, divide

_grad_accum_fusion_available = None


def _wgrad_accum_fp32(input_2d: torch.Tensor, grad_output_2d: torch.Tensor,
                      main_grad: torch.Tensor) -> None:
    """main_grad(fp32) += grad_output^T @ input, accumulated in fp32.

    On GPU this is a single hipBLASLt GEMM with HIPBLAS_COMPUTE_32F and beta=1
    from the ops extension (reference: apex fused_weight_gradient_mlp_cuda,
    fused_weight_gradient_dense.cu:128-151); CPU fallback accumulates through a
    fp32 matmul."""
    global _grad_accum_fusion_available
    if _grad_accum_fusion_available is None:
        try:
            from ..ops import ext as _ext
            _grad_accum_fusion_available = hasattr(_ext.load(), "wgrad_gemm_accum_fp32")
        except Exception:
            _grad_accum_fusion_available = False
    if _grad_accum_fusion_available and input_2d.is_cuda:
        from ..ops import ext as _ext
        _ext.load().wgrad_gemm_accum_fp32(input_2d, grad_output_2d, main_grad)
    else:
        main_grad.add_(
            torch.matmul(grad_output_2d.t().float(), input_2d.float())
        )


class LinearWithGradAccumulationAndAsyncCommunication(torch.autograd.Function):
    """Fused fwd/bwd of Y = X W^T [+ b] with optional:
      - sequence-parallel all-gather of X in fwd (and again in bwd),
      - async all-reduce (TP) or reduce-scatter (SP) of dX overlapped with dW,
      - fp32 wgrad accumulation directly into weight.main_grad.

    Reference: layers.py:213-317."""

    @staticmethod
    def forward(ctx, input, weight, bias, gradient_accumulation_fusion,
                async_grad_allreduce, sequence_parallel, fp8=False,
                fp8_meta=None):
        ctx.save_for_backward(input, weight)
        ctx.use_bias = bias is not None
        ctx.gradient_accumulation_fusion = gradient_accumulation_fusion
        ctx.async_grad_allreduce = async_grad_allreduce
        ctx.sequence_parallel = sequence_parallel
        ctx.fp8 = fp8 and input.is_cuda
        ctx.fp8_meta = fp8_meta

        if sequence_parallel:
            world_size = ps.get_tensor_model_parallel_world_size()
            dim_size = list(input.size())
            dim_size[0] = dim_size[0] * world_size
            all_gather_buffer = ps.get_global_memory_buffer().get_tensor(
                dim_size, input.dtype, "mpu"
            )
            torch.distributed.all_gather_into_tensor(
                all_gather_buffer, input.contiguous(),
                group=ps.get_tensor_model_parallel_group(),
            )
            total_input = all_gather_buffer
        else:
            total_input = input

        if ctx.fp8:
            from ..fp8 import fp8_linear_fwd, fp8_linear_fwd_delayed

            ti2d = total_input.contiguous().view(-1, total_input.shape[-1])
            if ctx.fp8_meta is not None:
                out2d = fp8_linear_fwd_delayed(ti2d, weight,
                                               ctx.fp8_meta["x"])
            else:
                out2d = fp8_linear_fwd(ti2d, weight)
            output = out2d.view(*total_input.shape[:-1], weight.shape[0])
        else:
            output = None
            if (total_input.is_cuda and not total_input.requires_grad
                    and total_input.dtype == torch.bfloat16
                    and weight.dtype == torch.bfloat16
                    and weight.shape[1] % 8 == 0):
                ti2d = total_input.reshape(-1, total_input.shape[-1])
                if ti2d.shape[0] <= 4:
                    # decode-time GEMV: the hand weight-stream kernel beats
                    # the library's M<=4 GEMM selection (ops/csrc/gemv.hip)
                    from ..ops import ext as _oext

                    out2d = _oext.load(required=True).gemv_bf16(
                        weight, ti2d.contiguous()
                    )
                    output = out2d.view(*total_input.shape[:-1],
                                        weight.shape[0])
            if output is None:
                output = torch.matmul(total_input, weight.t())
        if bias is not None:
            output = output + bias
        return output

    @staticmethod
    def backward(ctx, grad_output):
        input, weight = ctx.saved_tensors
        use_bias = ctx.use_bias

        if ctx.sequence_parallel:
            world_size = ps.get_tensor_model_parallel_world_size()
            dim_size = list(input.size())
            dim_size[0] = dim_size[0] * world_size
            all_gather_buffer = ps.get_global_memory_buffer().get_tensor(
                dim_size, input.dtype, "mpu"
            )
            gather_handle = torch.distributed.all_gather_into_tensor(
                all_gather_buffer, input.contiguous(),
                group=ps.get_tensor_model_parallel_group(), async_op=True,
            )
            total_input = all_gather_buffer
        else:
            total_input = input

        if ctx.fp8:
            from ..fp8 import fp8_linear_dgrad, fp8_linear_dgrad_delayed

            go2d = grad_output.contiguous().view(-1, grad_output.shape[-1])
            if ctx.fp8_meta is not None:
                gi2d = fp8_linear_dgrad_delayed(go2d, weight,
                                                ctx.fp8_meta["dy"])
            else:
                gi2d = fp8_linear_dgrad(go2d, weight)
            grad_input = gi2d.view(*grad_output.shape[:-1], weight.shape[1])
        else:
            grad_input = grad_output.matmul(weight)

        if ctx.sequence_parallel:
            gather_handle.wait()

        # flatten s,b dims for the wgrad GEMM
        grad_output_2d = grad_output.contiguous().view(
            -1, grad_output.shape[-1]
        )
        total_input_2d = total_input.contiguous().view(-1, total_input.shape[-1])

        if ctx.async_grad_allreduce:
            allreduce_handle = torch.distributed.all_reduce(
                grad_input, group=ps.get_tensor_model_parallel_group(),
                async_op=True,
            )
        elif ctx.sequence_parallel:
            dim_size = list(input.size())
            sub_grad_input = torch.empty(
                dim_size, dtype=input.dtype, device=input.device,
                requires_grad=False,
            )
            rs_handle = torch.distributed.reduce_scatter_tensor(
                sub_grad_input, grad_input,
                group=ps.get_tensor_model_parallel_group(), async_op=True,
            )

        if ctx.gradient_accumulation_fusion and hasattr(weight, "main_grad"):
            done = False
            if ctx.fp8:
                from ..fp8 import fp8_linear_wgrad, fp8_wgrad_enabled

                if fp8_wgrad_enabled():
                    meta = ctx.fp8_meta or {}
                    done = fp8_linear_wgrad(
                        total_input_2d, grad_output_2d, weight.main_grad,
                        meta_x=meta.get("wx"), meta_dy=meta.get("wdy"),
                    )
            if not done:
                _wgrad_accum_fp32(total_input_2d, grad_output_2d,
                                  weight.main_grad)
            grad_weight = None
        else:
            grad_weight = grad_output_2d.t().matmul(total_input_2d)
        grad_bias = grad_output_2d.sum(dim=0) if use_bias else None

        if ctx.sequence_parallel and not ctx.async_grad_allreduce:
            rs_handle.wait()
            return (sub_grad_input, grad_weight, grad_bias, None, None, None,
                    None, None)
        if ctx.async_grad_allreduce:
            allreduce_handle.wait()
        return (grad_input, grad_weight, grad_bias, None, None, None, None,
                None)


def linear_with_grad_accumulation_and_async_allreduce(
    input, weight, bias, gradient_accumulation_fusion,
    async_grad_allreduce, sequence_parallel_enabled, fp8=False,
    fp8_meta=None,
):
    return LinearWithGradAccumulationAndAsyncCommunication.apply(
        input, weight, bias, gradient_accumulation_fusion,
        async_grad_allreduce, sequence_parallel_enabled, fp8, fp8_meta,
    )


# ---------------------------------------------------------------------------


def _linear_fp8_meta(module, ref_tensor):
    """Lazily create per-call-site delayed-scaling state (device-bound)."""
    if not module.fp8 or not ref_tensor.is_cuda:
        return None
    if getattr(module, "_fp8_meta", None) is None:
        from ..fp8 import Fp8TensorMeta

        module._fp8_meta = {
            "x": Fp8TensorMeta(ref_tensor.device),
            "dy": Fp8TensorMeta(ref_tensor.device),
            # wgrad operand roles (fp8_wgrad): same tensors, separate
            # delayed-scaling state so the cast-transpose pass keeps its
            # own history
            "wx": Fp8TensorMeta(ref_tensor.device),
            "wdy": Fp8TensorMeta(ref_tensor.device),
        }
    return module._fp8_meta


def _initialize_affine_weight_gpu(weight, init_method, partition_dim, stride=1):
    weight.model_parallel = True
    weight.partition_dim = partition_dim
    weight.partition_stride = stride
    with get_cuda_rng_tracker().fork():
        init_method(weight)


def _initialize_affine_weight_cpu(
    weight, output_size, input_size, per_partition_size, partition_dim,
    init_method, stride=1, return_master_weight=False, params_dtype=torch.float32,
):
    """Init the full master weight on CPU and scatter the shard (keeps init
    bit-identical across TP sizes; reference layers.py:57-102)."""
    weight.model_parallel = True
    weight.partition_dim = partition_dim
    weight.partition_stride = stride

    master_weight = torch.empty(
        output_size, input_size, dtype=torch.float, requires_grad=False
    )
    init_method(master_weight)
    master_weight = master_weight.to(dtype=params_dtype)

    per_partition_per_stride_size = divide(per_partition_size, stride)
    weight_list = torch.split(
        master_weight, per_partition_per_stride_size, dim=partition_dim
    )
    rank = ps.get_tensor_model_parallel_rank()
    world_size = ps.get_tensor_model_parallel_world_size()
    my_weight_list = weight_list[rank::world_size]
    with torch.no_grad():
        torch.cat(my_weight_list, dim=partition_dim, out=weight)
    if return_master_weight:
        return master_weight
    return None


class VocabParallelEmbedding(torch.nn.Module):
    """Embedding sharded along vocab; fwd masks out-of-range ids, looks up the
    local shard and all-reduces across TP (reference layers.py:128-210)."""

    def __init__(self, num_embeddings, embedding_dim, *, init_method=init.xavier_normal_,
                 params_dtype=torch.float32, use_cpu_initialization=False,
                 perform_initialization=True):
        super().__init__()
        self.num_embeddings = num_embeddings
        self.embedding_dim = embedding_dim
        self.tensor_model_parallel_size = ps.get_tensor_model_parallel_world_size()
        (
            self.vocab_start_index,
            self.vocab_end_index,
        ) = VocabUtility.vocab_range_from_global_vocab_size(
            self.num_embeddings,
            ps.get_tensor_model_parallel_rank(),
            self.tensor_model_parallel_size,
        )
        self.num_embeddings_per_partition = (
            self.vocab_end_index - self.vocab_start_index
        )

        if use_cpu_initialization:
            self.weight = Parameter(
                torch.empty(self.num_embeddings_per_partition, self.embedding_dim,
                            dtype=params_dtype)
            )
            if perform_initialization:
                _initialize_affine_weight_cpu(
                    self.weight, self.num_embeddings, self.embedding_dim,
                    self.num_embeddings_per_partition, 0, init_method,
                    params_dtype=params_dtype,
                )
        else:
            device = torch.cuda.current_device() if torch.cuda.is_available() else None
            self.weight = Parameter(
                torch.empty(self.num_embeddings_per_partition, self.embedding_dim,
                            device=device, dtype=params_dtype)
            )
            if perform_initialization:
                _initialize_affine_weight_gpu(self.weight, init_method, partition_dim=0)

    def forward(self, input_):
        if self.tensor_model_parallel_size > 1:
            input_mask = (input_ < self.vocab_start_index) | (
                input_ >= self.vocab_end_index
            )
            masked_input = input_.clone() - self.vocab_start_index
            masked_input[input_mask] = 0
        else:
            masked_input = input_
        output_parallel = F.embedding(masked_input, self.weight)
        if self.tensor_model_parallel_size > 1:
            output_parallel[input_mask, :] = 0.0
        output = reduce_from_tensor_model_parallel_region(output_parallel)
        return output


class ColumnParallelLinear(torch.nn.Module):
    """Y = XA + b with A sharded along columns (output dim).

    Reference layers.py:410-563. With sequence_parallel the input arrives
    sharded s/tp and is all-gathered inside the fused autograd function."""

    def __init__(self, input_size, output_size, *, bias=True, gather_output=True,
                 init_method=init.xavier_normal_, stride=1,
                 keep_master_weight_for_test=False, skip_bias_add=False,
                 async_tensor_model_parallel_allreduce=True,
                 params_dtype=torch.float32, use_cpu_initialization=False,
                 perform_initialization=True, gradient_accumulation_fusion=False,
                 sequence_parallel_enabled: bool = False, world_size: Optional[int] = None,
                 fp8: bool = False):
        super().__init__()
        self.input_size = input_size
        self.output_size = output_size
        self.fp8 = fp8
        self.gather_output = gather_output
        world_size = world_size if world_size is not None else (
            ps.get_tensor_model_parallel_world_size()
        )
        self.output_size_per_partition = divide(output_size, world_size)
        self.skip_bias_add = skip_bias_add

        if use_cpu_initialization:
            self.weight = Parameter(
                torch.empty(self.output_size_per_partition, self.input_size,
                            dtype=params_dtype)
            )
            if perform_initialization:
                self.master_weight = _initialize_affine_weight_cpu(
                    self.weight, self.output_size, self.input_size,
                    self.output_size_per_partition, 0, init_method,
                    stride=stride, return_master_weight=keep_master_weight_for_test,
                    params_dtype=params_dtype,
                )
        else:
            device = torch.cuda.current_device() if torch.cuda.is_available() else None
            self.weight = Parameter(
                torch.empty(self.output_size_per_partition, self.input_size,
                            device=device, dtype=params_dtype)
            )
            if perform_initialization:
                _initialize_affine_weight_gpu(
                    self.weight, init_method, partition_dim=0, stride=stride
                )
        if bias:
            if use_cpu_initialization:
                self.bias = Parameter(
                    torch.empty(self.output_size_per_partition, dtype=params_dtype)
                )
            else:
                device = torch.cuda.current_device() if torch.cuda.is_available() else None
                self.bias = Parameter(
                    torch.empty(self.output_size_per_partition, device=device,
                                dtype=params_dtype)
                )
            self.bias.model_parallel = True
            self.bias.partition_dim = 0
            self.bias.partition_stride = stride
            with torch.no_grad():
                self.bias.zero_()
        else:
            self.register_parameter("bias", None)

        self.async_tensor_model_parallel_allreduce = (
            async_tensor_model_parallel_allreduce and world_size > 1
        )
        self.sequence_parallel_enabled = sequence_parallel_enabled and world_size > 1
        assert not (
            self.async_tensor_model_parallel_allreduce
            and self.sequence_parallel_enabled
        )
        self.gradient_accumulation_fusion = gradient_accumulation_fusion

    def forward(self, input_):
        bias = self.bias if not self.skip_bias_add else None

        if self.async_tensor_model_parallel_allreduce or self.sequence_parallel_enabled:
            input_parallel = input_
        else:
            input_parallel = copy_to_tensor_model_parallel_region(input_)

        output_parallel = linear_with_grad_accumulation_and_async_allreduce(
            input_parallel, self.weight, bias,
            self.gradient_accumulation_fusion,
            self.async_tensor_model_parallel_allreduce,
            self.sequence_parallel_enabled, self.fp8,
            _linear_fp8_meta(self, input_parallel),
        )
        if self.gather_output:
            assert not self.sequence_parallel_enabled
            output = gather_from_tensor_model_parallel_region(output_parallel)
        else:
            output = output_parallel
        output_bias = self.bias if self.skip_bias_add else None
        return output, output_bias


class RowParallelLinear(torch.nn.Module):
    """Y = XA + b with A sharded along rows (input dim); output all-reduced
    across TP (or reduce-scattered under SP). Reference layers.py:566-701."""

    def __init__(self, input_size, output_size, *, bias=True,
                 input_is_parallel=False, init_method=init.xavier_normal_,
                 stride=1, keep_master_weight_for_test=False, skip_bias_add=False,
                 params_dtype=torch.float32, use_cpu_initialization=False,
                 perform_initialization=True, gradient_accumulation_fusion=False,
                 sequence_parallel_enabled: bool = False, world_size: Optional[int] = None,
                 fp8: bool = False):
        super().__init__()
        self.input_size = input_size
        self.output_size = output_size
        self.fp8 = fp8
        self.input_is_parallel = input_is_parallel
        world_size = world_size if world_size is not None else (
            ps.get_tensor_model_parallel_world_size()
        )
        self.input_size_per_partition = divide(input_size, world_size)
        self.skip_bias_add = skip_bias_add
        self.gradient_accumulation_fusion = gradient_accumulation_fusion
        self.sequence_parallel_enabled = sequence_parallel_enabled and world_size > 1
        if self.sequence_parallel_enabled and not input_is_parallel:
            raise RuntimeError(
                "sequence_parallel_enabled requires input_is_parallel"
            )

        if use_cpu_initialization:
            self.weight = Parameter(
                torch.empty(self.output_size, self.input_size_per_partition,
                            dtype=params_dtype)
            )
            if perform_initialization:
                self.master_weight = _initialize_affine_weight_cpu(
                    self.weight, self.output_size, self.input_size,
                    self.input_size_per_partition, 1, init_method,
                    stride=stride, return_master_weight=keep_master_weight_for_test,
                    params_dtype=params_dtype,
                )
        else:
            device = torch.cuda.current_device() if torch.cuda.is_available() else None
            self.weight = Parameter(
                torch.empty(self.output_size, self.input_size_per_partition,
                            device=device, dtype=params_dtype)
            )
            if perform_initialization:
                _initialize_affine_weight_gpu(
                    self.weight, init_method, partition_dim=1, stride=stride
                )
        if bias:
            if use_cpu_initialization:
                self.bias = Parameter(torch.empty(self.output_size, dtype=params_dtype))
            else:
                device = torch.cuda.current_device() if torch.cuda.is_available() else None
                self.bias = Parameter(
                    torch.empty(self.output_size, device=device, dtype=params_dtype)
                )
            if self.sequence_parallel_enabled:
                self.bias.sequence_parallel = True
            with torch.no_grad():
                self.bias.zero_()
        else:
            self.register_parameter("bias", None)

    def forward(self, input_):
        if self.input_is_parallel:
            input_parallel = input_
        else:
            assert not self.sequence_parallel_enabled
            input_parallel = scatter_to_tensor_model_parallel_region(input_)

        output_parallel = linear_with_grad_accumulation_and_async_allreduce(
            input_parallel, self.weight, None,
            self.gradient_accumulation_fusion,
            False,  # row-parallel dgrad needs no all-reduce
            False, self.fp8, _linear_fp8_meta(self, input_parallel),
        )
        if self.sequence_parallel_enabled:
            output_ = reduce_scatter_to_sequence_parallel_region(output_parallel)
        else:
            output_ = reduce_from_tensor_model_parallel_region(output_parallel)
        if not self.skip_bias_add:
            output = output_ + self.bias if self.bias is not None else output_
            output_bias = None
        else:
            output = output_
            output_bias = self.bias
        return output, output_bias
