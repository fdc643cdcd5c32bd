"""Process-group topology for TP / PP / DP (+ embedding groups).

Same rank geometry as the reference (megatron/core/parallel_state.py:51-205) so
that checkpoints and launch scripts map one-to-one: with world size 16, tp=2,
pp=4 the groups are

    8 TP groups  [g, g+1]                (adjacent ranks — on an 8-GPU MI355X
                                          node this keeps every TP collective on
                                          the all-to-all xGMI mesh)
    8 DP groups  [g, g+2]
    4 PP groups  [g, g+4, g+8, g+12]

Rank order is TP-innermost, DP middle, PP outermost. On MI355X the xGMI fabric
is a fully-connected 7-link mesh per node (≈153 GB/s per link), so TP groups of
size ≤ 8 stay intra-node and every TP all-reduce / all-gather / reduce-scatter
runs as direct point-to-point RCCL transfers rather than multi-hop rings.
"""

from __future__ import annotations

import os
from typing import List, Optional

import torch
import torch.distributed as dist

# ---------------------------------------------------------------------------
# module state

_TENSOR_MODEL_PARALLEL_GROUP = None
_PIPELINE_MODEL_PARALLEL_GROUP = None
_MODEL_PARALLEL_GROUP = None
_DATA_PARALLEL_GROUP = None
_EMBEDDING_GROUP = None
_POSITION_EMBEDDING_GROUP = None

_TENSOR_MODEL_PARALLEL_WORLD_SIZE: Optional[int] = None
_PIPELINE_MODEL_PARALLEL_WORLD_SIZE: Optional[int] = None
_TENSOR_MODEL_PARALLEL_RANK: Optional[int] = None
_PIPELINE_MODEL_PARALLEL_RANK: Optional[int] = None

_PIPELINE_GLOBAL_RANKS: Optional[List[int]] = None
_DATA_PARALLEL_GLOBAL_RANKS: Optional[List[int]] = None
_EMBEDDING_GLOBAL_RANKS: Optional[List[int]] = None
_POSITION_EMBEDDING_GLOBAL_RANKS: Optional[List[int]] = None
_MPU_TENSOR_MODEL_PARALLEL_WORLD_SIZE = None  # test override
_MPU_PIPELINE_MODEL_PARALLEL_WORLD_SIZE = None

_VIRTUAL_PIPELINE_MODEL_PARALLEL_RANK: Optional[int] = None
_VIRTUAL_PIPELINE_MODEL_PARALLEL_WORLD_SIZE: Optional[int] = None
_PIPELINE_MODEL_PARALLEL_SPLIT_RANK: Optional[int] = None

_GLOBAL_MEMORY_BUFFER = None


class GlobalMemoryBuffer:
    """Reusable workspace tensors keyed by (shape, dtype, name) — avoids
    re-allocating all-gather outputs every microbatch
    (reference megatron/core/utils.py:24-42). With 288 GB of HBM3E per GPU
    holding these resident is free; churning them through the caching
    allocator is not."""

    def __init__(self):
        self.buffer = {}

    def get_tensor(self, tensor_shape, dtype, name):
        required_len = 1
        for s in tensor_shape:
            required_len *= s
        key = (name, dtype)
        buf = self.buffer.get(key)
        if buf is None or buf.numel() < required_len:
            device = torch.cuda.current_device() if torch.cuda.is_available() else "cpu"
            buf = torch.empty(required_len, dtype=dtype, device=device,
                              requires_grad=False)
            self.buffer[key] = buf
        return buf[0:required_len].view(*tensor_shape)


def initialize_model_parallel(
    tensor_model_parallel_size: int = 1,
    pipeline_model_parallel_size: int = 1,
    virtual_pipeline_model_parallel_size: Optional[int] = None,
    pipeline_model_parallel_split_rank: Optional[int] = None,
) -> None:
    """Build TP/PP/DP/model/embedding process groups.

    Group construction is rank-order-identical to the reference
    (parallel_state.py:68-205)."""
    assert dist.is_initialized()
    world_size = dist.get_world_size()
    tp = tensor_model_parallel_size
    pp = pipeline_model_parallel_size
    assert world_size % (tp * pp) == 0, (
        f"world_size {world_size} not divisible by tp*pp {tp * pp}"
    )
    dp = world_size // (tp * pp)
    rank = dist.get_rank()

    num_tp_groups = world_size // tp
    num_pp_groups = world_size // pp

    global _VIRTUAL_PIPELINE_MODEL_PARALLEL_RANK
    global _VIRTUAL_PIPELINE_MODEL_PARALLEL_WORLD_SIZE
    if virtual_pipeline_model_parallel_size is not None:
        assert pp > 2
        _VIRTUAL_PIPELINE_MODEL_PARALLEL_RANK = 0
        _VIRTUAL_PIPELINE_MODEL_PARALLEL_WORLD_SIZE = (
            virtual_pipeline_model_parallel_size
        )
    global _PIPELINE_MODEL_PARALLEL_SPLIT_RANK
    _PIPELINE_MODEL_PARALLEL_SPLIT_RANK = pipeline_model_parallel_split_rank

    # Data-parallel groups: stride tp within each pp stage
    global _DATA_PARALLEL_GROUP, _DATA_PARALLEL_GLOBAL_RANKS
    assert _DATA_PARALLEL_GROUP is None, "data parallel group already initialized"
    all_dp_group_ranks = []
    for i in range(pp):
        start_rank = i * num_pp_groups
        end_rank = (i + 1) * num_pp_groups
        for j in range(tp):
            ranks = list(range(start_rank + j, end_rank, tp))
            all_dp_group_ranks.append(ranks)
            group = dist.new_group(ranks)
            if rank in ranks:
                _DATA_PARALLEL_GROUP = group
                _DATA_PARALLEL_GLOBAL_RANKS = ranks

    # Model-parallel groups (TP x PP)
    global _MODEL_PARALLEL_GROUP
    assert _MODEL_PARALLEL_GROUP is None
    for i in range(dp):
        ranks = [dpr[i] for dpr in all_dp_group_ranks]
        group = dist.new_group(sorted(ranks))
        if rank in ranks:
            _MODEL_PARALLEL_GROUP = group

    # Tensor-parallel groups: contiguous rank blocks
    global _TENSOR_MODEL_PARALLEL_GROUP
    assert _TENSOR_MODEL_PARALLEL_GROUP is None
    for i in range(num_tp_groups):
        ranks = list(range(i * tp, (i + 1) * tp))
        group = dist.new_group(ranks)
        if rank in ranks:
            _TENSOR_MODEL_PARALLEL_GROUP = group

    # Pipeline groups (stride = world/pp) + embedding groups (first & last stage)
    global _PIPELINE_MODEL_PARALLEL_GROUP, _PIPELINE_GLOBAL_RANKS
    global _EMBEDDING_GROUP, _EMBEDDING_GLOBAL_RANKS
    global _POSITION_EMBEDDING_GROUP, _POSITION_EMBEDDING_GLOBAL_RANKS
    assert _PIPELINE_MODEL_PARALLEL_GROUP is None
    for i in range(num_pp_groups):
        ranks = list(range(i, world_size, num_pp_groups))
        group = dist.new_group(ranks)
        if rank in ranks:
            _PIPELINE_MODEL_PARALLEL_GROUP = group
            _PIPELINE_GLOBAL_RANKS = ranks
        # embedding group: first+last stage (+ split rank for T5-style)
        if len(ranks) > 1:
            embedding_ranks = [ranks[0], ranks[-1]]
            position_embedding_ranks = [ranks[0]]
            if pipeline_model_parallel_split_rank is not None:
                sr = pipeline_model_parallel_split_rank
                if ranks[sr] not in embedding_ranks:
                    embedding_ranks = [ranks[0], ranks[sr], ranks[-1]]
                if ranks[sr] not in position_embedding_ranks:
                    position_embedding_ranks = [ranks[0], ranks[sr]]
        else:
            embedding_ranks = ranks
            position_embedding_ranks = ranks
        group = dist.new_group(embedding_ranks)
        if rank in embedding_ranks:
            _EMBEDDING_GROUP = group
        if rank in ranks:
            _EMBEDDING_GLOBAL_RANKS = embedding_ranks
        group = dist.new_group(position_embedding_ranks)
        if rank in position_embedding_ranks:
            _POSITION_EMBEDDING_GROUP = group
        if rank in ranks:
            _POSITION_EMBEDDING_GLOBAL_RANKS = position_embedding_ranks

    global _GLOBAL_MEMORY_BUFFER
    _GLOBAL_MEMORY_BUFFER = GlobalMemoryBuffer()


def model_parallel_is_initialized() -> bool:
    return _TENSOR_MODEL_PARALLEL_GROUP is not None


# -- group getters ----------------------------------------------------------

def get_tensor_model_parallel_group(check_initialized=True):
    if check_initialized:
        assert _TENSOR_MODEL_PARALLEL_GROUP is not None
    return _TENSOR_MODEL_PARALLEL_GROUP


def get_pipeline_model_parallel_group():
    assert _PIPELINE_MODEL_PARALLEL_GROUP is not None
    return _PIPELINE_MODEL_PARALLEL_GROUP


def get_model_parallel_group():
    assert _MODEL_PARALLEL_GROUP is not None
    return _MODEL_PARALLEL_GROUP


def get_data_parallel_group():
    assert _DATA_PARALLEL_GROUP is not None
    return _DATA_PARALLEL_GROUP


def get_embedding_group():
    assert _EMBEDDING_GROUP is not None
    return _EMBEDDING_GROUP


def get_position_embedding_group():
    assert _POSITION_EMBEDDING_GROUP is not None
    return _POSITION_EMBEDDING_GROUP


# -- sizes / ranks ----------------------------------------------------------

def set_tensor_model_parallel_world_size(size):  # test hook
    global _MPU_TENSOR_MODEL_PARALLEL_WORLD_SIZE
    _MPU_TENSOR_MODEL_PARALLEL_WORLD_SIZE = size


def set_pipeline_model_parallel_world_size(size):  # test hook
    global _MPU_PIPELINE_MODEL_PARALLEL_WORLD_SIZE
    _MPU_PIPELINE_MODEL_PARALLEL_WORLD_SIZE = size


def get_tensor_model_parallel_world_size() -> int:
    if _MPU_TENSOR_MODEL_PARALLEL_WORLD_SIZE is not None:
        return _MPU_TENSOR_MODEL_PARALLEL_WORLD_SIZE
    return dist.get_world_size(group=get_tensor_model_parallel_group())


def get_pipeline_model_parallel_world_size() -> int:
    if _MPU_PIPELINE_MODEL_PARALLEL_WORLD_SIZE is not None:
        return _MPU_PIPELINE_MODEL_PARALLEL_WORLD_SIZE
    return dist.get_world_size(group=get_pipeline_model_parallel_group())


_MPU_TENSOR_MODEL_PARALLEL_RANK = None
_MPU_PIPELINE_MODEL_PARALLEL_RANK = None


def set_tensor_model_parallel_rank(rank):  # test hook
    global _MPU_TENSOR_MODEL_PARALLEL_RANK
    _MPU_TENSOR_MODEL_PARALLEL_RANK = rank


def set_pipeline_model_parallel_rank(rank):  # test hook
    global _MPU_PIPELINE_MODEL_PARALLEL_RANK
    _MPU_PIPELINE_MODEL_PARALLEL_RANK = rank


def get_tensor_model_parallel_rank() -> int:
    if _MPU_TENSOR_MODEL_PARALLEL_RANK is not None:
        return _MPU_TENSOR_MODEL_PARALLEL_RANK
    return dist.get_rank(group=get_tensor_model_parallel_group())


def get_pipeline_model_parallel_rank() -> int:
    if _MPU_PIPELINE_MODEL_PARALLEL_RANK is not None:
        return _MPU_PIPELINE_MODEL_PARALLEL_RANK
    return dist.get_rank(group=get_pipeline_model_parallel_group())


def get_pipeline_model_parallel_split_rank():
    return _PIPELINE_MODEL_PARALLEL_SPLIT_RANK


def set_pipeline_model_parallel_split_rank(rank):
    global _PIPELINE_MODEL_PARALLEL_SPLIT_RANK
    _PIPELINE_MODEL_PARALLEL_SPLIT_RANK = rank


def is_pipeline_first_stage(ignore_virtual=False) -> bool:
    if not ignore_virtual:
        if (
            get_virtual_pipeline_model_parallel_world_size() is not None
            and get_virtual_pipeline_model_parallel_rank() != 0
        ):
            return False
    return get_pipeline_model_parallel_rank() == 0


def is_pipeline_last_stage(ignore_virtual=False) -> bool:
    if not ignore_virtual:
        vpp = get_virtual_pipeline_model_parallel_world_size()
        if vpp is not None and get_virtual_pipeline_model_parallel_rank() != vpp - 1:
            return False
    return get_pipeline_model_parallel_rank() == (
        get_pipeline_model_parallel_world_size() - 1
    )


def is_rank_in_embedding_group(ignore_virtual=False) -> bool:
    rank = dist.get_rank()
    if _EMBEDDING_GLOBAL_RANKS is None:
        return False
    if ignore_virtual:
        return rank in _EMBEDDING_GLOBAL_RANKS
    if rank in _EMBEDDING_GLOBAL_RANKS:
        if rank == _EMBEDDING_GLOBAL_RANKS[0]:
            return is_pipeline_first_stage(ignore_virtual=False)
        elif rank == _EMBEDDING_GLOBAL_RANKS[-1]:
            return is_pipeline_last_stage(ignore_virtual=False)
        else:
            return True
    return False


def is_rank_in_position_embedding_group() -> bool:
    rank = dist.get_rank()
    return (
        _POSITION_EMBEDDING_GLOBAL_RANKS is not None
        and rank in _POSITION_EMBEDDING_GLOBAL_RANKS
    )


def is_pipeline_stage_before_split(rank=None) -> bool:
    if get_pipeline_model_parallel_world_size() == 1:
        return True
    if rank is None:
        rank = get_pipeline_model_parallel_rank()
    split = _PIPELINE_MODEL_PARALLEL_SPLIT_RANK
    if split is None:
        return True
    return rank < split


def is_pipeline_stage_after_split(rank=None) -> bool:
    if get_pipeline_model_parallel_world_size() == 1:
        return True
    if rank is None:
        rank = get_pipeline_model_parallel_rank()
    split = _PIPELINE_MODEL_PARALLEL_SPLIT_RANK
    if split is None:
        return True
    return rank >= split


def is_pipeline_stage_at_split() -> bool:
    rank = get_pipeline_model_parallel_rank()
    return is_pipeline_stage_before_split(rank) and is_pipeline_stage_after_split(
        rank + 1
    )


def get_virtual_pipeline_model_parallel_rank():
    return _VIRTUAL_PIPELINE_MODEL_PARALLEL_RANK


def set_virtual_pipeline_model_parallel_rank(rank):
    global _VIRTUAL_PIPELINE_MODEL_PARALLEL_RANK
    _VIRTUAL_PIPELINE_MODEL_PARALLEL_RANK = rank


def get_virtual_pipeline_model_parallel_world_size():
    return _VIRTUAL_PIPELINE_MODEL_PARALLEL_WORLD_SIZE


def set_virtual_pipeline_model_parallel_world_size(size):
    global _VIRTUAL_PIPELINE_MODEL_PARALLEL_WORLD_SIZE
    _VIRTUAL_PIPELINE_MODEL_PARALLEL_WORLD_SIZE = size


def get_tensor_model_parallel_src_rank() -> int:
    """Global rank of the first rank in this rank's TP group."""
    global_rank = dist.get_rank()
    local_world_size = get_tensor_model_parallel_world_size()
    return (global_rank // local_world_size) * local_world_size


def get_data_parallel_src_rank() -> int:
    assert _DATA_PARALLEL_GLOBAL_RANKS is not None
    return _DATA_PARALLEL_GLOBAL_RANKS[0]


def get_pipeline_model_parallel_first_rank() -> int:
    assert _PIPELINE_GLOBAL_RANKS is not None
    return _PIPELINE_GLOBAL_RANKS[0]


def get_pipeline_model_parallel_last_rank() -> int:
    assert _PIPELINE_GLOBAL_RANKS is not None
    return _PIPELINE_GLOBAL_RANKS[-1]


def get_pipeline_model_parallel_next_rank() -> int:
    assert _PIPELINE_GLOBAL_RANKS is not None
    rank_in_pipeline = get_pipeline_model_parallel_rank()
    world_size = get_pipeline_model_parallel_world_size()
    return _PIPELINE_GLOBAL_RANKS[(rank_in_pipeline + 1) % world_size]


def get_pipeline_model_parallel_prev_rank() -> int:
    assert _PIPELINE_GLOBAL_RANKS is not None
    rank_in_pipeline = get_pipeline_model_parallel_rank()
    world_size = get_pipeline_model_parallel_world_size()
    return _PIPELINE_GLOBAL_RANKS[(rank_in_pipeline - 1) % world_size]


def get_data_parallel_world_size() -> int:
    return dist.get_world_size(group=get_data_parallel_group())


def get_data_parallel_rank() -> int:
    return dist.get_rank(group=get_data_parallel_group())


def get_global_memory_buffer() -> GlobalMemoryBuffer:
    assert _GLOBAL_MEMORY_BUFFER is not None
    return _GLOBAL_MEMORY_BUFFER


def destroy_model_parallel() -> None:
    global _TENSOR_MODEL_PARALLEL_GROUP, _PIPELINE_MODEL_PARALLEL_GROUP
    global _MODEL_PARALLEL_GROUP, _DATA_PARALLEL_GROUP
    global _EMBEDDING_GROUP, _POSITION_EMBEDDING_GROUP
    global _VIRTUAL_PIPELINE_MODEL_PARALLEL_RANK
    global _VIRTUAL_PIPELINE_MODEL_PARALLEL_WORLD_SIZE
    global _MPU_TENSOR_MODEL_PARALLEL_WORLD_SIZE
    global _MPU_PIPELINE_MODEL_PARALLEL_WORLD_SIZE
    global _MPU_TENSOR_MODEL_PARALLEL_RANK, _MPU_PIPELINE_MODEL_PARALLEL_RANK
    global _GLOBAL_MEMORY_BUFFER, _PIPELINE_GLOBAL_RANKS
    global _DATA_PARALLEL_GLOBAL_RANKS, _EMBEDDING_GLOBAL_RANKS
    global _POSITION_EMBEDDING_GLOBAL_RANKS, _PIPELINE_MODEL_PARALLEL_SPLIT_RANK
    _TENSOR_MODEL_PARALLEL_GROUP = None
    _PIPELINE_MODEL_PARALLEL_GROUP = None
    _MODEL_PARALLEL_GROUP = None
    _DATA_PARALLEL_GROUP = None
    _EMBEDDING_GROUP = None
    _POSITION_EMBEDDING_GROUP = None
    _VIRTUAL_PIPELINE_MODEL_PARALLEL_RANK = None
    _VIRTUAL_PIPELINE_MODEL_PARALLEL_WORLD_SIZE = None
    _MPU_TENSOR_MODEL_PARALLEL_WORLD_SIZE = None
    _MPU_PIPELINE_MODEL_PARALLEL_WORLD_SIZE = None
    _MPU_TENSOR_MODEL_PARALLEL_RANK = None
    _MPU_PIPELINE_MODEL_PARALLEL_RANK = None
    _GLOBAL_MEMORY_BUFFER = None
    _PIPELINE_GLOBAL_RANKS = None
    _DATA_PARALLEL_GLOBAL_RANKS = None
    _EMBEDDING_GLOBAL_RANKS = None
    _POSITION_EMBEDDING_GLOBAL_RANKS = None
    _PIPELINE_MODEL_PARALLEL_SPLIT_RANK = None
