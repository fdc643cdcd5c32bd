"""Parallel runtime: process groups, TP mappings/layers, RNG, DDP, pipeline."""

from .state import (  # noqa: F401
    destroy_model_parallel,
    get_data_parallel_group,
    get_data_parallel_rank,
    get_data_parallel_src_rank,
    get_data_parallel_world_size,
    get_embedding_group,
    get_global_memory_buffer,
    get_model_parallel_group,
    get_pipeline_model_parallel_first_rank,
    get_pipeline_model_parallel_group,
    get_pipeline_model_parallel_last_rank,
    get_pipeline_model_parallel_next_rank,
    get_pipeline_model_parallel_prev_rank,
    get_pipeline_model_parallel_rank,
    get_pipeline_model_parallel_world_size,
    get_position_embedding_group,
    get_tensor_model_parallel_group,
    get_tensor_model_parallel_rank,
    get_tensor_model_parallel_src_rank,
    get_tensor_model_parallel_world_size,
    get_virtual_pipeline_model_parallel_rank,
    get_virtual_pipeline_model_parallel_world_size,
    initialize_model_parallel,
    is_pipeline_first_stage,
    is_pipeline_last_stage,
    is_rank_in_embedding_group,
    is_rank_in_position_embedding_group,
    model_parallel_is_initialized,
    set_pipeline_model_parallel_rank,
    set_pipeline_model_parallel_world_size,
    set_tensor_model_parallel_rank,
    set_tensor_model_parallel_world_size,
    set_virtual_pipeline_model_parallel_rank,
    set_virtual_pipeline_model_parallel_world_size,
)

from .mappings import (  # noqa: F401
    copy_to_tensor_model_parallel_region,
    gather_from_sequence_parallel_region,
    gather_from_tensor_model_parallel_region,
    reduce_from_tensor_model_parallel_region,
    reduce_scatter_to_sequence_parallel_region,
    scatter_to_sequence_parallel_region,
    scatter_to_tensor_model_parallel_region,
)

from .layers import (  # noqa: F401
    ColumnParallelLinear,
    RowParallelLinear,
    VocabParallelEmbedding,
    linear_with_grad_accumulation_and_async_allreduce,
)

from .cross_entropy import (  # noqa: F401
    vocab_parallel_cross_entropy,
    vocab_parallel_max_indices,
)

from .random import (  # noqa: F401
    checkpoint,
    get_cuda_rng_tracker,
    model_parallel_cuda_manual_seed,
)

from .data import broadcast_data  # noqa: F401

from .utils import (  # noqa: F401
    divide,
    gather_split_1d_tensor,
    split_tensor_along_last_dim,
    split_tensor_into_1d_equal_chunks,
    VocabUtility,
)
