from neuronx_distributed_inference_amd.parallel.state import (  # noqa: F401
    initialize_model_parallel,
    model_parallel_is_initialized,
    destroy_model_parallel,
    get_tensor_model_parallel_group,
    get_tensor_model_parallel_rank,
    get_tensor_model_parallel_size,
    get_context_parallel_group,
    get_context_parallel_rank,
    get_context_parallel_size,
    get_attention_dp_group,
    get_attention_dp_rank,
    get_attention_dp_size,
    get_expert_parallel_group,
    get_expert_parallel_rank,
    get_expert_parallel_size,
    get_world_group,
    get_prefill_tp_group,
)
from neuronx_distributed_inference_amd.parallel.mappings import (  # noqa: F401
    reduce_from_tensor_parallel_region,
    gather_from_tensor_parallel_region,
    scatter_to_tensor_parallel_region,
    reduce_scatter_along_dim,
    gather_along_dim,
    all_to_all,
)
from neuronx_distributed_inference_amd.parallel.layers import (  # noqa: F401
    ColumnParallelLinear,
    RowParallelLinear,
    ParallelEmbedding,
)
from neuronx_distributed_inference_amd.parallel.ops import (  # noqa: F401
    distributed_argmax,
    distributed_topk,
)
