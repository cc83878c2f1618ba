from fengshen_amd.parallel import groups  # noqa: F401
from fengshen_amd.parallel.groups import (  # noqa: F401
    initialize_model_parallel,
    model_parallel_is_initialized,
    destroy_model_parallel,
    get_tensor_model_parallel_group,
    get_tensor_model_parallel_rank,
    get_tensor_model_parallel_world_size,
    get_tensor_model_parallel_src_rank,
    get_data_parallel_group,
    get_data_parallel_rank,
    get_data_parallel_world_size,
    get_pipeline_model_parallel_group,
    get_pipeline_model_parallel_rank,
    get_pipeline_model_parallel_world_size,
)
