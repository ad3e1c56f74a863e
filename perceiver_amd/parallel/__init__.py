from perceiver_amd.parallel.ddp import BucketedGradReducer
from perceiver_amd.parallel.utils import (
    get_rank,
    get_world_size,
    init_distributed_from_env,
    is_main_process,
    rank_zero_only,
    split_dataset_by_node,
)
