from unionml_amd.parallel.ddp import (  # noqa: F401
    GradientAllReducer,
    distributed_is_active,
    get_rank,
    get_world_size,
    maybe_wrap,
    shard,
)
