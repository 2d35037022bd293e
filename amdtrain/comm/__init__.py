from .init import (
    init_from_env,
    init_from_tcp,
    init_from_file,
    get_rank,
    get_world_size,
    is_distributed,
    backend_for_device,
)
from .collectives import (
    reduce_mean,
    MetricReducer,
    barrier,
    broadcast_module_state,
    broadcast_optimizer_state,
)

__all__ = [
    "init_from_env", "init_from_tcp", "init_from_file",
    "get_rank", "get_world_size", "is_distributed", "backend_for_device",
    "reduce_mean", "MetricReducer", "barrier",
    "broadcast_module_state", "broadcast_optimizer_state",
]
