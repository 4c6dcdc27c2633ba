from .comm import init_distributed, all_to_all_rows, exchange_counts, backend
from .plan import HaloPlan, EpochState
from .halo import partition_aggregate, halo_exchange, comm_stream
from .reducer import GradReducer
from .boundary import discover_boundary, exchange_halo_degrees

__all__ = [
    "init_distributed", "all_to_all_rows", "exchange_counts", "backend",
    "HaloPlan", "EpochState", "partition_aggregate", "halo_exchange",
    "comm_stream", "GradReducer", "discover_boundary", "exchange_halo_degrees",
]
