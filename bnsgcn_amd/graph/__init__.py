from .csr import CSR, Graph, add_self_loops
from .synthetic import DATASETS, load_data
from .partition import assign_parts, partition_graph, partition_and_save
from .store import Partition, load_partition, load_meta, save_partitions

__all__ = [
    "CSR", "Graph", "add_self_loops", "DATASETS", "load_data",
    "assign_parts", "partition_graph", "partition_and_save",
    "Partition", "load_partition", "load_meta", "save_partitions",
]
