from .config import create_parser, graph_name_of
from .trainer import prepare_partitions, run, Evaluator

__all__ = ["create_parser", "graph_name_of", "prepare_partitions", "run",
           "Evaluator"]
