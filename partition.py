"""Partition-only CLI (reference-compatible: partition.py) — for
heterogeneous clusters without shared storage (reference README.md:116)."""
from bnsgcn_amd.runtime.config import create_parser, graph_name_of
from bnsgcn_amd.runtime.trainer import prepare_partitions

if __name__ == "__main__":
    args = create_parser().parse_args()
    args.graph_name = graph_name_of(args)
    out = prepare_partitions(args)
    print(f"partition store written to {out}")
