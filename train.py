"""Single-process training entry (reference-compatible: train.py).

Run under torchrun / mpirun / by main.py's spawner; rank and world size
come from the environment (RANK/WORLD_SIZE or OMPI_COMM_WORLD_*), matching
the reference's mpirun path (train.py:459-475).
"""
from __future__ import annotations

import os

import torch.distributed as dist

from bnsgcn_amd.runtime.config import create_parser, graph_name_of
from bnsgcn_amd.runtime.trainer import run


def main():
    args = create_parser().parse_args()
    args.graph_name = graph_name_of(args)
    rank = int(os.environ.get("RANK",
               os.environ.get("OMPI_COMM_WORLD_RANK", 0)))
    world = int(os.environ.get("WORLD_SIZE",
                os.environ.get("OMPI_COMM_WORLD_SIZE", args.n_partitions)))
    try:
        run(args, rank=rank, world_size=world)
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


if __name__ == "__main__":
    main()
