"""Launcher CLI (reference-compatible: main.py of GATECH-EIC/BNS-GCN).

Names the graph, partitions it on node-rank 0, then spawns one training
process per local partition (reference main.py:10-64). On a GPU node each
process binds one MI355X and the group runs over RCCL; without GPUs the
same code runs on gloo (CPU).
"""
from __future__ import annotations

import os

import torch.multiprocessing as mp

from bnsgcn_amd.runtime.config import create_parser, graph_name_of
from bnsgcn_amd.runtime.trainer import prepare_partitions, run


def _worker(local_idx: int, start: int, world: int, args):
    import torch.distributed as dist
    try:
        run(args, rank=start + local_idx, world_size=world)
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def main():
    args = create_parser().parse_args()
    args.graph_name = graph_name_of(args)
    if args.node_rank == 0 and not args.skip_partition:
        prepare_partitions(args)

    world = args.n_partitions
    if args.backend == "mpi":
        # reference main.py:51-62: exec `mpirun -n P python train.py ...`;
        # train.py discovers its rank from OMPI_COMM_WORLD_RANK
        import shutil
        import subprocess
        import sys
        if shutil.which("mpirun") is None:
            raise RuntimeError("--backend mpi requires mpirun on PATH "
                               "(reference main.py:51-62); use nccl/gloo")
        cmd = ["mpirun", "-n", str(world), sys.executable, "train.py",
               "--skip-partition"] + sys.argv[1:]
        raise SystemExit(subprocess.call(cmd))
    start = args.node_rank * args.parts_per_node
    local = min(args.parts_per_node, world - start)
    os.environ.setdefault("MASTER_ADDR", args.master_addr)
    os.environ.setdefault("MASTER_PORT", str(args.port))
    os.environ.pop("RANK", None)        # ranks are passed explicitly
    os.environ.pop("WORLD_SIZE", None)
    if world == 1:
        _worker(0, 0, 1, args)
    else:
        mp.start_processes(_worker, args=(start, world, args), nprocs=local,
                           join=True, start_method="spawn")


if __name__ == "__main__":
    main()
