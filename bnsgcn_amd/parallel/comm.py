"""Backend-agnostic collectives for the partition-parallel runtime.

The GPU path is one process per GPU over RCCL (`torch.distributed` backend
"nccl" IS RCCL on ROCm); on the 8×MI355X node the halo exchange maps to
`all_to_all_single`, which RCCL executes as direct pairwise sends over the
fully connected xGMI clique — all 7 links of every GPU concurrently (the
design target of SURVEY.md §5.8). CPU tests use gloo; gloo has no
all-to-all, so a ring-ordered isend/irecv fallback (the reference's own
gloo pattern, helper/utils.py:204-211, minus the pinned-host staging)
provides identical semantics.
"""
from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist


def init_distributed(backend: str | None = None, rank: int | None = None,
                     world_size: int | None = None, master_addr: str = "127.0.0.1",
                     master_port: int = 18118) -> tuple[int, int]:
    """Initialise the process group; returns (rank, world_size).

    Reference counterpart: train.py:459-468 (init_processes). Backend
    default: nccl(RCCL) when a GPU is visible, else gloo.
    """
    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    rank = int(os.environ.get("RANK", rank if rank is not None else 0))
    world_size = int(os.environ.get("WORLD_SIZE",
                                    world_size if world_size is not None else 1))
    os.environ.setdefault("MASTER_ADDR", master_addr)
    os.environ.setdefault("MASTER_PORT", str(master_port))
    if backend is None or backend == "auto":
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    # generous timeout: rank 0 may be generating + partitioning a
    # 100M-edge synthetic graph while the others wait at the first barrier
    dist.init_process_group(backend, rank=rank, world_size=world_size,
                            timeout=datetime.timedelta(seconds=1800))
    return rank, world_size


def backend() -> str:
    return dist.get_backend()


def _supports_alltoall() -> bool:
    return dist.get_backend() in ("nccl", "mpi")


def all_to_all_rows(recv: torch.Tensor, send: torch.Tensor,
                    recv_counts: list[int], send_counts: list[int],
                    async_op: bool = False):
    """Variable all-to-all of row blocks (dim-0 splits).

    send: [sum(send_counts), F] row blocks destined to ranks 0..P-1 in rank
    order; recv: preallocated [sum(recv_counts), F]. Returns a waitable
    handle when async_op (nccl) else None.
    """
    if sum(send_counts) == 0 and sum(recv_counts) == 0:
        return None          # sampling-rate 0: nothing moves (NCCL dislikes
                             # empty tensors in collectives)
    if not dist.is_initialized() or dist.get_world_size() == 1:
        # single-process: only the (normally empty) self block
        if recv_counts and recv_counts[0] > 0:
            recv[:recv_counts[0]].copy_(send[:send_counts[0]])
        return None
    if _supports_alltoall():
        return dist.all_to_all_single(recv, send,
                                      output_split_sizes=recv_counts,
                                      input_split_sizes=send_counts,
                                      async_op=async_op)
    # gloo fallback: non-blocking pairwise isend/irecv, ring-ordered.
    # gloo p2p is CPU-only; CUDA tensors are staged through host copies
    # (the reference's gloo+GPU oversubscription mode, main.py:45 /
    # utils.py:197-211 — minus the persistent pinned mirrors: this path
    # is a test/compat fallback, the production path is RCCL above).
    rank, size = dist.get_rank(), dist.get_world_size()
    staged = recv.is_cuda or send.is_cuda
    d_recv, d_send = recv, send
    if staged:
        send = send.cpu()
        recv = torch.empty_like(d_recv, device="cpu")
    s_off = [0]
    for c in send_counts:
        s_off.append(s_off[-1] + c)
    r_off = [0]
    for c in recv_counts:
        r_off.append(r_off[-1] + c)
    reqs = []
    for step in range(1, size):
        right = (rank + step) % size
        left = (rank - step + size) % size
        if recv_counts[left] > 0:
            reqs.append(dist.irecv(recv[r_off[left]:r_off[left + 1]], src=left))
        if send_counts[right] > 0:
            chunk = send[s_off[right]:s_off[right + 1]].contiguous()
            reqs.append(dist.isend(chunk, dst=right))
    for r in reqs:
        r.wait()
    # self block
    if size >= 1 and recv_counts[rank] > 0:
        recv[r_off[rank]:r_off[rank + 1]].copy_(send[s_off[rank]:s_off[rank + 1]])
    if staged:
        d_recv.copy_(recv)
    return None


def exchange_counts(my_counts: torch.Tensor) -> torch.Tensor:
    """Each rank contributes a length-P int64 vector (what I send to each
    peer); returns the length-P vector of what each peer sends to ME.
    Implemented with all_gather so it works on every backend."""
    if not dist.is_initialized():
        return my_counts.clone()
    size = dist.get_world_size()
    rank = dist.get_rank()
    dev = my_counts.device
    if dist.get_backend() == "gloo" and my_counts.is_cuda:
        my_counts = my_counts.cpu()   # gloo all_gather is CPU-only
    elif dist.get_backend() == "nccl" and not my_counts.is_cuda:
        my_counts = my_counts.cuda()  # RCCL is device-only
    gathered = [torch.zeros_like(my_counts) for _ in range(size)]
    dist.all_gather(gathered, my_counts)
    return torch.stack(gathered)[:, rank].to(dev)
