"""Runtime boundary discovery + halo-degree exchange.

The partition store already carries the outgoing boundary lists and halo
degrees (graph/store.py — the partitioner has the global view). These
collectives re-derive them from halo info only, matching the reference's
startup protocol (get_boundary, helper/utils.py:150-184, and
collect_out_degree, train.py:148-167); they are used to cross-check the
store in tests and allow stores produced without global degree info.
"""
from __future__ import annotations

import torch
import torch.distributed as dist

from .comm import all_to_all_rows, exchange_counts


def discover_boundary(halo_part: torch.Tensor, halo_owner_local: torch.Tensor,
                      n_parts: int) -> list[torch.Tensor]:
    """Each rank announces, per owner peer j, the owner-local ids of j's
    nodes in its halo; returns boundary[j] = my inner-local ids that peer j
    needs (sorted ascending — both sides sorted identically)."""
    rank = dist.get_rank()
    send_counts = [int((halo_part == j).sum()) for j in range(n_parts)]
    send_counts[rank] = 0
    recv_counts = exchange_counts(torch.tensor(send_counts, dtype=torch.int64)).tolist()
    recv_counts[rank] = 0
    # halo rows are peer-major sorted, so halo_owner_local is already grouped
    send = halo_owner_local.to(torch.int64).reshape(-1, 1)
    recv = torch.empty(sum(recv_counts), 1, dtype=torch.int64)
    all_to_all_rows(recv, send, recv_counts, send_counts)
    out, off = [], 0
    for j in range(n_parts):
        out.append(recv[off:off + recv_counts[j], 0].clone())
        off += recv_counts[j]
    return out


def exchange_halo_degrees(boundary: list[torch.Tensor],
                          my_deg: torch.Tensor,
                          recv_counts: list[int]) -> torch.Tensor:
    """Send my inner nodes' degree for each peer's boundary request;
    receive the degrees of my halo rows (peer-major order).
    Reference counterpart: collect_out_degree (train.py:148-167)."""
    rank = dist.get_rank()
    send_parts = [my_deg[b.long()].to(torch.int64) for j, b in enumerate(boundary)
                  if j != rank]
    send_counts = [0 if j == rank else len(boundary[j])
                   for j in range(len(boundary))]
    send = (torch.cat(send_parts) if send_parts
            else torch.zeros(0, dtype=torch.int64)).reshape(-1, 1)
    rc = list(recv_counts)
    rc[rank] = 0
    recv = torch.empty(sum(rc), 1, dtype=torch.int64)
    all_to_all_rows(recv, send, rc, send_counts)
    return recv[:, 0]
