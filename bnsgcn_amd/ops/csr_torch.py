"""Torch-side CSR utilities used by the runtime (device-capable).

These implement the per-epoch block assembly that the reference does with
DGL graph surgery (reference: train.py:256-281 construct_graph /
out_edges / repeat_interleave) — here it is pure offset bookkeeping over
precomputed CSR fragments, cheap enough for the epoch loop.
"""
from __future__ import annotations

import torch


def row_lengths(indptr: torch.Tensor) -> torch.Tensor:
    return indptr[1:] - indptr[:-1]


def transpose_csr(indptr: torch.Tensor, indices: torch.Tensor, n_cols: int
                  ) -> tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Transpose a CSR. Returns (indptr_T, indices_T, eperm) where
    eperm[k] = original edge id of the k-th transposed edge (so per-edge
    payloads can be permuted alongside)."""
    device = indices.device
    n_rows = indptr.numel() - 1
    row = torch.repeat_interleave(torch.arange(n_rows, device=device, dtype=indices.dtype),
                                  row_lengths(indptr))
    eperm = torch.argsort(indices.long(), stable=True)
    indices_t = row[eperm]
    # counts via searchsorted on the (already sorted) column sequence —
    # an atomic histogram serializes on power-law hub columns (measured
    # 11 ms on a 14M-edge GAT block vs ~0.2 ms for this)
    sorted_cols = indices[eperm]
    bounds = torch.arange(n_cols + 1, device=device, dtype=sorted_cols.dtype)
    indptr_t = torch.searchsorted(sorted_cols, bounds).to(indptr.dtype)
    return indptr_t, indices_t, eperm


import os

# Defaults from an on-device sweep (profiles/, Reddit-shaped bench):
# SEG 512 / 131072 waves beat (2048, 65536) by ~6% epoch time.
SEG = int(os.environ.get("BNSGCN_SEG", 512))         # heavy-row split granularity
MAX_WAVES = int(os.environ.get("BNSGCN_MAX_WAVES", 131072))


def build_worklist(indptr: torch.Tensor, seg: int = None, max_waves: int = None
                   ) -> tuple[torch.Tensor, torch.Tensor, torch.Tensor, torch.Tensor]:
    """Edge-balanced work schedule for the gfx950 SpMM kernel.

    Three ideas (each measured on the Reddit-shaped bench):
    * heavy-row split: power-law hubs (max degree ~750k vs mean ~490) are
      split into <= `seg`-edge items combined via atomicAdd (item row id
      bitwise-negated), so no single wave serializes a hub row;
    * items stay in natural (ascending dst) order: the locality partitioner
      gives consecutive rows overlapping neighborhoods, so consecutive
      waves re-hit the same feature rows in L2;
    * per-wave contiguous ranges balanced by EDGE count (wave_start), with
      the kernel's XCD-aware block remap giving each XCD chiplet a
      contiguous slab of the row space for its private L2.

    Returns (wrow int32 [W_items] (negative ~row = atomic), wbeg, wend
    int64, wave_start int32 [n_waves+1], zero_rows int64); n_waves is a
    multiple of 8 (one 256-thread block = 4 waves of 64 lanes, or 8
    half-wave subgroups for the narrow-F kernel variant). zero_rows lists
    the rows the kernel does NOT fully overwrite (split rows combined via
    atomicAdd + empty rows) — the launcher zero-fills only those instead
    of the whole output (a measurable cost at 8-way partition scale).
    """
    seg = seg or SEG
    max_waves = max_waves or MAX_WAVES
    device = indptr.device
    deg = (indptr[1:] - indptr[:-1])
    n = deg.numel()
    nseg = torch.clamp((deg + (seg - 1)) // seg, min=1)
    total = int(nseg.sum())
    rows = torch.repeat_interleave(torch.arange(n, device=device), nseg)
    item_first = torch.cumsum(nseg, 0)
    item_first = torch.cat([torch.zeros(1, dtype=item_first.dtype, device=device),
                            item_first[:-1]])
    k = torch.arange(total, device=device) - item_first[rows]
    wbeg = indptr[rows] + k * seg
    wend = torch.minimum(wbeg + seg, indptr[rows + 1])
    split = nseg[rows] > 1
    wrow = torch.where(split, ~rows, rows).to(torch.int32)

    # drop empty items (deg-0 rows): the launcher zero-initializes out, so
    # they contribute nothing — and zero-length items degenerate the
    # edge-balancing searchsorted (every empty item lands on one wave)
    empty_rows = torch.nonzero(deg == 0, as_tuple=True)[0]
    atomic_rows = torch.nonzero(nseg > 1, as_tuple=True)[0]
    zero_rows = torch.cat([empty_rows, atomic_rows])
    keep = wend > wbeg
    wrow, wbeg, wend = wrow[keep], wbeg[keep], wend[keep]
    total = int(keep.sum())
    if total == 0:
        z = torch.zeros(0, dtype=torch.int32, device=device)
        return (z, wbeg, wend, torch.zeros(1, dtype=torch.int32, device=device),
                zero_rows)

    lens = (wend - wbeg)
    cum = torch.cumsum(torch.cat([torch.zeros(1, dtype=lens.dtype, device=device),
                                  lens]), 0)
    # graphs with millions of items want more waves in flight: 262144
    # beat 131072 by ~3% on ogbn-products (2.7M items; profiles r02)
    if total >= 1_000_000:
        max_waves = max(max_waves, 262144)
    n_waves = min(max_waves, max(8, total))
    n_waves = (n_waves + 7) // 8 * 8
    targets = (cum[-1] * torch.arange(n_waves + 1, device=device).double()
               / n_waves).to(cum.dtype)
    wave_start = torch.searchsorted(cum, targets).to(torch.int32)
    wave_start[0] = 0
    wave_start[-1] = total
    return (wrow.contiguous(), wbeg.contiguous(), wend.contiguous(),
            wave_start.contiguous(), zero_rows)


def merge_csr(ip1: torch.Tensor, ix1: torch.Tensor,
              ip2: torch.Tensor, ix2: torch.Tensor,
              col_offset2: int) -> tuple[torch.Tensor, torch.Tensor]:
    """Row-wise union of two CSRs over the same row set; the second CSR's
    columns are shifted by `col_offset2` (used to append halo sources after
    inner sources for the GAT attention block). Within each output row, the
    first CSR's edges precede the second's."""
    device = ix1.device
    n = ip1.numel() - 1
    l1 = row_lengths(ip1)
    l2 = row_lengths(ip2)
    ip = torch.zeros(n + 1, dtype=ip1.dtype, device=device)
    torch.cumsum(l1 + l2, 0, out=ip[1:])
    ix = torch.empty(int(ip[-1]), dtype=ix1.dtype, device=device)
    rows = torch.arange(n, device=device)
    # positions of csr1 edges: ip[r] + offset-within-row
    if ix1.numel():
        r1 = torch.repeat_interleave(rows, l1)
        e1 = torch.arange(ix1.numel(), device=device)
        ix[ip[r1] + (e1 - ip1[r1])] = ix1
    if ix2.numel():
        r2 = torch.repeat_interleave(rows, l2)
        e2 = torch.arange(ix2.numel(), device=device)
        ix[ip[r2] + l1[r2] + (e2 - ip2[r2])] = (ix2 + col_offset2).to(ix1.dtype)
    return ip, ix


def gather_rows_csr(indptr: torch.Tensor, indices: torch.Tensor,
                    rows: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
    """Sub-CSR containing the listed rows, re-numbered 0..len(rows)-1 in the
    given order: new row k = old row rows[k]. Columns unchanged.

    This is the per-epoch sampled-halo CSR: rows = halo indices of the
    sampled boundary nodes (in packed-receive order), columns = inner
    destination ids."""
    device = indices.device
    lens = row_lengths(indptr)[rows]
    new_indptr = torch.zeros(rows.numel() + 1, dtype=indptr.dtype, device=device)
    torch.cumsum(lens, 0, out=new_indptr[1:])
    total = int(new_indptr[-1])
    if total == 0:
        return new_indptr, torch.zeros(0, dtype=indices.dtype, device=device)
    # edge gather: for new row k, copy indices[indptr[rows[k]] : +lens[k]]
    starts = indptr[rows]
    pos = torch.arange(total, device=device)
    row_of_new_edge = torch.repeat_interleave(
        torch.arange(rows.numel(), device=device), lens)
    offset_in_row = pos - new_indptr[row_of_new_edge]
    src_edge = starts[row_of_new_edge] + offset_in_row
    return new_indptr, indices[src_edge]
