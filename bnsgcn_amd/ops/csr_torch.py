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
    counts = torch.bincount(indices.long(), minlength=n_cols)
    indptr_t = torch.zeros(n_cols + 1, dtype=indptr.dtype, device=device)
    torch.cumsum(counts, 0, out=indptr_t[1:])
    return indptr_t, indices_t, eperm


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
