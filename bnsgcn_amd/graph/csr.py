"""CSR graph container (numpy, CPU) — the offline/graph-construction core.

Replaces the reference's reliance on DGL graph objects (reference:
helper/utils.py:21-140 builds DGLGraphs; train.py:71-73 casts to int32 on
GPU). Indices are int32 throughout (node counts here are < 2^31; edge
counts can exceed 2^31 only for papers100M-scale, so indptr is int64).
"""
from __future__ import annotations

import numpy as np


class CSR:
    """Row-indexed adjacency: for row r, cols are indices[indptr[r]:indptr[r+1]].

    Interpreted as in-edges (row = destination, col = source) unless stated
    otherwise. indptr: int64 [n_rows+1]; indices: int32 [n_edges].
    """

    __slots__ = ("indptr", "indices", "n_cols")

    def __init__(self, indptr: np.ndarray, indices: np.ndarray, n_cols: int):
        self.indptr = np.ascontiguousarray(indptr, dtype=np.int64)
        self.indices = np.ascontiguousarray(indices, dtype=np.int32)
        self.n_cols = int(n_cols)

    @property
    def n_rows(self) -> int:
        return len(self.indptr) - 1

    @property
    def n_edges(self) -> int:
        return len(self.indices)

    def degrees(self) -> np.ndarray:
        """Per-row edge counts (int32)."""
        return np.diff(self.indptr).astype(np.int32)

    @staticmethod
    def from_edges(src: np.ndarray, dst: np.ndarray, n_rows: int, n_cols: int,
                   sort_cols: bool = False) -> "CSR":
        """Build CSR keyed by dst (row = dst, col = src) via counting sort."""
        dst = np.asarray(dst)
        src = np.asarray(src)
        counts = np.bincount(dst, minlength=n_rows).astype(np.int64)
        indptr = np.zeros(n_rows + 1, dtype=np.int64)
        np.cumsum(counts, out=indptr[1:])
        if len(dst) > 4_000_000:
            # torch's parallel stable sort is ~2x numpy on big arrays
            import torch
            order = torch.argsort(torch.from_numpy(np.ascontiguousarray(dst)),
                                  stable=True).numpy()
        else:
            order = np.argsort(dst, kind="stable")
        indices = src[order].astype(np.int32)
        csr = CSR(indptr, indices, n_cols)
        if sort_cols:
            csr.sort_within_rows()
        return csr

    def sort_within_rows(self) -> None:
        """Sort each row's column list ascending (stable, vectorized)."""
        n = self.n_rows
        if self.n_edges == 0:
            return
        row_of_edge = np.repeat(np.arange(n, dtype=np.int64), np.diff(self.indptr))
        order = np.lexsort((self.indices, row_of_edge))
        self.indices = np.ascontiguousarray(self.indices[order])

    def transpose(self) -> "CSR":
        """Swap row/col roles: returns CSR keyed by the former column index."""
        n = self.n_rows
        row_of_edge = np.repeat(np.arange(n, dtype=np.int32), np.diff(self.indptr))
        return CSR.from_edges(src=row_of_edge, dst=self.indices.astype(np.int64),
                              n_rows=self.n_cols, n_cols=n)

    def to_edges(self) -> tuple[np.ndarray, np.ndarray]:
        """Return (src=cols, dst=rows) edge arrays."""
        dst = np.repeat(np.arange(self.n_rows, dtype=np.int32), np.diff(self.indptr))
        return self.indices.copy(), dst


class Graph:
    """Full graph: in-edge CSR (row = dst) plus node payloads.

    Mirrors the semantic content of the reference's loaded DGL graph +
    node_dict (reference: helper/utils.py:101-140): features, labels,
    train/val/test masks, and full-graph in/out degrees precomputed BEFORE
    partitioning (reference: helper/utils.py:92-93) so every partition sees
    consistent normalization constants.
    """

    def __init__(self, adj_in: CSR, feat: np.ndarray, label: np.ndarray,
                 train_mask: np.ndarray, val_mask: np.ndarray, test_mask: np.ndarray,
                 n_class: int, multilabel: bool = False, name: str = "graph"):
        assert adj_in.n_rows == adj_in.n_cols == feat.shape[0]
        self.adj_in = adj_in
        self.feat = feat.astype(np.float32, copy=False)
        self.label = label
        self.train_mask = train_mask.astype(bool)
        self.val_mask = val_mask.astype(bool)
        self.test_mask = test_mask.astype(bool)
        self.n_class = int(n_class)
        self.multilabel = bool(multilabel)
        self.name = name
        # Full-graph degrees (include self-loops if present).
        self.in_deg = adj_in.degrees()
        self.out_deg = np.bincount(adj_in.indices, minlength=adj_in.n_cols).astype(np.int32)

    @property
    def n_nodes(self) -> int:
        return self.adj_in.n_rows

    @property
    def n_edges(self) -> int:
        return self.adj_in.n_edges

    @property
    def n_feat(self) -> int:
        return self.feat.shape[1]

    @property
    def n_train(self) -> int:
        return int(self.train_mask.sum())

    def subgraph(self, node_mask: np.ndarray, name: str | None = None) -> "Graph":
        """Node-induced subgraph (for inductive training — reference:
        helper/utils.py:76-77 restricts to the train subgraph)."""
        keep = np.flatnonzero(node_mask)
        remap = np.full(self.n_nodes, -1, dtype=np.int64)
        remap[keep] = np.arange(len(keep), dtype=np.int64)
        src, dst = self.adj_in.to_edges()
        em = node_mask[src] & node_mask[dst]
        new_src = remap[src[em]]
        new_dst = remap[dst[em]]
        adj = CSR.from_edges(new_src, new_dst, len(keep), len(keep), sort_cols=True)
        return Graph(adj, self.feat[keep], self.label[keep],
                     self.train_mask[keep], self.val_mask[keep], self.test_mask[keep],
                     self.n_class, self.multilabel,
                     name=name or (self.name + "-sub"))


def add_self_loops(src: np.ndarray, dst: np.ndarray, n_nodes: int
                   ) -> tuple[np.ndarray, np.ndarray]:
    """Drop existing self-loops and append one per node (reference:
    helper/utils.py:67-69 remove+re-add)."""
    keep = src != dst
    loop = np.arange(n_nodes, dtype=src.dtype)
    return np.concatenate([src[keep], loop]), np.concatenate([dst[keep], loop])
