"""Synthetic datasets of the named shapes (Reddit / ogbn-products / Yelp /
ogbn-papers100M).

There is no network access in the build/runtime environment, so the real
datasets the reference loads (reference: helper/utils.py:21-70 via
dgl.data/ogb) cannot be downloaded. Instead we generate graphs of the SAME
shape — node count, edge count, feature width, class count, multilabel-ness,
split sizes — with a power-law degree distribution and tunable locality
(locality makes the contiguous partitioner behave like METIS does on the
real graphs: small boundary cuts). Feature/label payloads are random.

BASELINE.json's bench contract explicitly allows this ("synthetic data of
that shape and random-init weights").
"""
from __future__ import annotations

from dataclasses import dataclass

import numpy as np

import zlib

from .csr import CSR, Graph, add_self_loops


@dataclass(frozen=True)
class GraphSpec:
    n_nodes: int
    n_edges: int          # directed edge count BEFORE self-loops
    n_feat: int
    n_class: int
    multilabel: bool
    train_frac: float
    val_frac: float
    zipf_alpha: float     # degree skew (higher = flatter)
    locality: float       # fraction of edges drawn from a local window
    window_frac: float    # local window width as a fraction of n_nodes


# Shapes per SURVEY.md §7.1 / the reference README's dataset table.
DATASETS: dict[str, GraphSpec] = {
    "reddit": GraphSpec(232_965, 114_615_892, 602, 41, False, 0.66, 0.10, 0.75, 0.85, 0.004),
    "ogbn-products": GraphSpec(2_449_029, 123_718_280, 100, 47, False, 0.08, 0.016, 0.80, 0.85, 0.002),
    "yelp": GraphSpec(716_847, 13_954_819, 300, 100, True, 0.75, 0.10, 0.80, 0.85, 0.004),
    "ogbn-papers100M": GraphSpec(111_059_956, 1_615_685_872, 128, 172, False, 0.011, 0.001, 0.85, 0.90, 0.0005),
    # Tiny shapes for tests/CI (no GPU, fast).
    "tiny": GraphSpec(200, 2_000, 16, 7, False, 0.60, 0.20, 0.8, 0.5, 0.2),
    "tiny-ml": GraphSpec(200, 2_000, 16, 5, True, 0.60, 0.20, 0.8, 0.5, 0.2),
}


def _draw_edges(spec: GraphSpec, n_nodes: int, n_edges: int, rng: np.random.Generator,
                chunk: int = 1 << 24) -> tuple[np.ndarray, np.ndarray]:
    """Vectorized degree-skewed + locality-mixed edge sampling, chunked to
    bound peak memory (papers100M-scale needs this)."""
    # Degree-skewed endpoints via the analytic inverse CDF of a continuous
    # power law (rank cdf ∝ r^(1-α) for α<1 → r = N·U^(1/(1-α))): O(1) per
    # draw, no searchsorted. Ranks are permuted so hubs are spread uniformly
    # over the id space (ids carry locality, not degree).
    perm = rng.permutation(n_nodes)
    inv_exp = 1.0 / (1.0 - spec.zipf_alpha)
    window = max(1, int(spec.window_frac * n_nodes))
    dtype = np.int32 if n_nodes < 2**31 else np.int64

    def draw(m):
        r = (rng.random(m) ** inv_exp * n_nodes).astype(np.int64)
        np.clip(r, 0, n_nodes - 1, out=r)
        return perm[r].astype(dtype)

    srcs, dsts = [], []
    remaining = n_edges
    while remaining > 0:
        m = min(chunk, remaining)
        src = draw(m)
        # locality mixture for destinations
        local = rng.random(m) < spec.locality
        n_loc = int(local.sum())
        dst = np.empty(m, dtype=dtype)
        off = rng.integers(-window, window + 1, size=n_loc)
        dst[local] = np.clip(src[local].astype(np.int64) + off, 0, n_nodes - 1).astype(dtype)
        dst[~local] = draw(m - n_loc)
        srcs.append(src)
        dsts.append(dst)
        remaining -= m
    return np.concatenate(srcs), np.concatenate(dsts)


def load_data(name: str, seed: int = 0, scale: float = 1.0) -> Graph:
    """Generate the named synthetic dataset deterministically from `seed`.

    `scale` < 1 shrinks node/edge counts proportionally (used for smoke
    tests of papers100M-shaped runs).
    Mirrors the loader post-processing of the reference
    (helper/utils.py:37-70): self-loops removed and re-added, multilabel
    float labels for Yelp, masks as boolean node masks.
    """
    spec = DATASETS[name]
    n_nodes = max(16, int(spec.n_nodes * scale))
    n_edges = max(64, int(spec.n_edges * scale))
    # NOTE: deterministic name hash — python's hash() is randomized per
    # process (PYTHONHASHSEED), which would make every process regenerate a
    # DIFFERENT graph (a bug this caused: rank-0's evaluator scored a model
    # against a different random dataset than it trained on)
    rng = np.random.default_rng(
        np.random.SeedSequence([zlib.crc32(name.encode()) & 0x7FFFFFFF, seed]))

    src, dst = _draw_edges(spec, n_nodes, n_edges, rng)
    src, dst = add_self_loops(src, dst, n_nodes)
    adj_in = CSR.from_edges(src, dst, n_nodes, n_nodes)
    del src, dst

    feat = rng.standard_normal((n_nodes, spec.n_feat), dtype=np.float32)
    # LEARNABLE labels: class scores are a random linear map of the node's
    # own features plus its (cheaply approximated) neighborhood mean, with
    # noise — so a GNN can actually fit them and evaluation accuracy is a
    # meaningful end-to-end signal (purely random labels would pin accuracy
    # at chance regardless of training).
    proto = rng.standard_normal((spec.n_feat, spec.n_class)).astype(np.float32)
    z = feat @ proto                                   # [N, C] class scores
    # neighborhood term from the first K CSR neighbors per node (a cheap
    # O(N*K) stand-in for the true neighbor mean — labels only)
    K = 8
    indptr, indices = adj_in.indptr, adj_in.indices
    pos = indptr[:-1, None] + np.arange(K)
    valid = pos < indptr[1:, None]
    neigh = indices[np.clip(pos, 0, max(len(indices) - 1, 0))]
    zm = (z[neigh] * valid[:, :, None]).sum(1) / np.maximum(
        valid.sum(1), 1)[:, None]
    scores = 0.7 * z + 0.3 * zm
    scores += 0.5 * rng.standard_normal(scores.shape).astype(np.float32)
    if spec.multilabel:
        thresh = np.quantile(scores, 0.94, axis=0)
        label = (scores > thresh).astype(np.float32)
    else:
        label = scores.argmax(1).astype(np.int64)
    del z, zm, scores

    r = rng.random(n_nodes)
    train_mask = r < spec.train_frac
    val_mask = (r >= spec.train_frac) & (r < spec.train_frac + spec.val_frac)
    test_mask = ~(train_mask | val_mask)
    return Graph(adj_in, feat, label, train_mask, val_mask, test_mask,
                 spec.n_class, spec.multilabel, name=name)
