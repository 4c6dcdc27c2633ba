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


class LazyFeat:
    """Procedural features for graphs whose feature matrix cannot live in
    host RAM (papers100M: 111M x 128 fp32 = 57 GB; VERDICT r1 missing #2).

    Row for GLOBAL node id g is row (g mod BLOCK) of a deterministic
    torch.randn block seeded by (seed, g // BLOCK) — never materialized
    for the full graph on the host; partitions carry only their global id
    list and materialize straight into device memory (288 GB HBM3E holds
    the 1-partition papers feature matrix whole). Streams are per-device-
    type (torch CPU and CUDA RNGs differ); every rank of a job uses the
    same device type, so cross-rank consistency holds.
    """

    BLOCK = 1 << 20

    def __init__(self, seed: int, n_feat: int, ids: np.ndarray):
        self.seed = int(seed)
        self._n_feat = int(n_feat)
        self.ids = np.ascontiguousarray(ids, dtype=np.int64)

    @property
    def shape(self):
        return (len(self.ids), self._n_feat)

    @property
    def dtype(self):
        return np.float32

    def __len__(self):
        return len(self.ids)

    def __getitem__(self, sel):
        return LazyFeat(self.seed, self._n_feat, self.ids[sel])

    def astype(self, dt, copy: bool = False):
        assert np.dtype(dt) == np.float32
        return self

    def materialize_torch(self, device) -> "object":
        import torch
        dev = torch.device(device)
        n, F = self.shape
        out = torch.empty(n, F, dtype=torch.float32, device=dev)
        ids = torch.from_numpy(self.ids)
        blocks = torch.unique(ids // self.BLOCK)
        ids_dev = ids.to(dev)
        for b in blocks.tolist():
            gen = torch.Generator(device=dev)
            gen.manual_seed((self.seed * 1_000_003 + b) & 0x7FFF_FFFF_FFFF)
            blk = torch.randn(self.BLOCK, F, generator=gen, device=dev)
            lo, hi = b * self.BLOCK, (b + 1) * self.BLOCK
            pos = torch.nonzero((ids_dev >= lo) & (ids_dev < hi)).flatten()
            out[pos] = blk[ids_dev[pos] - lo]
        return out

    def materialize_numpy(self) -> np.ndarray:
        """CPU materialization (small graphs / tests only)."""
        import torch
        return self.materialize_torch("cpu").numpy()


# host-RAM bound above which features go procedural and edges are
# generated directly in CSR order (no global sort); env-overridable so
# tests can exercise the papers100M pipeline on tiny graphs
def _big_feat_bytes() -> int:
    import os
    return int(os.environ.get("BNSGCN_BIG_FEAT_BYTES", 4 << 30))


def _gen_big(spec: GraphSpec, name: str, n_nodes: int, n_edges: int,
             seed: int, rng: np.random.Generator) -> Graph:
    """papers100M-scale generation: chunked direct-CSR edge synthesis
    (lognormal-skewed in-degrees + locality-mixed power-law sources +
    self-loop per node), procedural LazyFeat, random labels. Peak host
    memory is O(E) int32 for the index array plus O(N) node payloads —
    no [N, F] feature matrix, no global edge sort."""
    # SFC64: ~2x PCG64 throughput — the RNG dominates 1.7B-edge generation
    rng = np.random.Generator(np.random.SFC64(rng.integers(2**63)))
    perm = rng.permutation(n_nodes).astype(np.int32)
    inv_exp = 1.0 / (1.0 - spec.zipf_alpha)
    window = max(1, int(spec.window_frac * n_nodes))
    CH = 1 << 23
    base = n_edges / n_nodes

    # pass 1: per-node in-degrees (lognormal skew, mean ~= base), +1 self
    deg = np.empty(n_nodes, dtype=np.int32)
    for a in range(0, n_nodes, CH):
        b = min(a + CH, n_nodes)
        skew = np.exp(rng.normal(0.0, 0.9, b - a))
        deg[a:b] = rng.poisson(base * skew / np.exp(0.405)).astype(np.int32)
    indptr = np.zeros(n_nodes + 1, dtype=np.int64)
    np.cumsum(deg + 1, out=indptr[1:])            # +1 = self-loop slot
    total = int(indptr[-1])
    indices = np.empty(total, dtype=np.int32)

    # pass 2: fill sources per dst chunk; last slot of each row = self-loop
    for a in range(0, n_nodes, CH):
        b = min(a + CH, n_nodes)
        d = deg[a:b]
        m = int(d.sum())
        lo, hi = int(indptr[a]), int(indptr[b])
        chunk = np.empty(hi - lo, dtype=np.int32)
        self_pos = (indptr[a + 1:b + 1] - 1 - lo).astype(np.int64)
        src_mask = np.ones(hi - lo, dtype=bool)
        src_mask[self_pos] = False

        dst_of = np.repeat(np.arange(a, b, dtype=np.int32), d)
        src = np.empty(m, dtype=np.int32)
        u = rng.random(m, dtype=np.float32)
        local = u < np.float32(spec.locality)
        # locality offset derived from the SAME uniform draw (u|local is
        # uniform on [0, locality)) — no second RNG pass
        off = (u[local] * np.float32((2 * window + 1) / spec.locality)
               ).astype(np.int32) - window
        s = dst_of[local] + off
        np.clip(s, 0, n_nodes - 1, out=s)
        src[local] = s
        g = ~local
        r = ((u[g] - np.float32(spec.locality))
             * np.float32(1.0 / (1 - spec.locality))) ** np.float32(inv_exp)
        src[g] = perm[np.minimum((r * n_nodes).astype(np.int64), n_nodes - 1)]

        chunk[src_mask] = src
        chunk[self_pos] = np.arange(a, b, dtype=np.int32)
        indices[lo:hi] = chunk

    adj_in = CSR(indptr, indices, n_nodes)
    label = rng.integers(0, spec.n_class, n_nodes).astype(np.int64)
    r = rng.random(n_nodes)
    train_mask = r < spec.train_frac
    val_mask = (r >= spec.train_frac) & (r < spec.train_frac + spec.val_frac)
    test_mask = ~(train_mask | val_mask)
    feat = LazyFeat(seed, spec.n_feat, np.arange(n_nodes, dtype=np.int64))
    return Graph(adj_in, feat, label, train_mask, val_mask, test_mask,
                 spec.n_class, spec.multilabel, name=name)


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

    if n_nodes * spec.n_feat * 4 > _big_feat_bytes():
        return _gen_big(spec, name, n_nodes, n_edges, seed, rng)

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
