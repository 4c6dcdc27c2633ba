"""Graph partitioner (offline, CPU, vectorized numpy).

Reference counterpart: helper/utils.py:73-98 (`graph_partition`, which
delegates to DGL/METIS). libmetis/DGL are not available in this image, so:

* method="random": balanced random assignment (the reference's
  `--partition-method random`, which BNS-GCN's paper uses for papers100M);
* method="metis": a locality partitioner — balanced contiguous ranges of
  the node-id space. Our synthetic graphs (graph/synthetic.py) carry edge
  locality in the id space, so this plays the role METIS plays on the real
  datasets: a small boundary cut. The flag name is kept for CLI
  compatibility (helper/parser.py).

The partition objective flag (vol/cut) is accepted and recorded in
meta.json but does not change the algorithm.
"""
from __future__ import annotations

import numpy as np

from .csr import Graph
from .store import Partition, save_partitions


def assign_parts(n_nodes: int, n_parts: int, method: str, seed: int = 0,
                 adj=None) -> np.ndarray:
    if method == "random":
        rng = np.random.default_rng(seed)
        part = np.arange(n_nodes, dtype=np.int32) % n_parts
        rng.shuffle(part)
        return part
    if method == "metis":  # locality/contiguous (see module docstring)
        bounds = np.linspace(0, n_nodes, n_parts + 1).astype(np.int64)
        part = np.zeros(n_nodes, dtype=np.int32)
        for p in range(n_parts):
            part[bounds[p]:bounds[p + 1]] = p
        return part
    if method == "bfs":
        assert adj is not None, "bfs partitioning needs the adjacency"
        return _bfs_grow_parts(adj, n_parts, seed)
    raise ValueError(f"unknown partition method: {method}")


def _bfs_grow_parts(adj, n_parts: int, seed: int) -> np.ndarray:
    """Balanced multi-source BFS region growing — a cut-reducing
    partitioner that does NOT rely on node-id locality (the contiguous
    "metis" stand-in does; real-world node orderings may be arbitrary).
    P seeds grow breadth-first with a node-count capacity of ~N/P each;
    round-robin growth order rotates so no partition starves; leftover
    (unreached/over-capacity) nodes are filled round-robin.

    Vectorized per (partition, level): each expansion is one CSR gather
    over the current frontier — O(E) total like a plain BFS.
    """
    rng = np.random.default_rng(seed)
    n = adj.n_rows
    indptr, indices = adj.indptr, adj.indices
    cap = int(np.ceil(n / n_parts))
    part = np.full(n, -1, dtype=np.int32)
    sizes = np.zeros(n_parts, dtype=np.int64)
    seeds = rng.choice(n, size=n_parts, replace=False)
    frontiers: list[np.ndarray] = []
    for p, s in enumerate(seeds):
        part[s] = p
        sizes[p] = 1
        frontiers.append(np.array([s], dtype=np.int64))

    active = True
    order = list(range(n_parts))
    while active:
        active = False
        for p in order:
            if sizes[p] >= cap or len(frontiers[p]) == 0:
                continue
            f = frontiers[p]
            # neighbors of the frontier (gather all adjacency rows)
            lens = (indptr[f + 1] - indptr[f]).astype(np.int64)
            total = int(lens.sum())
            if total == 0:
                frontiers[p] = np.zeros(0, dtype=np.int64)
                continue
            starts = indptr[f]
            pos = np.arange(total)
            row_of = np.repeat(np.arange(len(f)), lens)
            offs = np.zeros(len(f) + 1, dtype=np.int64)
            np.cumsum(lens, out=offs[1:])
            nbr = indices[starts[row_of] + (pos - offs[row_of])].astype(np.int64)
            nbr = np.unique(nbr)
            nbr = nbr[part[nbr] < 0]
            room = cap - int(sizes[p])
            if len(nbr) > room:
                nbr = nbr[:room]
            if len(nbr):
                part[nbr] = p
                sizes[p] += len(nbr)
                frontiers[p] = nbr
                active = True
            else:
                frontiers[p] = np.zeros(0, dtype=np.int64)
        order = order[1:] + order[:1]   # rotate growth priority

    left = np.flatnonzero(part < 0)
    if len(left):
        # fill stragglers into the emptiest partitions
        fill_order = np.argsort(sizes)
        assign = np.empty(len(left), dtype=np.int32)
        i = 0
        for p in fill_order:
            take = min(len(left) - i, max(0, cap - int(sizes[p])))
            assign[i:i + take] = p
            sizes[p] += take
            i += take
            if i >= len(left):
                break
        if i < len(left):
            assign[i:] = np.arange(len(left) - i) % n_parts
        part[left] = assign
    return part


def partition_graph(g: Graph, n_parts: int, method: str = "metis", seed: int = 0,
                    objective: str = "vol") -> tuple[list[Partition], dict]:
    """Split `g` into per-rank Partition objects (see store.Partition)."""
    n = g.n_nodes
    part = assign_parts(n, n_parts, method, seed, adj=g.adj_in)

    # inner-local id of every node within its partition (sorted-global order)
    inner_local = np.zeros(n, dtype=np.int64)
    inner_lists = []
    for p in range(n_parts):
        nodes = np.flatnonzero(part == p)          # ascending global ids
        inner_local[nodes] = np.arange(len(nodes))
        inner_lists.append(nodes)

    src, dst = g.adj_in.to_edges()                  # dst is NONDECREASING (CSR order)
    spart = part[src]
    dpart = part[dst]

    parts: list[Partition] = []
    boundary_all: list[list[np.ndarray]] = [[None] * n_parts for _ in range(n_parts)]

    from .csr import CSR
    for p in range(n_parts):
        nodes = inner_lists[p]
        n_inner = len(nodes)
        # flatnonzero keeps ascending edge order, so dst stays nondecreasing
        # inside the bucket — per-partition CSRs need NO re-sort.
        eidx = np.flatnonzero(dpart == p)
        e_src, e_dst = src[eidx], dst[eidx]
        e_spart = spart[eidx]
        dst_local = inner_local[e_dst].astype(np.int32)

        # inner->inner edges: rows (dst_local) already nondecreasing
        im = e_spart == p
        i_dst = dst_local[im]
        counts = np.bincount(i_dst, minlength=n_inner).astype(np.int64)
        inner_indptr = np.zeros(n_inner + 1, dtype=np.int64)
        np.cumsum(counts, out=inner_indptr[1:])
        inner_csr = CSR(inner_indptr, inner_local[e_src[im]].astype(np.int32),
                        n_inner)

        # halo edges grouped by (owner, owner-local id)
        hm = ~im
        h_owner = e_spart[hm]
        h_ol = inner_local[e_src[hm]].astype(np.int64)   # owner-local src id
        h_dst = dst_local[hm]
        # unique halo rows in (owner, owner_local) order == peer-major sorted
        key = h_owner.astype(np.int64) * n + h_ol
        uniq, inv = np.unique(key, return_inverse=True)
        halo_part = (uniq // n).astype(np.int32)
        halo_ol = (uniq % n).astype(np.int32)
        halo_csr = CSR.from_edges(h_dst, inv.astype(np.int64), len(uniq), n_inner)

        # halo degrees come straight from the full graph (global view)
        halo_global = np.empty(len(uniq), dtype=np.int64)
        for j in range(n_parts):
            m = halo_part == j
            if m.any():
                halo_global[m] = inner_lists[j][halo_ol[m]]
        halo_out_deg = g.out_deg[halo_global].astype(np.int32)
        halo_in_deg = g.in_deg[halo_global].astype(np.int32)

        # outgoing boundary of each OWNER j toward me (p): owner-local ids,
        # sorted — record into boundary_all[j][p]
        for j in range(n_parts):
            m = halo_part == j
            boundary_all[j][p] = halo_ol[m].copy()   # already sorted within peer

        parts.append(Partition(
            rank=p, n_parts=n_parts,
            inner_global_nid=nodes.astype(np.int64),
            feat=g.feat[nodes], label=g.label[nodes],
            train_mask=g.train_mask[nodes], val_mask=g.val_mask[nodes],
            test_mask=g.test_mask[nodes],
            in_deg=g.in_deg[nodes].astype(np.int32),
            out_deg=g.out_deg[nodes].astype(np.int32),
            inner_indptr=inner_csr.indptr, inner_indices=inner_csr.indices,
            halo_part=halo_part, halo_owner_local=halo_ol,
            halo_out_deg=halo_out_deg, halo_in_deg=halo_in_deg,
            halo_indptr=halo_csr.indptr, halo_indices=halo_csr.indices,
        ))

    for p in range(n_parts):
        parts[p].boundary = [
            boundary_all[p][j] if boundary_all[p][j] is not None
            else np.zeros(0, dtype=np.int32)
            for j in range(n_parts)]
        parts[p].boundary[p] = np.zeros(0, dtype=np.int32)

    meta = {
        "n_parts": n_parts, "method": method, "objective": objective, "seed": seed,
        "n_nodes": int(g.n_nodes), "n_edges": int(g.n_edges),
        "n_feat": int(g.n_feat), "n_class": int(g.n_class),
        "n_train": int(g.n_train), "multilabel": bool(g.multilabel),
        "dataset": g.name,
    }
    return parts, meta


def partition_and_save(g: Graph, n_parts: int, method: str, out_dir: str,
                       graph_name: str, seed: int = 0, objective: str = "vol",
                       extra_meta: dict | None = None) -> str:
    parts, meta = partition_graph(g, n_parts, method, seed, objective)
    if extra_meta:
        meta.update(extra_meta)
    return save_partitions(parts, meta, out_dir, graph_name)
