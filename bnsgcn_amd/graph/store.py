"""On-disk partition store + in-memory Partition object.

Semantically equivalent to what the reference's `load_partition` yields
(reference: helper/utils.py:101-140 — local subgraph with inner + 1-hop halo
nodes, flat feature dict, partition book) but in our own layout, designed
for the MI355X runtime:

* inner nodes are local ids [0, n_inner) in sorted-global-id order;
* halo (boundary-in) nodes are NOT materialized as graph nodes — instead we
  store, per peer, the sorted owner-local ids of the halo nodes
  (`halo_owner_local`) and a halo-row-indexed CSR of halo→inner edges
  (`halo_csr`), which is exactly the shape the per-epoch sampled halo SpMM
  consumes (no per-epoch DGL-style graph rebuild; cf. reference
  train.py:256-281);
* full-graph in/out degrees travel with the store (the reference exchanges
  halo out-degrees at startup instead — train.py:148-167; we keep a runtime
  exchange too, used to cross-check the store, see parallel/boundary.py).

Directory layout (reference-compatible naming, main.py:18-24):
  <partition_dir>/<graph_name>/meta.json
  <partition_dir>/<graph_name>/part<rank>.npz
"""
from __future__ import annotations

import json
import os
from dataclasses import dataclass, field

import numpy as np


@dataclass
class Partition:
    rank: int
    n_parts: int
    # inner nodes
    inner_global_nid: np.ndarray      # int64 [n_inner], sorted ascending
    feat: np.ndarray                  # float32 [n_inner, F]
    label: np.ndarray                 # int64 [n_inner] or float32 [n_inner, C]
    train_mask: np.ndarray            # bool [n_inner]
    val_mask: np.ndarray
    test_mask: np.ndarray
    in_deg: np.ndarray                # int32 [n_inner] full-graph in-degree
    out_deg: np.ndarray               # int32 [n_inner] full-graph out-degree
    # inner->inner in-edges (row = inner dst, col = inner src)
    inner_indptr: np.ndarray          # int64 [n_inner+1]
    inner_indices: np.ndarray         # int32
    # halo rows: peer-major, within peer sorted by owner-local id
    halo_part: np.ndarray             # int32 [n_halo] owner rank per halo row
    halo_owner_local: np.ndarray      # int32 [n_halo] owner-local id per halo row
    halo_out_deg: np.ndarray          # int32 [n_halo] full-graph out-degree
    halo_in_deg: np.ndarray           # int32 [n_halo] full-graph in-degree
    # halo->inner edges (row = halo row index, col = inner dst)
    halo_indptr: np.ndarray           # int64 [n_halo+1]
    halo_indices: np.ndarray          # int32
    # outgoing boundary: per peer j, my inner-local ids that j needs,
    # sorted ascending == j's halo_owner_local for owner==me
    boundary: list = field(default_factory=list)   # list of int32 arrays, len n_parts
    meta: dict = field(default_factory=dict)

    @property
    def n_inner(self) -> int:
        return len(self.inner_global_nid)

    @property
    def n_halo(self) -> int:
        return len(self.halo_part)

    def halo_peer_slices(self) -> list[slice]:
        """slice of halo rows owned by each peer (empty slice for self)."""
        out = []
        for j in range(self.n_parts):
            lo, hi = np.searchsorted(self.halo_part, [j, j + 1])
            out.append(slice(int(lo), int(hi)))
        return out


def save_partitions(parts: list[Partition], meta: dict, out_dir: str, graph_name: str) -> str:
    from .synthetic import LazyFeat
    d = os.path.join(out_dir, graph_name)
    os.makedirs(d, exist_ok=True)
    if parts and isinstance(parts[0].feat, LazyFeat):
        # procedural features (papers100M scale): the store carries only
        # the generator seed — partitions rebuild LazyFeat from their
        # inner_global_nid at load and materialize on the training device
        meta = dict(meta)
        meta["procedural_feat"] = {"seed": parts[0].feat.seed,
                                   "n_feat": parts[0].feat.shape[1]}
    for p in parts:
        arrs = {k: getattr(p, k) for k in (
            "inner_global_nid", "feat", "label", "train_mask", "val_mask", "test_mask",
            "in_deg", "out_deg", "inner_indptr", "inner_indices",
            "halo_part", "halo_owner_local", "halo_out_deg", "halo_in_deg",
            "halo_indptr", "halo_indices")}
        if isinstance(p.feat, LazyFeat):
            arrs["feat"] = np.zeros((0, p.feat.shape[1]), dtype=np.float32)
        for j, b in enumerate(p.boundary):
            arrs[f"boundary_{j}"] = b
        np.savez(os.path.join(d, f"part{p.rank}.npz"), **arrs)
    # meta.json LAST: its presence marks the store complete (skip_partition
    # checks it — a crash mid-save must not leave a trusted partial store)
    with open(os.path.join(d, "meta.json"), "w") as f:
        json.dump(meta, f, indent=1)
    return d


def load_meta(part_dir: str, graph_name: str) -> dict:
    with open(os.path.join(part_dir, graph_name, "meta.json")) as f:
        return json.load(f)


def load_partition(part_dir: str, graph_name: str, rank: int) -> Partition:
    meta = load_meta(part_dir, graph_name)
    z = np.load(os.path.join(part_dir, graph_name, f"part{rank}.npz"))
    n_parts = int(meta["n_parts"])
    boundary = [z[f"boundary_{j}"] for j in range(n_parts)]
    kw = {k: z[k] for k in (
        "inner_global_nid", "feat", "label", "train_mask", "val_mask", "test_mask",
        "in_deg", "out_deg", "inner_indptr", "inner_indices",
        "halo_part", "halo_owner_local", "halo_out_deg", "halo_in_deg",
        "halo_indptr", "halo_indices")}
    pf = meta.get("procedural_feat")
    if pf:
        from .synthetic import LazyFeat
        kw["feat"] = LazyFeat(int(pf["seed"]), int(pf["n_feat"]),
                              kw["inner_global_nid"])
    return Partition(rank=rank, n_parts=n_parts, boundary=boundary, meta=meta, **kw)
