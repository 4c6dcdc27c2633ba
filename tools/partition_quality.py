"""Report partition quality (edge cut, comm volume, balance) per method.

    python tools/partition_quality.py --dataset reddit --data-scale 0.1 \
        --n-partitions 8 [--methods metis random bfs contiguous] [--permute]

comm volume = total (owner-node, consumer-partition) boundary pairs —
exactly the per-layer BNS payload unit at sampling-rate 1.0.
--permute applies a random node-id permutation first (tests that a
method does not depend on planted id locality).
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from bnsgcn_amd.graph import CSR, load_data
from bnsgcn_amd.graph.partition import assign_parts


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--dataset", default="reddit")
    ap.add_argument("--data-scale", type=float, default=0.1)
    ap.add_argument("--n-partitions", type=int, default=8)
    ap.add_argument("--methods", nargs="+",
                    default=["metis", "bfs", "contiguous", "random"])
    ap.add_argument("--objective", default="cut", choices=["cut", "vol"])
    ap.add_argument("--permute", action="store_true")
    ap.add_argument("--seed", type=int, default=0)
    a = ap.parse_args()

    g = load_data(a.dataset, seed=a.seed, scale=a.data_scale)
    adj = g.adj_in
    n, P = g.n_nodes, a.n_partitions
    if a.permute:
        rng = np.random.default_rng(1)
        perm = rng.permutation(n)
        s, d = adj.to_edges()
        adj = CSR.from_edges(perm[s.astype(np.int64)],
                             perm[d.astype(np.int64)], n, n)
    src, dst = adj.to_edges()
    src = src.astype(np.int64)

    print(f"{a.dataset} scale={a.data_scale:g} n={n} e={adj.n_edges} "
          f"P={P} objective={a.objective} permuted={a.permute}")
    for m in a.methods:
        t0 = time.time()
        part = assign_parts(n, P, m, seed=a.seed, adj=adj,
                            objective=a.objective)
        dt = time.time() - t0
        cut = float((part[src] != part[dst]).mean())
        key = np.unique(src * P + part[dst])
        owners = part[key // P]
        vol = int((owners != (key % P)).sum())
        counts = np.bincount(part, minlength=P)
        print(f"  {m:10s} cut={cut:.4f}  comm_volume={vol}  "
              f"balance={counts.max() / (n / P):.3f}  t={dt:.1f}s")


if __name__ == "__main__":
    main()
