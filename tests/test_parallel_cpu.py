"""Distributed primitives (gloo, CPU): boundary discovery and halo-degree
exchange must reproduce what the partition store already knows
(reference protocol parity: get_boundary helper/utils.py:150-184,
collect_out_degree train.py:148-167)."""
import numpy as np
import torch

from bnsgcn_amd.graph import load_data, partition_graph

from util_dist import run_dist


def _check_rank(rank, world, parts_data):
    import torch.distributed as dist
    from bnsgcn_amd.parallel import (init_distributed, discover_boundary,
                                     exchange_halo_degrees, exchange_counts)
    init_distributed("gloo", rank, world)
    part = parts_data[rank]
    halo_part = torch.from_numpy(part["halo_part"])
    halo_ol = torch.from_numpy(part["halo_owner_local"])
    boundary = discover_boundary(halo_part, halo_ol, world)
    # matches the store's precomputed outgoing boundary
    for j in range(world):
        np.testing.assert_array_equal(boundary[j].numpy(),
                                      part["boundary"][j])
    # degree exchange returns the owners' true out-degrees for my halo
    recv_counts = [int((halo_part == j).sum()) for j in range(world)]
    my_deg = torch.from_numpy(part["out_deg"].astype(np.int64))
    deg = exchange_halo_degrees([b.to(torch.int64) for b in boundary],
                                my_deg, recv_counts)
    np.testing.assert_array_equal(deg.numpy(), part["halo_out_deg"])
    # exchange_counts sanity
    got = exchange_counts(torch.tensor([rank * 10 + j for j in range(world)],
                                       dtype=torch.int64))
    want = torch.tensor([j * 10 + rank for j in range(world)])
    assert torch.equal(got, want)
    return True


def test_boundary_discovery_matches_store():
    g = load_data("tiny", seed=11)
    parts, meta = partition_graph(g, 2, method="random", seed=3)
    data = [{"halo_part": p.halo_part, "halo_owner_local": p.halo_owner_local,
             "boundary": p.boundary, "out_deg": p.out_deg,
             "halo_out_deg": p.halo_out_deg} for p in parts]
    res = run_dist(2, _check_rank, (data,))
    assert all(res)
