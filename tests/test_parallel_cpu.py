"""Distributed primitives (gloo, CPU): boundary discovery and halo-degree
exchange must reproduce what the partition store already knows
(reference protocol parity: get_boundary helper/utils.py:150-184,
collect_out_degree train.py:148-167)."""
import numpy as np
import pytest
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


def _syncbn_rank(rank, world, xs):
    from bnsgcn_amd.parallel import init_distributed
    from bnsgcn_amd.models.sync_bn import SyncBatchNorm
    import torch
    init_distributed("gloo", rank, world)
    torch.manual_seed(0)
    bn = SyncBatchNorm(xs[0].shape[1], whole_size=sum(x.shape[0] for x in xs))
    bn.train()
    x = xs[rank].clone().requires_grad_(True)
    y = bn(x)
    g = torch.ones_like(y) * (rank + 1)
    y.backward(g.clone())
    # return numpy copies: torch tensors through an mp queue travel as
    # shared-memory handles that vanish when the child exits
    return {k: v.detach().numpy().copy() for k, v in
            {"y": y, "dx": x.grad, "dw": bn.weight.grad, "db": bn.bias.grad,
             "rm": bn.running_mean, "rv": bn.running_var}.items()}


def test_syncbn_matches_single_process_batchnorm():
    """2-rank SyncBN == torch BatchNorm1d over the concatenated rows
    (reference module/sync_bn.py semantics with whole_size = total rows)."""
    import torch
    torch.manual_seed(3)
    xs = [torch.randn(40, 6), torch.randn(24, 6)]
    res = run_dist(2, _syncbn_rank, (xs,))
    xfull = torch.cat(xs).requires_grad_(True)
    bn = torch.nn.BatchNorm1d(6, eps=1e-5, momentum=0.1)
    bn.train()
    yfull = bn(xfull)
    gfull = torch.cat([torch.ones_like(xs[0]) * 1, torch.ones_like(xs[1]) * 2])
    yfull.backward(gfull)
    res = [{k: torch.from_numpy(v) for k, v in r.items()} for r in res]
    ycat = torch.cat([res[0]["y"], res[1]["y"]])
    torch.testing.assert_close(ycat, yfull.detach(), rtol=1e-4, atol=1e-5)
    dxcat = torch.cat([res[0]["dx"], res[1]["dx"]])
    torch.testing.assert_close(dxcat, xfull.grad, rtol=1e-4, atol=1e-5)
    # weight/bias grads are all-reduced: same on both ranks, equal to full
    torch.testing.assert_close(res[0]["dw"], bn.weight.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(res[0]["db"], bn.bias.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(res[0]["rm"], bn.running_mean, rtol=1e-4, atol=1e-5)
    # running_var: torch tracks the UNBIASED variance; the reference's
    # SyncBN (module/sync_bn.py:21) and ours track the biased one
    n = xfull.shape[0]
    ours_unbiased = (res[0]["rv"] - 0.9) * n / (n - 1) + 0.9  # undo momentum mix
    torch.testing.assert_close(ours_unbiased, bn.running_var, rtol=1e-3, atol=1e-4)


def _agg_rank(rank, world, parts_data, rate, n_epochs, mode="mean"):
    """Average the sampled partition-aggregate over many epochs; the
    1/ratio-unbiased estimator must converge to the full (p=1.0)
    aggregation (reference estimator semantics,
    feature_buffer.py:117,129)."""
    import torch
    from bnsgcn_amd.parallel import init_distributed, HaloPlan
    from bnsgcn_amd.models.context import GraphContext
    init_distributed("gloo", rank, world)
    part = parts_data[rank]
    feat = torch.from_numpy(part.feat)

    plan_full = HaloPlan(part, 1.0, seed=5, device="cpu")
    ctx_full = GraphContext.for_partition(part, plan_full, "cpu")
    plan_full.set_epoch(0)
    exact = ctx_full.aggregate(feat, mode)

    plan = HaloPlan(part, rate, seed=5, device="cpu")
    ctx = GraphContext.for_partition(part, plan, "cpu")
    acc = torch.zeros_like(exact)
    for ep in range(n_epochs):
        plan.set_epoch(ep)
        acc += ctx.aggregate(feat, mode)
    mean_est = acc / n_epochs
    err = (mean_est - exact).norm() / exact.norm()
    return float(err)


def _reducer_rank(rank, world, overlap, bucket_bytes):
    """Train a tiny MLP (one branch unused some steps) for 4 steps; return
    the full grad arena after each synchronize. With overlap=True the
    buckets launch from post-accumulate-grad hooks DURING backward; the
    reduced values must be bit-identical to the post-backward path
    (reference train.py:337-338 hook overlap)."""
    import torch
    from bnsgcn_amd.parallel import init_distributed
    from bnsgcn_amd.parallel.reducer import GradReducer
    init_distributed("gloo", rank, world)
    torch.manual_seed(7)

    class Net(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.a = torch.nn.Linear(17, 33)   # odd sizes: params straddle
            self.b = torch.nn.Linear(33, 9)    # tiny-bucket boundaries
            self.unused = torch.nn.Linear(5, 5)

        def forward(self, x, use_extra):
            h = torch.relu(self.a(x))
            out = self.b(h)
            if use_extra:
                out = out + self.unused(out[:, :5]).sum() * 0
            return out

    net = Net()
    red = GradReducer(net, n_train_global=10, bucket_bytes=bucket_bytes,
                      overlap=overlap)
    arenas = []
    for step in range(4):
        torch.manual_seed(100 + step)       # same data on both ranks except
        x = torch.randn(6, 17) + rank       # the rank shift
        red.zero_grad()
        loss = net(x, use_extra=(step % 2 == 0)).pow(2).sum()
        loss.backward()
        red.reduce()
        red.synchronize()
        arenas.append(red.flat.detach().numpy().copy())
    return arenas


@pytest.mark.parametrize("bucket_bytes", [128, 1 << 20])
def test_reducer_overlap_bit_identical(bucket_bytes):
    a = run_dist(2, _reducer_rank, (True, bucket_bytes))
    b = run_dist(2, _reducer_rank, (False, bucket_bytes))
    for r in range(2):
        for sa, sb in zip(a[r], b[r]):
            np.testing.assert_array_equal(sa, sb)
    # grads identical across ranks after all-reduce
    for sa, sb in zip(a[0], a[1]):
        np.testing.assert_array_equal(sa, sb)


@pytest.mark.parametrize("mode", ["mean", "gcn"])
def test_bns_estimator_unbiased(mode):
    from bnsgcn_amd.graph import load_data, partition_graph
    g = load_data("tiny", seed=21)
    parts, meta = partition_graph(g, 2, method="random", seed=1)
    errs = run_dist(2, _agg_rank, (parts, 0.5, 300, mode), timeout=600)
    # Monte-Carlo average over 300 epochs: relative error shrinks well
    # below the single-sample deviation (~0.3 at p=0.5)
    assert max(errs) < 0.05, errs
