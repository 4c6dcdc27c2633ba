"""GPU (MI355X) numerics: every HIP kernel vs the pure-torch fp32
reference on the same inputs, plus an end-to-end training step.

Run on the GPU box: python -m pytest tests -m gpu -x -q
"""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from bnsgcn_amd.ops._ext import require_ext
    ext = require_ext()
else:
    ext = None

needs_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="no GPU")

from bnsgcn_amd.graph import CSR
from bnsgcn_amd.ops import reference as ref
from bnsgcn_amd.ops.philox import bns_keys


def rand_csr(n_rows, n_cols, e, seed=0):
    rng = np.random.default_rng(seed)
    src = rng.integers(0, n_cols, e)
    dst = rng.integers(0, n_rows, e)
    c = CSR.from_edges(src, dst, n_rows, n_cols)
    return (torch.from_numpy(c.indptr), torch.from_numpy(c.indices))


@needs_gpu
@pytest.mark.parametrize("F", [256, 602, 17, 512, 128, 64, 30])
def test_spmm_matches_reference(F):
    from bnsgcn_amd.ops.functional import spmm_sum_raw
    torch.manual_seed(0)
    indptr, indices = rand_csr(500, 700, 6000)
    x = torch.randn(700, F)
    ss = torch.rand(700) + 0.5
    ds = torch.rand(500) + 0.5
    want = ref.spmm_sum(indptr, indices, x, ss, ds)
    got = spmm_sum_raw(indptr.cuda(), indices.cuda(), x.cuda(), ss.cuda(),
                       ds.cuda()).cpu()
    torch.testing.assert_close(got, want, rtol=2e-5, atol=1e-5)
    # accumulate path + no-scale path
    base = torch.randn(500, F)
    got2 = spmm_sum_raw(indptr.cuda(), indices.cuda(), x.cuda(), None, None,
                        base.clone().cuda()).cpu()
    want2 = ref.spmm_sum(indptr, indices, x, None, None, base.clone())
    torch.testing.assert_close(got2, want2, rtol=2e-5, atol=1e-5)


@needs_gpu
def test_spmm_heavy_row_split():
    """Power-law CSR with a hub row >> SEG: exercises the work-list split
    + atomic combine path."""
    from bnsgcn_amd.ops.functional import spmm_sum_raw
    rng = np.random.default_rng(3)
    n_rows, n_cols = 300, 400
    dst = np.concatenate([np.zeros(9000, dtype=np.int64),          # hub row
                          rng.integers(0, n_rows, 4000)])
    src = rng.integers(0, n_cols, dst.shape[0])
    c = CSR.from_edges(src, dst, n_rows, n_cols)
    indptr, indices = torch.from_numpy(c.indptr), torch.from_numpy(c.indices)
    for F in (256, 31):
        x = torch.randn(n_cols, F)
        ss = torch.rand(n_cols) + 0.5
        ds = torch.rand(n_rows) + 0.5
        want = ref.spmm_sum(indptr, indices, x, ss, ds)
        got = spmm_sum_raw(indptr.cuda(), indices.cuda(), x.cuda(), ss.cuda(),
                           ds.cuda()).cpu()
        torch.testing.assert_close(got, want, rtol=1e-4, atol=1e-3)


@needs_gpu
@pytest.mark.parametrize("Fdim", [64, 31])
def test_pack_scatter_match(Fdim):
    torch.manual_seed(1)
    x = torch.randn(300, Fdim)
    idx = torch.randint(0, 300, (120,))
    scale = torch.rand(120) + 0.5
    want = ref.pack_rows(x, idx, scale)
    got = ext.pack_rows(x.cuda(), idx.cuda(), scale.cuda()).cpu()
    torch.testing.assert_close(got, want)
    out_c = torch.zeros(300, Fdim)
    src = torch.randn(120, Fdim)
    ref.scatter_add_rows(out_c, idx, src, scale)
    out_g = torch.zeros(300, Fdim).cuda()
    ext.scatter_add_rows(out_g, idx.cuda(), src.cuda(), scale.cuda())
    torch.testing.assert_close(out_g.cpu(), out_c, rtol=1e-5, atol=1e-5)


@needs_gpu
def test_philox_keys_bitwise_matches_numpy():
    for (n, seed, epoch, s, d) in [(1000, 42, 7, 1, 3), (5000, 123456789, 0, 0, 7)]:
        want = bns_keys(n, seed, epoch, s, d)
        got = ext.philox_keys(n, seed, epoch, s, d).cpu().numpy()
        np.testing.assert_array_equal(got, want)


@needs_gpu
@pytest.mark.parametrize("M,N,K", [(128, 64, 32), (300, 41, 602),
                                   (1000, 256, 256), (77, 33, 129)])
def test_gemm_matches_torch(M, N, K):
    torch.manual_seed(2)
    x = torch.randn(M, K).cuda()
    w = torch.randn(N, K).cuda()
    b = torch.randn(N).cuda()
    got = ext.gemm_nt_bias(x, w, b)
    want = torch.nn.functional.linear(x, w, b)
    torch.testing.assert_close(got, want, rtol=2e-5, atol=2e-4)
    g = torch.randn(M, N).cuda()
    torch.testing.assert_close(ext.gemm_nn(g, w), g @ w, rtol=2e-5, atol=2e-4)
    torch.testing.assert_close(ext.gemm_tn(g, x), g.t().contiguous() @ x,
                               rtol=2e-5, atol=2e-4)


@needs_gpu
def test_syncbn_stats():
    x = torch.randn(5000, 96).cuda()
    got = ext.syncbn_stats(x)
    want = torch.stack((x.sum(0), (x * x).sum(0)))
    torch.testing.assert_close(got, want, rtol=1e-4, atol=1e-3)


@needs_gpu
def test_gpu_sampling_matches_cpu_plan():
    """Device Philox sampling must select the same positions as numpy."""
    from bnsgcn_amd.ops.philox import sample_boundary
    n, s = 5000, 500
    keys = ext.philox_keys(n, 9, 3, 2, 5)
    order = torch.argsort(keys, stable=True)[:s]
    got = torch.sort(order)[0].cpu().numpy()
    want = sample_boundary(n, s, 9, 3, 2, 5)
    np.testing.assert_array_equal(got, want)


@needs_gpu
@pytest.mark.parametrize("model", ["graphsage", "gcn", "gat"])
def test_single_rank_training_step_gpu(model):
    """One-rank e2e on GPU: forward+backward+step, finite loss, and the
    loss decreases over a few epochs (HIP path exercised throughout)."""
    from bnsgcn_amd.graph import load_data, partition_graph
    from bnsgcn_amd.models.models import create_model
    from bnsgcn_amd.runtime.config import create_parser
    from bnsgcn_amd.runtime.trainer import RankState, _forward, forward_train_logits
    from bnsgcn_amd.parallel import GradReducer

    args = create_parser().parse_args([])
    args.dataset = "tiny"
    args.model = model
    args.n_layers = 2
    args.n_hidden = 32
    args.heads = 2
    args.sampling_rate = 1.0
    args.use_pp = model != "gat"
    args.dropout = 0.0
    torch.manual_seed(0)

    g = load_data("tiny", seed=0)
    parts, meta = partition_graph(g, 1, method="metis")
    parts[0].meta = meta
    state = RankState(parts[0], args, "cuda:0")
    state.plan.set_epoch(0)
    m = create_model(args, n_feat=g.n_feat, n_class=g.n_class,
                     train_size=g.n_train).to("cuda:0")
    if args.use_pp or model == "gat":
        state.precompute()
    reducer = GradReducer(m, g.n_train)
    opt = torch.optim.Adam(m.parameters(), lr=1e-2)
    lf = torch.nn.CrossEntropyLoss(reduction="sum")
    losses = []
    for ep in range(15):
        state.plan.set_epoch(ep)
        m.train()
        logits = forward_train_logits(m, state)
        loss = lf(logits, state.label[state.train_mask].long())
        reducer.zero_grad()
        loss.backward()
        reducer.synchronize()
        opt.step()
        losses.append(loss.item())
    assert np.isfinite(losses).all()
    assert losses[-1] < losses[0]


@needs_gpu
def test_fused_attn_dropout_softmax2():
    """Fused attention dropout in segment_softmax2: dropped weights equal
    a*mask/keep for the (regenerated-Philox) mask, the kept fraction is
    ~keep, and the backward equals the manual dropout∘softmax chain
    computed with the SAME mask."""
    from bnsgcn_amd.ops import functional as BF
    from bnsgcn_amd.ops import reference as ref
    torch.manual_seed(0)
    n, H = 500, 4
    dev = "cuda:0"
    deg1 = torch.randint(0, 6, (n,))
    deg2 = torch.randint(0, 4, (n,))
    ip1 = torch.zeros(n + 1, dtype=torch.long)
    ip1[1:] = deg1.cumsum(0)
    ip2 = torch.zeros(n + 1, dtype=torch.long)
    ip2[1:] = deg2.cumsum(0)
    ip1, ip2 = ip1.to(dev), ip2.to(dev)
    l1 = torch.randn(int(deg1.sum()), H, device=dev, requires_grad=True)
    l2 = torch.randn(int(deg2.sum()), H, device=dev, requires_grad=True)
    keep = 0.7

    da1, da2 = BF.segment_softmax2(l1, l2, ip1, ip2, p_drop=1 - keep)
    a1, a2 = BF.segment_softmax2_raw(ip1, l1.detach(), ip2, l2.detach())
    m1 = (da1 != 0).float()
    m2 = (da2 != 0).float()
    torch.testing.assert_close(da1, a1 * m1 / keep, rtol=1e-6, atol=1e-7)
    torch.testing.assert_close(da2, a2 * m2 / keep, rtol=1e-6, atol=1e-7)
    tot = m1.numel() + m2.numel()
    frac = float(m1.sum() + m2.sum()) / tot
    assert abs(frac - keep) < 0.03, frac

    g1 = torch.randn_like(da1)
    g2 = torch.randn_like(da2)
    (da1 * g1).sum().backward(retain_graph=True)
    (da2 * g2).sum().backward()

    # manual chain with the same mask on the CPU reference
    d1r, d2r = ref.segment_softmax2_backward(
        ip1.cpu(), a1.cpu(), (g1 * m1 / keep).cpu(),
        ip2.cpu(), a2.cpu(), (g2 * m2 / keep).cpu())
    torch.testing.assert_close(l1.grad.cpu(), d1r, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(l2.grad.cpu(), d2r, rtol=1e-4, atol=1e-5)


@needs_gpu
@pytest.mark.parametrize("F", [41, 64, 128, 256, 600, 1024])
@pytest.mark.parametrize("act", [False, True])
def test_layernorm_matches_torch(F, act):
    """HIP ln_fwd/ln_bwd (K7, + fused ReLU when act — K9) vs torch
    LayerNorm(+relu): outputs and all three gradients, including F not a
    multiple of the 64-lane wave."""
    from bnsgcn_amd.ops.functional import layer_norm
    torch.manual_seed(0)
    n = 4097
    x = (torch.randn(n, F, device="cuda:0") * 3 + 1).requires_grad_(True)
    w = torch.randn(F, device="cuda:0").requires_grad_(True)
    b = torch.randn(F, device="cuda:0").requires_grad_(True)
    y = layer_norm(x, w, b, act=act)
    g = torch.randn_like(y)
    y.backward(g)

    x2 = x.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)
    y2 = torch.nn.functional.layer_norm(x2, (F,), w2, b2, 1e-5)
    if act:
        y2 = torch.relu(y2)
    y2.backward(g)
    torch.testing.assert_close(y, y2, rtol=2e-5, atol=2e-5)
    torch.testing.assert_close(x.grad, x2.grad, rtol=2e-4, atol=2e-4)
    torch.testing.assert_close(w.grad, w2.grad, rtol=2e-3, atol=2e-3)
    torch.testing.assert_close(b.grad, b2.grad, rtol=2e-3, atol=2e-3)


@needs_gpu
def test_hip_dropout_unbiased_and_consistent():
    """Mask-free HIP dropout (K8): kept fraction ~= keep, kept values
    scaled 1/keep, and backward applies the SAME regenerated mask."""
    from bnsgcn_amd.ops.functional import dropout
    torch.manual_seed(0)
    x = torch.rand(1000, 257, device="cuda:0") + 1.0   # strictly positive
    x.requires_grad_(True)
    p = 0.4
    y = dropout(x, p, training=True)
    mask = (y != 0).float()
    frac = float(mask.mean())
    assert abs(frac - (1 - p)) < 0.01, frac
    torch.testing.assert_close(y, x * mask / (1 - p), rtol=1e-6, atol=1e-7)
    g = torch.randn_like(y)
    y.backward(g)
    torch.testing.assert_close(x.grad, g * mask / (1 - p),
                               rtol=1e-6, atol=1e-7)
    # eval mode: identity
    assert dropout(x, p, training=False) is x


@needs_gpu
def test_stream_overlap_ordering_stress(monkeypatch):
    """Stream-ordering stress: 25 sampled epochs with the
    side-stream halo overlap ON must be BITWISE identical to the
    sequential path (BNSGCN_NO_OVERLAP=1) and run-to-run deterministic.
    Heavy-row splits are disabled (SEG=1e9 -> no atomicAdd combines) so
    every kernel is deterministic — any missing cross-stream event
    (exchange vs SpMM, prefetch vs consume) shows up as a value change."""
    from bnsgcn_amd.graph import load_data, partition_graph
    from bnsgcn_amd.models.models import create_model
    from bnsgcn_amd.runtime.config import create_parser
    from bnsgcn_amd.runtime.trainer import RankState, forward_train_logits
    from bnsgcn_amd.parallel import GradReducer
    from bnsgcn_amd.ops import csr_torch
    monkeypatch.setattr(csr_torch, "SEG", 10**9)

    def train(no_overlap: bool):
        monkeypatch.setenv("BNSGCN_NO_OVERLAP", "1" if no_overlap else "0")
        args = create_parser().parse_args([])
        args.dataset = "tiny"
        args.model = "gcn"
        args.n_layers = 3
        args.n_hidden = 32
        args.sampling_rate = 0.4
        args.use_pp = False          # layer 0 exchanges too (more traffic)
        args.dropout = 0.2
        torch.manual_seed(11)
        g = load_data("tiny", seed=3)
        parts, meta = partition_graph(g, 1, method="metis")
        parts[0].meta = meta
        state = RankState(parts[0], args, "cuda:0")
        state.plan.set_epoch(0)
        m = create_model(args, n_feat=g.n_feat, n_class=g.n_class,
                         train_size=g.n_train).to("cuda:0")
        reducer = GradReducer(m, g.n_train)
        opt = torch.optim.Adam(m.parameters(), lr=1e-2, fused=True)
        lf = torch.nn.CrossEntropyLoss(reduction="sum")
        losses = []
        for ep in range(25):
            state.plan.set_epoch(ep)
            m.train()
            logits = forward_train_logits(m, state)
            loss = lf(logits, state.label[state.train_mask].long())
            reducer.zero_grad()
            loss.backward()
            if not no_overlap:
                state.prefetch(ep + 1)
            reducer.synchronize()
            opt.step()
            losses.append(loss.item())
        torch.cuda.synchronize()
        return np.array(losses)

    a1 = train(False)
    a2 = train(False)
    b = train(True)
    np.testing.assert_array_equal(a1, a2)   # run-to-run deterministic
    np.testing.assert_array_equal(a1, b)    # overlap == sequential


@needs_gpu
def test_plan_prefetch_trajectory_identical():
    """Side-stream plan prefetch (RankState.prefetch) must be trajectory-
    identical to building the sampling plan on the main stream — same
    Philox draws, only the stream changes."""
    from bnsgcn_amd.graph import load_data, partition_graph
    from bnsgcn_amd.models.models import create_model
    from bnsgcn_amd.runtime.config import create_parser
    from bnsgcn_amd.runtime.trainer import RankState, forward_train_logits
    from bnsgcn_amd.parallel import GradReducer

    def train(prefetch: bool):
        args = create_parser().parse_args([])
        args.dataset = "tiny"
        args.model = "graphsage"
        args.n_layers = 3
        args.n_hidden = 16
        args.sampling_rate = 0.5
        args.use_pp = True
        args.dropout = 0.0
        torch.manual_seed(0)
        g = load_data("tiny", seed=0)
        parts, meta = partition_graph(g, 1, method="metis")
        parts[0].meta = meta
        state = RankState(parts[0], args, "cuda:0")
        state.plan.set_epoch(0)
        m = create_model(args, n_feat=g.n_feat, n_class=g.n_class,
                         train_size=g.n_train).to("cuda:0")
        state.precompute()
        reducer = GradReducer(m, g.n_train)
        opt = torch.optim.Adam(m.parameters(), lr=1e-2)
        lf = torch.nn.CrossEntropyLoss(reduction="sum")
        losses = []
        for ep in range(12):
            state.plan.set_epoch(ep)
            m.train()
            logits = forward_train_logits(m, state)
            loss = lf(logits, state.label[state.train_mask].long())
            reducer.zero_grad()
            loss.backward()
            if prefetch:
                state.prefetch(ep + 1)
            reducer.synchronize()
            opt.step()
            losses.append(loss.item())
        torch.cuda.synchronize()
        return np.array(losses)

    a = train(prefetch=True)
    b = train(prefetch=False)
    np.testing.assert_array_equal(a, b)


@needs_gpu
def test_gpu_matches_cpu_training():
    """GPU single-rank loss trajectory ≈ CPU single-rank (same seed, no
    dropout): validates the whole HIP op set against the torch reference."""
    from bnsgcn_amd.graph import load_data, partition_graph
    from bnsgcn_amd.models.models import create_model
    from bnsgcn_amd.runtime.config import create_parser
    from bnsgcn_amd.runtime.trainer import RankState, _forward, forward_train_logits
    from bnsgcn_amd.parallel import GradReducer

    def train(device):
        args = create_parser().parse_args([])
        args.dataset = "tiny"
        args.model = "graphsage"
        args.n_layers = 3
        args.n_hidden = 32
        args.sampling_rate = 1.0
        args.use_pp = True
        args.dropout = 0.0
        torch.manual_seed(5)
        g = load_data("tiny", seed=0)
        parts, meta = partition_graph(g, 1, method="metis")
        parts[0].meta = meta
        state = RankState(parts[0], args, device)
        state.plan.set_epoch(0)
        m = create_model(args, n_feat=g.n_feat, n_class=g.n_class,
                         train_size=g.n_train).to(device)
        state.precompute()
        reducer = GradReducer(m, g.n_train)
        opt = torch.optim.Adam(m.parameters(), lr=1e-2)
        lf = torch.nn.CrossEntropyLoss(reduction="sum")
        losses = []
        for ep in range(10):
            m.train()
            logits = forward_train_logits(m, state)
            loss = lf(logits, state.label[state.train_mask].long())
            reducer.zero_grad()
            loss.backward()
            reducer.synchronize()
            opt.step()
            losses.append(loss.item())
        return np.array(losses)

    lc = train("cpu")
    lg = train("cuda:0")
    np.testing.assert_allclose(lg, lc, rtol=5e-3, atol=1e-3)


@needs_gpu
def test_gpu_matches_cpu_training_at_scale():
    """Same oracle at reddit shape scale 0.02 (4.6k nodes, ~2.3M edges,
    power-law hubs with degree >> SEG=512): engages the worklist
    heavy-row SPLIT + atomic-combine path, multi-wave scheduling, and the
    restricted final layer at a size the tiny graphs never reach
    (VERDICT r1 weak #7)."""
    from bnsgcn_amd.graph import load_data, partition_graph
    from bnsgcn_amd.models.models import create_model
    from bnsgcn_amd.runtime.config import create_parser
    from bnsgcn_amd.runtime.trainer import RankState, forward_train_logits
    from bnsgcn_amd.parallel import GradReducer

    g = load_data("reddit", seed=0, scale=0.02)
    assert int(np.diff(g.adj_in.indptr).max()) > 512  # splits engage

    def train(device):
        args = create_parser().parse_args([])
        args.model = "graphsage"
        args.n_layers = 3
        args.n_hidden = 64
        args.sampling_rate = 1.0
        args.use_pp = True
        args.dropout = 0.0
        torch.manual_seed(5)
        parts, meta = partition_graph(g, 1, method="contiguous")
        parts[0].meta = meta
        state = RankState(parts[0], args, device)
        state.plan.set_epoch(0)
        m = create_model(args, n_feat=g.n_feat, n_class=g.n_class,
                         train_size=g.n_train).to(device)
        state.precompute()
        reducer = GradReducer(m, g.n_train)
        opt = torch.optim.Adam(m.parameters(), lr=1e-2)
        lf = torch.nn.CrossEntropyLoss(reduction="sum")
        losses = []
        for ep in range(5):
            m.train()
            logits = forward_train_logits(m, state)
            loss = lf(logits, state.label[state.train_mask].long())
            reducer.zero_grad()
            loss.backward()
            reducer.synchronize()
            opt.step()
            losses.append(loss.item())
        return np.array(losses)

    lc = train("cpu")
    lg = train("cuda:0")
    # 2.3M-edge atomics + fp32 reduction-order differences accumulate a
    # little faster than on the tiny graphs
    np.testing.assert_allclose(lg, lc, rtol=2e-2, atol=1e-2)


@needs_gpu
@pytest.mark.parametrize("H,D", [(4, 128), (4, 100), (2, 8), (1, 64)])
def test_gat_kernels_match_reference(H, D):
    """GAT kernel set at the shapes the Yelp config actually hits (incl.
    D=100, the non-pow2 output layer that takes the general path)."""
    from bnsgcn_amd.ops.functional import (spmm_edge_raw, sddmm_dot_raw,
                                           sddmm_add_raw, segment_softmax_raw,
                                           segment_softmax_bwd_raw)
    torch.manual_seed(4)
    n_rows, n_cols, E = 400, 300, 5000
    indptr, indices = rand_csr(n_rows, n_cols, E, seed=7)
    w = torch.randn(E, H)
    x = torch.randn(n_cols, H, D)
    g = torch.randn(n_rows, H, D)
    cu = lambda t: t.cuda()
    want = ref.spmm_edge_sum(indptr, indices, w, x)
    got = spmm_edge_raw(cu(indptr), cu(indices), cu(w), cu(x)).cpu()
    torch.testing.assert_close(got, want, rtol=1e-4, atol=1e-4)
    want = ref.sddmm_dot(indptr, indices, g, x)
    got = sddmm_dot_raw(cu(indptr), cu(indices), cu(g), cu(x)).cpu()
    torch.testing.assert_close(got, want, rtol=1e-4, atol=1e-4)
    el = torch.randn(n_cols, H)
    er = torch.randn(n_rows, H)
    want = ref.sddmm_add(indptr, indices, el, er)
    got = sddmm_add_raw(cu(indptr), cu(indices), cu(el), cu(er)).cpu()
    torch.testing.assert_close(got, want, rtol=1e-5, atol=1e-5)
    logits = torch.randn(E, H)
    want = ref.segment_softmax(indptr, logits)
    got = segment_softmax_raw(cu(indptr), cu(logits)).cpu()
    torch.testing.assert_close(got, want, rtol=1e-4, atol=1e-5)
    ga = torch.randn(E, H)
    want = ref.segment_softmax_backward(indptr, want, ga)
    got = segment_softmax_bwd_raw(cu(indptr), got.cuda(), cu(ga)).cpu()
    torch.testing.assert_close(got, want, rtol=1e-4, atol=1e-4)


@needs_gpu
def test_segment_sum_edges_and_bincount():
    from bnsgcn_amd.ops.csr_torch import transpose_csr
    from bnsgcn_amd.ops.functional import _worklist_of
    H = 4
    indptr, indices = rand_csr(200, 150, 3000, seed=9)
    grad = torch.randn(3000, H)
    # g_er oracle: segment sum by row
    row = torch.repeat_interleave(torch.arange(200), indptr[1:] - indptr[:-1])
    want = torch.zeros(200, H).index_add(0, row, grad)
    ip, ix = indptr.cuda(), indices.cuda()
    got = ext.segment_sum_edges(*_worklist_of(ip)[:4], None, grad.cuda(), 200).cpu()
    torch.testing.assert_close(got, want, rtol=1e-4, atol=1e-4)
    # g_el oracle via transpose + eperm
    tip, tix, eperm = transpose_csr(ip, ix, 150)
    want2 = torch.zeros(150, H).index_add(0, indices.long(), grad)
    got2 = ext.segment_sum_edges(*_worklist_of(tip)[:4], eperm, grad.cuda(), 150).cpu()
    torch.testing.assert_close(got2, want2, rtol=1e-4, atol=1e-4)
    # bincount kernel
    v = torch.randint(0, 150, (3000,), dtype=torch.int32).cuda()
    torch.testing.assert_close(ext.bincount_i32(v, 150).cpu(),
                               torch.bincount(v.cpu().long(), minlength=150))


@needs_gpu
def test_train_and_eval_on_gpu_accuracy():
    """Full pipeline on GPU incl. --eval-device cuda: train tiny 300 epochs,
    evaluate on the GPU with the HIP kernels, accuracy well above chance."""
    from bnsgcn_amd.graph import load_data, partition_graph
    from bnsgcn_amd.models.models import create_model
    from bnsgcn_amd.runtime.config import create_parser
    from bnsgcn_amd.runtime.trainer import RankState, _forward, forward_train_logits, forward_train_logits, Evaluator
    from bnsgcn_amd.parallel import GradReducer

    args = create_parser().parse_args([])
    args.dataset = "tiny"
    args.model = "graphsage"
    args.n_layers = 2
    args.n_hidden = 32
    args.sampling_rate = 1.0
    args.use_pp = True
    args.dropout = 0.0
    args.lr = 0.05
    args.eval_device = "cuda:0"
    torch.manual_seed(7)

    g = load_data("tiny", seed=7)
    parts, meta = partition_graph(g, 1, method="metis")
    parts[0].meta = meta
    state = RankState(parts[0], args, "cuda:0")
    state.plan.set_epoch(0)
    m = create_model(args, n_feat=g.n_feat, n_class=g.n_class,
                     train_size=g.n_train).to("cuda:0")
    state.precompute()
    reducer = GradReducer(m, g.n_train)
    opt = torch.optim.Adam(m.parameters(), lr=args.lr)
    lf = torch.nn.CrossEntropyLoss(reduction="sum")
    for ep in range(300):
        m.train()
        logits = forward_train_logits(m, state)
        loss = lf(logits, state.label[state.train_mask].long())
        reducer.zero_grad()
        loss.backward()
        reducer.synchronize()
        opt.step()
    # dist_evaluate on GPU (world=1): exercises the full-state eval forward
    from bnsgcn_amd.runtime.trainer import dist_evaluate
    dres = dist_evaluate(state, m)
    assert 0.0 <= dres["test"] <= 1.0
    args.seed = 7
    ev = Evaluator(args)
    final = create_model(args, n_feat=g.n_feat, n_class=g.n_class,
                         train_size=g.n_train)
    final.load_state_dict({k: v.cpu() for k, v in m.state_dict().items()})
    out = ev.evaluate(final)
    assert out["test"] > 0.30, out
    # GPU dist eval == CPU full-graph evaluator on the same weights
    assert abs(dres["test"] - out["test"]) < 1e-4, (dres, out)


@needs_gpu
@pytest.mark.parametrize("H", [4, 3, 1, 16])   # pow2 -> interleaved kernel
def test_segment_softmax2_matches_reference(H):
    from bnsgcn_amd.ops.functional import segment_softmax2_raw
    ip1, _ = rand_csr(300, 10, 4000, seed=41)
    ip2, _ = rand_csr(300, 10, 700, seed=42)
    l1 = torch.randn(4000, H)
    l2 = torch.randn(700, H)
    w1, w2 = ref.segment_softmax2(ip1, l1, ip2, l2)
    g1, g2 = segment_softmax2_raw(ip1.cuda(), l1.cuda(), ip2.cuda(), l2.cuda())
    torch.testing.assert_close(g1.cpu(), w1, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(g2.cpu(), w2, rtol=1e-4, atol=1e-5)
    ga, gb = torch.randn_like(l1), torch.randn_like(l2)
    wd1, wd2 = ref.segment_softmax2_backward(ip1, w1, ga, ip2, w2, gb)
    from bnsgcn_amd.ops.functional import segment_softmax2_bwd_raw
    gd1, gd2 = segment_softmax2_bwd_raw(ip1.cuda(), g1, ga.cuda(),
                                        ip2.cuda(), g2, gb.cuda())
    torch.testing.assert_close(gd1.cpu(), wd1, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(gd2.cpu(), wd2, rtol=1e-4, atol=1e-5)


@needs_gpu
def test_gemm_empty_inputs():
    """0-row / 0-col GEMMs must not crash (empty-halo GAT at world size 1
    hit a divide-by-zero in split-K block sizing)."""
    x0 = torch.zeros(0, 300).cuda()
    w = torch.randn(64, 300).cuda()
    b = torch.randn(64).cuda()
    out = ext.gemm_nt_bias(x0, w, b)
    assert out.shape == (0, 64)
    g0 = torch.zeros(0, 64).cuda()
    assert ext.gemm_nn(g0, w).shape == (0, 300)
    dw = ext.gemm_tn(g0, x0)
    torch.testing.assert_close(dw, torch.zeros(64, 300).cuda())


@needs_gpu
@pytest.mark.parametrize("H,D", [(4, 128), (4, 100), (2, 8)])
def test_attn_project_gpu(H, D):
    from bnsgcn_amd.ops import functional as BF
    z = torch.randn(500, H, D).cuda().requires_grad_(True)
    al = torch.randn(1, H, D).cuda().requires_grad_(True)
    ar = torch.randn(1, H, D).cuda().requires_grad_(True)
    el, er = BF.attn_project(z, al, ar)
    z2 = z.detach().clone().requires_grad_(True)
    al2 = al.detach().clone().requires_grad_(True)
    ar2 = ar.detach().clone().requires_grad_(True)
    el2 = (z2 * al2).sum(-1)
    er2 = (z2 * ar2).sum(-1)
    torch.testing.assert_close(el, el2, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(er, er2, rtol=1e-4, atol=1e-4)
    g1, g2 = torch.randn_like(el), torch.randn_like(er)
    (el * g1 + er * g2).sum().backward()
    (el2 * g1 + er2 * g2).sum().backward()
    torch.testing.assert_close(z.grad, z2.grad, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(al.grad, al2.grad, rtol=1e-3, atol=1e-3)
    torch.testing.assert_close(ar.grad, ar2.grad, rtol=1e-3, atol=1e-3)


@needs_gpu
def test_syncbn_module_gpu_matches_cpu():
    from bnsgcn_amd.models.sync_bn import SyncBatchNorm
    torch.manual_seed(3)
    x = torch.randn(500, 24)
    def run(dev):
        torch.manual_seed(4)
        bn = SyncBatchNorm(24, whole_size=500).to(dev)
        bn.train()
        xx = x.clone().to(dev).requires_grad_(True)
        y = bn(xx)
        y.sum().backward()
        return (y.detach().cpu(), xx.grad.cpu(), bn.weight.grad.cpu(),
                bn.running_var.cpu())
    yc, gc, wc, rvc = run("cpu")
    yg, gg, wg, rvg = run("cuda:0")
    torch.testing.assert_close(yg, yc, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(gg, gc, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(wg, wc, rtol=1e-4, atol=1e-3)
    torch.testing.assert_close(rvg, rvc, rtol=1e-4, atol=1e-4)
