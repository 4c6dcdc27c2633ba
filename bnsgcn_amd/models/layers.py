"""GNN layers over a GraphContext.

Math matches the reference exactly (module/layer.py for GCN/SAGE; DGL
GATConv as instantiated at module/model.py:102 for GAT), but expressed on
our partition-aggregate / SDDMM / segment-softmax primitives:

* GCNLayer   — h' = Linear( in_norm^-1 · Σ_src out_norm^-1 · h )
               (layer.py:32-38; use_pp fast path layer.py:29-30)
* SAGELayer  — h' = Linear1(h_dst) + Linear2( (Σ_src h)/in_deg )
               (layer.py:85-92; use_pp path = one Linear over [h ‖ agg],
               layer.py:59,82-83)
* GATLayer   — multi-head attention: z=fc(h); e=LeakyReLU(a_l·z_src+a_r·z_dst);
               α=edge-softmax_dst(e); out=Σ α·z_src + bias (DGL GATConv
               semantics with feat_drop/attn_drop, negative_slope=0.2).

Parameter init mirrors the reference (uniform ±1/sqrt(fan_in),
layer.py:20-24,65-77; GATConv uses xavier with gain=sqrt(2)).
"""
from __future__ import annotations

import math

import torch
from torch import nn

from ..ops import functional as F
from .context import GraphContext


def _uniform_init(*tensors, fan_in):
    stdv = 1.0 / math.sqrt(fan_in)
    for t in tensors:
        nn.init.uniform_(t, -stdv, stdv)


class LayerNorm(nn.Module):
    """Drop-in affine nn.LayerNorm over the last dim: HIP ln_fwd/ln_bwd
    kernels on GPU (K7), torch on CPU. Same parameter names as
    nn.LayerNorm so `norm.<i>.weight/bias` state-dict keys (and reference
    checkpoints) stay compatible."""

    def __init__(self, dim: int, elementwise_affine: bool = True, eps=1e-5):
        super().__init__()
        assert elementwise_affine
        self.weight = nn.Parameter(torch.ones(dim))
        self.bias = nn.Parameter(torch.zeros(dim))
        self.eps = eps

    def forward(self, x, act: bool = False):
        """act=True fuses the following ReLU into the kernel pair (K9)."""
        return F.layer_norm(x, self.weight, self.bias, self.eps, act=act)


class Dropout(nn.Module):
    """Drop-in nn.Dropout: mask-free HIP kernel on GPU (K8 — splitmix64
    mask regenerated in backward), torch on CPU. No parameters, so
    state-dict keys are unchanged."""

    def __init__(self, p: float):
        super().__init__()
        self.p = float(p)

    def forward(self, x):
        return F.dropout(x, self.p, self.training)


class GCNLayer(nn.Module):
    """State-dict keys match the reference exactly (layers.<i>.linear.*,
    module/layer.py:17) so checkpoints are interchangeable."""

    def __init__(self, in_feats, out_feats, bias=True, use_pp=False):
        super().__init__()
        self.use_pp = use_pp
        self.linear = nn.Linear(in_feats, out_feats, bias=bias)
        _uniform_init(self.linear.weight, fan_in=in_feats)
        if self.linear.bias is not None:
            _uniform_init(self.linear.bias, fan_in=in_feats)

    def forward(self, ctx: GraphContext, x, rows: torch.Tensor | None = None):
        """rows: final-layer loss-row restriction — output [len(rows), F]
        (identical math for those rows; see GraphContext.aggregate)."""
        if self.training and self.use_pp:
            if rows is not None:
                x = x[rows]
            return F.linear(x, self.linear.weight, self.linear.bias)
        h = ctx.aggregate(x, "gcn", rows=rows)
        return F.linear(h, self.linear.weight, self.linear.bias)


class SAGELayer(nn.Module):
    """State-dict keys match the reference (layers.<i>.linear.* under
    use_pp, layers.<i>.linear{1,2}.* otherwise — module/layer.py:59-62)."""

    def __init__(self, in_feats, out_feats, bias=True, use_pp=False):
        super().__init__()
        self.use_pp = use_pp
        if use_pp:
            self.linear = nn.Linear(2 * in_feats, out_feats, bias=bias)
            _uniform_init(self.linear.weight, fan_in=2 * in_feats)
            if self.linear.bias is not None:
                _uniform_init(self.linear.bias, fan_in=2 * in_feats)
        else:
            self.linear1 = nn.Linear(in_feats, out_feats, bias=bias)
            self.linear2 = nn.Linear(in_feats, out_feats, bias=bias)
            _uniform_init(self.linear1.weight, self.linear2.weight,
                          fan_in=in_feats)
            if bias:
                _uniform_init(self.linear1.bias, self.linear2.bias,
                              fan_in=in_feats)

    def forward(self, ctx: GraphContext, x, rows: torch.Tensor | None = None):
        if self.training and self.use_pp:
            # x = [feat ‖ precomputed neighbor mean], width 2F
            if rows is not None:
                x = x[rows]
            return F.linear(x, self.linear.weight, self.linear.bias)
        ah = ctx.aggregate(x, "mean", rows=rows)
        x_dst = x[rows] if rows is not None else x
        if self.use_pp:  # eval path of a pp layer (reference layer.py:98-100)
            return F.linear(torch.cat((x_dst, ah), dim=1), self.linear.weight,
                            self.linear.bias)
        return (F.linear(x_dst, self.linear1.weight, self.linear1.bias)
                + F.linear(ah, self.linear2.weight, self.linear2.bias))


class GATLayer(nn.Module):
    def __init__(self, in_feats, out_feats, heads, feat_drop=0.0, attn_drop=0.0,
                 negative_slope=0.2, bias=True, use_pp=False):
        super().__init__()
        self.heads, self.out_feats = heads, out_feats
        self.use_pp = use_pp
        # param names follow DGL GATConv (fc.weight, attn_l, attn_r, bias)
        # so reference GAT checkpoints load directly
        self.fc = nn.Linear(in_feats, heads * out_feats, bias=False)
        self.attn_l = nn.Parameter(torch.empty(1, heads, out_feats))
        self.attn_r = nn.Parameter(torch.empty(1, heads, out_feats))
        self.bias = nn.Parameter(torch.zeros(heads * out_feats)) if bias else None
        self.feat_drop = Dropout(feat_drop)
        self.attn_drop = nn.Dropout(attn_drop)   # p only; fused in softmax2
        self.negative_slope = negative_slope
        gain = math.sqrt(2.0)
        nn.init.xavier_normal_(self.fc.weight, gain=gain)
        nn.init.xavier_normal_(self.attn_l, gain=gain)
        nn.init.xavier_normal_(self.attn_r, gain=gain)

    def forward(self, ctx: GraphContext, x, halo_feat: torch.Tensor | None = None,
                rows: torch.Tensor | None = None):
        """x: [n_local, F]. In partition mode, halo sources are fetched via
        halo_exchange — except when `halo_feat` is given (GAT layer 0 under
        use_pp: the FULL unsampled halo features captured at precompute,
        reference model.py:118-120 / train.py:208-209).

        Partition mode uses the SPLIT block: attention logits are computed
        over the static inner edge set and the per-epoch sampled-halo edge
        set separately, normalized jointly with the union softmax
        (ops segment_softmax2) — no per-epoch merge/transpose of the big
        combined CSR. Evaluation (full graph, no plan) keeps the single
        combined path."""
        from ..parallel.halo import halo_exchange

        h = self.feat_drop(x)
        H, D = self.heads, self.out_feats
        if ctx.plan is None:                       # full-graph eval path
            ip, ix, tip, tix, eperm, _ = ctx.gat_block()
            z = F.linear(h, self.fc.weight).view(-1, H, D)
            el, er = F.attn_project(z, self.attn_l, self.attn_r)
            logits = F.sddmm_add(el, er, ip, ix, tip, tix, eperm,
                                 slope=self.negative_slope)
            alpha = self.attn_drop(F.segment_softmax(logits, ip))
            out = F.spmm_edge_sum(z, alpha, ip, ix, tip, tix, eperm)
            return out + self.bias.view(1, H, D) if self.bias is not None else out

        if halo_feat is not None:                  # layer 0 under use_pp
            src_halo = self.feat_drop(halo_feat)
            hip, hix, hbip, hbix, heperm = ctx.gat_split_full()
        else:
            src_halo = halo_exchange(h, ctx.plan)
            hip, hix, hbip, hbix, heperm = ctx.gat_split_halo()

        z_in = F.linear(h, self.fc.weight).view(-1, H, D)
        z_h = F.linear(src_halo, self.fc.weight).view(-1, H, D)
        el_in, er = F.attn_project(z_in, self.attn_l, self.attn_r)
        el_h, _ = F.attn_project(z_h, self.attn_l, self.attn_r)  # halo: el
        slope = self.negative_slope
        if rows is not None:
            # final-layer loss-row restriction: edge sets gathered to the
            # labeled dst rows (identical math — unlabeled rows carry zero
            # gradient); sources (z/el) stay full, er shrinks to the rows
            iip, iix, itip, itix, ieperm = ctx.gat_rows_inner()
            hip, hix, hbip, hbix, heperm = ctx.gat_rows_halo(ctx.plan.state)
            er = er[rows]
            inner5 = (iip, iix, itip, itix, ieperm)
        else:
            inner5 = (ctx.indptr, ctx.indices, ctx.t_indptr, ctx.t_indices,
                      ctx.t_eperm)
        li = F.sddmm_add(el_in, er, *inner5, slope=slope)
        lh = F.sddmm_add(el_h, er, hip, hix, hbip, hbix, heperm, slope=slope)
        # attention dropout FUSED into the union softmax (no separate
        # dropout kernels/masks; identity when eval or p=0)
        p_attn = self.attn_drop.p if self.training else 0.0
        ai, ah = F.segment_softmax2(li, lh, inner5[0], hip, p_drop=p_attn)
        out = F.spmm_edge_sum2(z_in, ai, z_h, ah, inner5,
                               (hip, hix, hbip, hbix, heperm))
        if self.bias is not None:
            out = out + self.bias.view(1, H, D)
        return out
