"""GCN / GraphSAGE / GAT models (reference: module/model.py).

Structure preserved: `layer_size` list of widths; the last `n_linear`
layers are plain Linear (MLP tail, used by Yelp); LayerNorm or SyncBN +
activation between layers (not after the last); dropout before every
conv/linear for GCN/SAGE, only before the linear tail for GAT (GAT carries
its own feat_drop/attn_drop); GAT head-means after every conv
(model.py:124). State-dict naming keeps `layers.<i>.*` / `norm.<i>.*` like
the reference so checkpoints stay structurally compatible.
"""
from __future__ import annotations

from torch import nn

import torch.nn.functional as TF

from .context import GraphContext
from .layers import GCNLayer, SAGELayer, GATLayer, LayerNorm, Dropout
from .sync_bn import SyncBatchNorm


class GNNBase(nn.Module):
    def __init__(self, layer_size, activation, use_pp=False, dropout=0.5,
                 norm="layer", n_linear=0, train_size=None):
        super().__init__()
        self.n_layers = len(layer_size) - 1
        self.layers = nn.ModuleList()
        self.activation = activation
        self.use_pp = use_pp
        self.n_linear = n_linear
        self.use_norm = norm in ("layer", "batch")
        if self.use_norm:
            self.norm = nn.ModuleList()
            for i in range(self.n_layers - 1):
                if norm == "layer":
                    self.norm.append(LayerNorm(layer_size[i + 1],
                                               elementwise_affine=True))
                else:
                    self.norm.append(SyncBatchNorm(layer_size[i + 1], train_size))
        self.dropout = Dropout(dropout)

    def _post(self, i, h):
        if i < self.n_layers - 1:
            if self.use_norm:
                nm = self.norm[i]
                if isinstance(nm, LayerNorm) and self.activation is TF.relu:
                    return nm(h, act=True)     # fused LN+ReLU (K7+K9)
                h = nm(h)
            h = self.activation(h)
        return h


class GCN(GNNBase):
    conv_cls = GCNLayer
    agg_mode = "gcn"

    def __init__(self, layer_size, activation, use_pp=False, dropout=0.5,
                 norm="layer", n_linear=0, train_size=None):
        super().__init__(layer_size, activation, use_pp, dropout, norm,
                         n_linear, train_size)
        pp = use_pp
        for i in range(self.n_layers):
            if i < self.n_layers - self.n_linear:
                self.layers.append(self.conv_cls(layer_size[i], layer_size[i + 1],
                                                 use_pp=pp))
            else:
                self.layers.append(nn.Linear(layer_size[i], layer_size[i + 1]))
            pp = False

    def forward(self, ctx: GraphContext, feat):
        """When training and ctx.loss_rows is set, the FINAL layer computes
        only the loss rows (logits shape [len(loss_rows), C]) — identical
        math, since unlabeled rows carry zero gradient (differences are
        GEMM reduction-order float noise). Dropout stays full-shaped so
        RNG draws match the unrestricted run."""
        h = feat
        restrict = self.training and getattr(ctx, "loss_rows", None) is not None
        for i in range(self.n_layers):
            rows = ctx.loss_rows if (restrict and i == self.n_layers - 1) \
                else None
            h = self.dropout(h)
            if i < self.n_layers - self.n_linear:
                h = self.layers[i](ctx, h, rows=rows)
            else:
                h = self.layers[i](h[rows] if rows is not None else h)
            h = self._post(i, h)
        return h


class GraphSAGE(GCN):
    conv_cls = SAGELayer
    agg_mode = "mean"


class GAT(GNNBase):
    def __init__(self, layer_size, activation, use_pp=False, heads=1,
                 dropout=0.5, norm="layer", n_linear=0, train_size=None):
        super().__init__(layer_size, activation, use_pp, dropout, norm,
                         n_linear, train_size)
        for i in range(self.n_layers):
            if i < self.n_layers - self.n_linear:
                self.layers.append(GATLayer(layer_size[i], layer_size[i + 1],
                                            heads, feat_drop=dropout,
                                            attn_drop=dropout,
                                            use_pp=(use_pp and i == 0)))
            else:
                self.layers.append(nn.Linear(layer_size[i], layer_size[i + 1]))

    def forward(self, ctx: GraphContext, feat, halo_feat0=None):
        """halo_feat0: full-halo raw features for layer 0 under use_pp
        (captured once at precompute; reference train.py:208-209).
        Training with ctx.loss_rows set restricts the FINAL layer to the
        labeled rows (same math; see GNNBase/GCN.forward)."""
        h = feat
        restrict = (self.training and ctx.plan is not None
                    and getattr(ctx, "loss_rows", None) is not None)
        for i in range(self.n_layers):
            last_conv_restrict = (restrict and i == self.n_layers - 1
                                  and not (self.use_pp and i == 0))
            if i < self.n_layers - self.n_linear:
                lay = self.layers[i]
                rows = ctx.loss_rows if last_conv_restrict else None
                if (self.training and self.use_pp and i == 0
                        and ctx.plan is not None):
                    h = lay(ctx, h, halo_feat=halo_feat0)
                else:
                    h = lay(ctx, h, rows=rows)
                h = h.mean(1)  # head mean (model.py:124)
            else:
                h = self.dropout(h)
                if restrict and i == self.n_layers - 1:
                    h = h[ctx.loss_rows]
                h = self.layers[i](h)
            h = self._post(i, h)
        if restrict and h.shape[0] != ctx.loss_rows.shape[0]:
            h = h[ctx.loss_rows]   # 1-layer use_pp edge case
        return h


def create_model(args, n_feat, n_class, train_size) -> nn.Module:
    """Reference: train.py:214-222. layer_size = [in, hidden×(L-1), out];
    use_pp SAGE doubles the first width via the pp-linear (handled in
    SAGELayer); GAT forces use_pp semantics the reference way
    (train.py:222: GAT always use_pp=True)."""
    import torch.nn.functional as TF
    hidden = [args.n_hidden] * (args.n_layers - 1)
    layer_size = [n_feat] + hidden + [n_class]
    act = TF.relu
    if args.model == "gcn":
        return GCN(layer_size, act, use_pp=args.use_pp, dropout=args.dropout,
                   norm=args.norm, n_linear=args.n_linear, train_size=train_size)
    if args.model == "graphsage":
        return GraphSAGE(layer_size, act, use_pp=args.use_pp, dropout=args.dropout,
                         norm=args.norm, n_linear=args.n_linear,
                         train_size=train_size)
    if args.model == "gat":
        return GAT(layer_size, act, use_pp=True, heads=args.heads,
                   dropout=args.dropout, norm=args.norm,
                   n_linear=args.n_linear, train_size=train_size)
    raise ValueError(f"unknown model {args.model}")
