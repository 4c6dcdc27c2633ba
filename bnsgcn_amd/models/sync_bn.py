"""Distributed SyncBatchNorm (reference: module/sync_bn.py).

Forward all-reduces per-feature Σx and Σx² and normalizes with the GLOBAL
mean/var over `whole_size` rows (= global train-node count — correct when
partition row counts sum to whole_size, i.e. inductive mode; SURVEY.md
§2.5.8). Backward all-reduces dbias/dweight and forms dx analytically.
Selected by --norm batch.
"""
from __future__ import annotations

import torch
import torch.distributed as dist
from torch import nn
from torch.autograd import Function


def _maybe_all_reduce(t: torch.Tensor) -> torch.Tensor:
    if dist.is_initialized() and dist.get_world_size() > 1:
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return t


class _SyncBNFunc(Function):
    @staticmethod
    def forward(ctx, x, weight, bias, whole_size, running_mean, running_var,
                training, momentum, eps):
        if not training:
            mean, var = running_mean, running_var
        else:
            from ..ops._ext import use_hip, get_ext
            if use_hip(x):
                sums = get_ext().syncbn_stats(x.contiguous())
            else:
                sums = torch.stack((x.sum(0), (x * x).sum(0)))
            # The global row count rides along in the same all-reduce. The
            # reference divides by the CONSTRUCTION-TIME whole_size
            # (= n_train), which is the true row count only in inductive
            # mode — transductively its variance goes negative -> NaN
            # (SURVEY.md §2.5.8). Counting rows dynamically is identical
            # inductively and correct transductively.
            stats = torch.cat([sums.flatten(),
                               sums.new_tensor([float(x.shape[0])])])
            _maybe_all_reduce(stats)
            nf = x.shape[1]
            n_total = stats[-1].clamp_min(1.0)
            sum_x = stats[:nf]
            sum_x2 = stats[nf:2 * nf]
            mean = sum_x / n_total
            var = (sum_x2 - mean * sum_x) / n_total
            with torch.no_grad():
                running_mean.mul_(1 - momentum).add_(mean * momentum)
                running_var.mul_(1 - momentum).add_(var * momentum)
        std = torch.sqrt(var.clamp_min(0.0) + eps) if training \
            else torch.sqrt(var + eps)
        x_hat = (x - mean) / std
        if training:
            ctx.save_for_backward(x_hat, weight, std)
            ctx.whole_size = float(n_total)
        return x_hat * weight + bias

    @staticmethod
    def backward(ctx, grad):
        x_hat, weight, std = ctx.saved_tensors
        n = ctx.whole_size
        red = torch.stack((grad.sum(0), (grad * x_hat).sum(0)))
        _maybe_all_reduce(red)
        dbias, dweight = red[0], red[1]
        # standard BN input gradient: the batch couples every row through
        # mean/var, so dx = w/std * (g - mean_rows(g) - x_hat*mean_rows(g·x_hat))
        g_mean = dbias / n
        gx_mean = dweight / n
        dx = weight / std * (grad - g_mean - x_hat * gx_mean)
        return dx, dweight, dbias, None, None, None, None, None, None


class SyncBatchNorm(nn.Module):
    def __init__(self, num_features, whole_size, eps=1e-5, momentum=0.1):
        super().__init__()
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.whole_size = whole_size
        self.eps = eps
        self.momentum = momentum
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))

    def forward(self, x):
        return _SyncBNFunc.apply(x, self.weight, self.bias, self.whole_size,
                                 self.running_mean, self.running_var,
                                 self.training, self.momentum, self.eps)
