"""Pure-torch reference implementations of every compute op.

These are (a) the CPU execution path of the framework and (b) the numerics
oracle that the HIP/CDNA4 kernels are tested against (fp32, same math).
Raw computations only — autograd wiring lives in ops/functional.py.

Reference-kernel parity: spmm_sum=K1/K2/K3 (+fused K18
norms), spmm_edge/sddmm/segment_softmax=K4/K5, pack_rows=K13,
scatter_add_rows=K14, syncbn stats=K12.
"""
from __future__ import annotations

import torch


def _row_of_edge(indptr: torch.Tensor) -> torch.Tensor:
    n = indptr.numel() - 1
    return torch.repeat_interleave(torch.arange(n, device=indptr.device),
                                   indptr[1:] - indptr[:-1])


def spmm_sum(indptr: torch.Tensor, indices: torch.Tensor, x: torch.Tensor,
             src_scale: torch.Tensor | None = None,
             dst_scale: torch.Tensor | None = None,
             out: torch.Tensor | None = None) -> torch.Tensor:
    """out[r] (+)= dst_scale[r] * sum_{c in row r} src_scale[c] * x[c].

    x: [n_src, F]; returns [n_rows, F]. If `out` is given, accumulates into
    it (used to combine inner + halo contributions into one output).
    """
    n_rows = indptr.numel() - 1
    xs = x if src_scale is None else x * src_scale.unsqueeze(1)
    if indices.numel() == 0:
        agg = torch.zeros(n_rows, x.shape[1], dtype=x.dtype, device=x.device)
    else:
        a = torch.sparse_csr_tensor(
            indptr.to(torch.int64), indices.to(torch.int64),
            torch.ones(indices.numel(), dtype=x.dtype, device=x.device),
            size=(n_rows, x.shape[0]))
        agg = a @ xs
    if dst_scale is not None:
        agg = agg * dst_scale.unsqueeze(1)
    if out is not None:
        out += agg
        return out
    return agg


def spmm_edge_sum(indptr: torch.Tensor, indices: torch.Tensor,
                  eweight: torch.Tensor, x: torch.Tensor,
                  out: torch.Tensor | None = None) -> torch.Tensor:
    """Multi-head edge-weighted aggregation (GAT message reduce, K4):
    out[r, h] (+)= sum_{e in row r} eweight[e, h] * x[col_e, h, :].

    eweight: [E, H]; x: [n_src, H, D]; returns [n_rows, H, D].
    """
    n_rows = indptr.numel() - 1
    H, D = x.shape[1], x.shape[2]
    agg = torch.zeros(n_rows, H, D, dtype=x.dtype, device=x.device)
    if indices.numel():
        row = _row_of_edge(indptr)
        msg = x[indices.long()] * eweight.unsqueeze(-1)   # [E, H, D]
        agg.index_add_(0, row, msg)
    if out is not None:
        out += agg
        return out
    return agg


def sddmm_dot(indptr: torch.Tensor, indices: torch.Tensor,
              a_dst: torch.Tensor, b_src: torch.Tensor) -> torch.Tensor:
    """Per-edge per-head dot product (grad of spmm_edge_sum wrt eweight):
    out[e, h] = <a_dst[row_e, h, :], b_src[col_e, h, :]>."""
    row = _row_of_edge(indptr)
    return (a_dst[row] * b_src[indices.long()]).sum(-1)


def sddmm_add(indptr: torch.Tensor, indices: torch.Tensor,
              el_src: torch.Tensor, er_dst: torch.Tensor) -> torch.Tensor:
    """GAT attention logits (u_add_v SDDMM): out[e,h] = el_src[col_e,h] + er_dst[row_e,h]."""
    row = _row_of_edge(indptr)
    return el_src[indices.long()] + er_dst[row]


def segment_softmax(indptr: torch.Tensor, logits: torch.Tensor) -> torch.Tensor:
    """Numerically stable softmax over each CSR row's edges.
    logits: [E, H] grouped by row in CSR order. Returns [E, H]."""
    n_rows = indptr.numel() - 1
    row = _row_of_edge(indptr)
    H = logits.shape[1]
    m = torch.full((n_rows, H), float("-inf"), dtype=logits.dtype, device=logits.device)
    m = m.index_reduce_(0, row, logits, "amax", include_self=True)
    ex = torch.exp(logits - m[row])
    s = torch.zeros(n_rows, H, dtype=logits.dtype, device=logits.device)
    s.index_add_(0, row, ex)
    return ex / s.clamp_min(1e-38)[row]


def segment_softmax_backward(indptr: torch.Tensor, alpha: torch.Tensor,
                             grad_alpha: torch.Tensor) -> torch.Tensor:
    """d logits given alpha = segment_softmax(logits) and d alpha."""
    n_rows = indptr.numel() - 1
    row = _row_of_edge(indptr)
    ag = alpha * grad_alpha
    s = torch.zeros(n_rows, alpha.shape[1], dtype=alpha.dtype, device=alpha.device)
    s.index_add_(0, row, ag)
    return ag - alpha * s[row]


def pack_rows(x: torch.Tensor, idx: torch.Tensor,
              scale: torch.Tensor | None = None) -> torch.Tensor:
    """Gather rows + optional per-row scale (the send-pack, K13;
    reference feature_buffer.py:117 `feat[selected] / ratio`)."""
    out = x[idx.long()]
    if scale is not None:
        out = out * scale.unsqueeze(1)
    return out


def scatter_add_rows(out: torch.Tensor, idx: torch.Tensor, src: torch.Tensor,
                     scale: torch.Tensor | None = None) -> torch.Tensor:
    """out[idx[r]] += scale[r] * src[r] (the grad unpack, K14;
    reference feature_buffer.py:129 `grad[selected] += recv / ratio`)."""
    s = src if scale is None else src * scale.unsqueeze(1)
    out.index_add_(0, idx.long(), s)
    return out


def segment_softmax2(indptr1, logits1, indptr2, logits2):
    """Softmax over the UNION of two per-row edge segments (the split GAT
    block: static inner edges + per-epoch sampled halo edges), numerically
    stable via the shared running max. Returns (alpha1, alpha2)."""
    n_rows = indptr1.numel() - 1
    H = logits1.shape[1]
    dev, dt = logits1.device, logits1.dtype
    row1 = _row_of_edge(indptr1)
    row2 = _row_of_edge(indptr2)
    m = torch.full((n_rows, H), float("-inf"), dtype=dt, device=dev)
    if logits1.numel():
        m = m.index_reduce_(0, row1, logits1, "amax", include_self=True)
    if logits2.numel():
        m = m.index_reduce_(0, row2, logits2, "amax", include_self=True)
    m = torch.where(torch.isfinite(m), m, torch.zeros_like(m))
    ex1 = torch.exp(logits1 - m[row1]) if logits1.numel() else logits1
    ex2 = torch.exp(logits2 - m[row2]) if logits2.numel() else logits2
    s = torch.zeros(n_rows, H, dtype=dt, device=dev)
    if logits1.numel():
        s.index_add_(0, row1, ex1)
    if logits2.numel():
        s.index_add_(0, row2, ex2)
    s = s.clamp_min(1e-38)
    a1 = ex1 / s[row1] if logits1.numel() else ex1
    a2 = ex2 / s[row2] if logits2.numel() else ex2
    return a1, a2


def segment_softmax2_backward(indptr1, alpha1, grad1, indptr2, alpha2, grad2):
    """d logits for the union softmax: dl_e = a_e * (g_e - S[row_e]) with
    S[r] = sum over BOTH segments of a*g."""
    n_rows = indptr1.numel() - 1
    H = alpha1.shape[1]
    dev, dt = alpha1.device, alpha1.dtype
    row1 = _row_of_edge(indptr1)
    row2 = _row_of_edge(indptr2)
    s = torch.zeros(n_rows, H, dtype=dt, device=dev)
    if alpha1.numel():
        s.index_add_(0, row1, alpha1 * grad1)
    if alpha2.numel():
        s.index_add_(0, row2, alpha2 * grad2)
    d1 = alpha1 * (grad1 - s[row1]) if alpha1.numel() else alpha1
    d2 = alpha2 * (grad2 - s[row2]) if alpha2.numel() else alpha2
    return d1, d2


def attn_project(z, al, ar):
    """GAT attention projections in one conceptual pass:
    el[n,h] = <z[n,h,:], al[h,:]>, er[n,h] = <z[n,h,:], ar[h,:]>.
    z: [N,H,D]; al/ar: [1,H,D] (the GATConv attn_l/attn_r parameters)."""
    el = (z * al).sum(-1)
    er = (z * ar).sum(-1)
    return el, er


def attn_project_backward(z, al, ar, g_el, g_er):
    dz = al * g_el.unsqueeze(-1) + ar * g_er.unsqueeze(-1)
    dal = (z * g_el.unsqueeze(-1)).sum(0, keepdim=True)
    dar = (z * g_er.unsqueeze(-1)).sum(0, keepdim=True)
    return dz, dal, dar
