"""Autograd-wired compute ops with CPU(torch-reference) / GPU(HIP) dispatch.

Each op dispatches per-tensor-device: CPU → ops/reference.py; CUDA(HIP) →
the gfx950 extension (required — no silent eager fallback on GPU, see
ops/_ext.py). Backward passes reuse the same primitives on precomputed
transposed CSRs, so one kernel family serves both directions
(reference counterparts: SURVEY.md §2.3 K1-K6, K13-K14).
"""
from __future__ import annotations

import torch
from torch.autograd import Function

from . import reference as ref
from ._ext import get_ext, use_hip


# ---------------------------------------------------------------- raw ops

def _worklist_of(indptr):
    """Edge-balanced work list for the device SpMM (csr_torch.build_worklist),
    cached on the indptr tensor itself (same lifetime as the CSR)."""
    wl = getattr(indptr, "_bns_worklist", None)
    if wl is None:
        from .csr_torch import build_worklist
        wl = build_worklist(indptr)
        indptr._bns_worklist = wl
    return wl


def spmm_sum_raw(indptr, indices, x, src_scale=None, dst_scale=None, out=None):
    if use_hip(x):
        wrow, wbeg, wend, wave_start, zero_rows = _worklist_of(indptr)
        acc = out is not None
        if not acc:
            # allocate uninitialized and zero ONLY the rows the kernel does
            # not fully overwrite (split/atomic + empty rows)
            out = torch.empty(indptr.numel() - 1, x.shape[1],
                              dtype=x.dtype, device=x.device)
            if zero_rows.numel():
                out.index_fill_(0, zero_rows, 0.0)
        return get_ext().spmm_sum(wrow, wbeg, wend, wave_start, indices, x,
                                  src_scale, dst_scale, out, acc)
    return ref.spmm_sum(indptr, indices, x, src_scale, dst_scale, out)


def spmm_edge_raw(indptr, indices, eweight, x, out=None, wperm=None):
    """wperm: optional edge permutation applied to eweight INSIDE the HIP
    kernel (fuses the w[eperm] gather of the transposed backward pass)."""
    if use_hip(x):
        wrow, wbeg, wend, wstart, zero_rows = _worklist_of(indptr)
        acc = out is not None
        if not acc:
            out = torch.empty(indptr.numel() - 1, x.shape[1], x.shape[2],
                              dtype=x.dtype, device=x.device)
            if zero_rows.numel():
                out.index_fill_(0, zero_rows, 0.0)
        return get_ext().spmm_edge_sum(wrow, wbeg, wend, wstart, indptr,
                                       indices, eweight.contiguous(), wperm,
                                       x.contiguous(), out, acc)
    ew = eweight if wperm is None else eweight[wperm]
    return ref.spmm_edge_sum(indptr, indices, ew, x, out)


def sddmm_dot_raw(indptr, indices, a_dst, b_src):
    if use_hip(a_dst):
        wl = _worklist_of(indptr)[:4]
        return get_ext().sddmm_dot(*wl, indptr, indices,
                                   a_dst.contiguous(), b_src.contiguous())
    return ref.sddmm_dot(indptr, indices, a_dst, b_src)


def sddmm_add_raw(indptr, indices, el_src, er_dst, slope=-1.0):
    """slope >= 0 fuses LeakyReLU into the kernel (HIP path); the torch
    path applies it separately."""
    if use_hip(el_src):
        wl = _worklist_of(indptr)[:4]
        return get_ext().sddmm_add(*wl, indptr, indices,
                                   el_src.contiguous(), er_dst.contiguous(),
                                   slope)
    out = ref.sddmm_add(indptr, indices, el_src, er_dst)
    if slope >= 0:
        out = torch.nn.functional.leaky_relu(out, slope)
    return out


def segment_softmax_raw(indptr, logits):
    if use_hip(logits):
        return get_ext().segment_softmax(indptr, logits)
    return ref.segment_softmax(indptr, logits)


def segment_softmax_bwd_raw(indptr, alpha, grad_alpha):
    if use_hip(alpha):
        return get_ext().segment_softmax_backward(indptr, alpha, grad_alpha)
    return ref.segment_softmax_backward(indptr, alpha, grad_alpha)


def pack_rows_raw(x, idx, scale=None):
    if use_hip(x):
        return get_ext().pack_rows(x, idx, scale)
    return ref.pack_rows(x, idx, scale)


def scatter_add_rows_raw(out, idx, src, scale=None):
    if use_hip(out):
        get_ext().scatter_add_rows(out, idx, src, scale)
        return out
    return ref.scatter_add_rows(out, idx, src, scale)


# ----------------------------------------------------------- autograd ops

class _SpMMSum(Function):
    """y[r] = dst_scale[r] * Σ_{c∈row r} src_scale[c] * x[c]  (K1/K2 + K18).

    Backward runs the SAME kernel on the precomputed transposed CSR with the
    scale roles swapped — no atomics anywhere (SURVEY.md §7 'per-epoch block
    assembly' design)."""

    @staticmethod
    def forward(ctx, x, indptr, indices, indptr_t, indices_t, src_scale, dst_scale):
        ctx.save_for_backward(indptr_t, indices_t, src_scale, dst_scale)
        return spmm_sum_raw(indptr, indices, x, src_scale, dst_scale)

    @staticmethod
    def backward(ctx, grad):
        indptr_t, indices_t, src_scale, dst_scale = ctx.saved_tensors
        gx = spmm_sum_raw(indptr_t, indices_t, grad.contiguous(),
                          src_scale=dst_scale, dst_scale=src_scale)
        return gx, None, None, None, None, None, None


def spmm_sum(x, indptr, indices, indptr_t, indices_t,
             src_scale=None, dst_scale=None):
    return _SpMMSum.apply(x, indptr, indices, indptr_t, indices_t,
                          src_scale, dst_scale)


class _SpMMEdge(Function):
    """y[r,h] = Σ_{e∈row r} w[e,h] * x[col_e,h,:]  (GAT aggregate, K4/K5)."""

    @staticmethod
    def forward(ctx, x, w, indptr, indices, indptr_t, indices_t, eperm_t):
        ctx.save_for_backward(x, w, indptr, indices, indptr_t, indices_t, eperm_t)
        return spmm_edge_raw(indptr, indices, w, x)

    @staticmethod
    def backward(ctx, grad):
        x, w, indptr, indices, indptr_t, indices_t, eperm_t = ctx.saved_tensors
        grad = grad.contiguous()
        gx = gw = None
        if ctx.needs_input_grad[0]:
            gx = spmm_edge_raw(indptr_t, indices_t, w, grad, wperm=eperm_t)
        if ctx.needs_input_grad[1]:
            gw = sddmm_dot_raw(indptr, indices, grad, x)
        return gx, gw, None, None, None, None, None


def spmm_edge_sum(x, w, indptr, indices, indptr_t, indices_t, eperm_t):
    return _SpMMEdge.apply(x, w, indptr, indices, indptr_t, indices_t, eperm_t)


class _SpMMEdge2(Function):
    """Inner + halo GAT aggregation accumulated into ONE output buffer —
    removes the [N, H, D]-sized elementwise add between the two edge-set
    results (~5% of the Yelp GAT epoch in bare CUDAFunctor_add kernels,
    profiles/topk_gat2_r02.txt)."""

    @staticmethod
    def forward(ctx, x_in, w_in, x_h, w_h,
                ip, ix, tip, tix, eperm_t,
                hip_, hix_, hbip, hbix, heperm_t):
        ctx.save_for_backward(x_in, w_in, x_h, w_h, ip, ix, tip, tix,
                              eperm_t, hip_, hix_, hbip, hbix, heperm_t)
        out = spmm_edge_raw(ip, ix, w_in, x_in)
        spmm_edge_raw(hip_, hix_, w_h, x_h, out=out)
        return out

    @staticmethod
    def backward(ctx, grad):
        (x_in, w_in, x_h, w_h, ip, ix, tip, tix, eperm_t,
         hip_, hix_, hbip, hbix, heperm_t) = ctx.saved_tensors
        grad = grad.contiguous()
        gx_in = gw_in = gx_h = gw_h = None
        if ctx.needs_input_grad[0]:
            gx_in = spmm_edge_raw(tip, tix, w_in, grad, wperm=eperm_t)
        if ctx.needs_input_grad[1]:
            gw_in = sddmm_dot_raw(ip, ix, grad, x_in)
        if ctx.needs_input_grad[2]:
            gx_h = spmm_edge_raw(hbip, hbix, w_h, grad, wperm=heperm_t)
        if ctx.needs_input_grad[3]:
            gw_h = sddmm_dot_raw(hip_, hix_, grad, x_h)
        return (gx_in, gw_in, gx_h, gw_h) + (None,) * 10


def spmm_edge_sum2(x_in, w_in, x_h, w_h, inner_csrs5, halo_csrs5):
    """inner_csrs5 = (ip, ix, tip, tix, eperm_t); halo_csrs5 likewise."""
    return _SpMMEdge2.apply(x_in, w_in, x_h, w_h,
                            *inner_csrs5, *halo_csrs5)


class _SDDMMAdd(Function):
    """logits[e,h] = [leaky_relu](el[col_e,h] + er[row_e,h])  (u_add_v
    SDDMM, optionally fused with the GAT LeakyReLU — the activation is
    monotonic with fixpoint 0, so the backward mask comes from the sign
    of the saved OUTPUT)."""

    @staticmethod
    def forward(ctx, el, er, indptr, indices, indptr_t, indices_t, eperm_t,
                slope):
        out = sddmm_add_raw(indptr, indices, el, er,
                            -1.0 if slope is None else float(slope))
        ctx.slope = slope
        if slope is not None:
            ctx.save_for_backward(indptr, indices, indptr_t, indices_t,
                                  eperm_t, out)
        else:
            ctx.save_for_backward(indptr, indices, indptr_t, indices_t,
                                  eperm_t)
        ctx.n_src, ctx.n_dst = el.shape[0], er.shape[0]
        return out

    @staticmethod
    def backward(ctx, grad):
        if ctx.slope is not None:
            indptr, indices, indptr_t, indices_t, eperm_t, out = ctx.saved_tensors
            grad = torch.where(out > 0, grad, grad * ctx.slope).contiguous()
        else:
            indptr, indices, indptr_t, indices_t, eperm_t = ctx.saved_tensors
            grad = grad.contiguous()
        g_el = g_er = None
        if use_hip(grad) and grad.shape[1] <= 8:
            e = get_ext()
            if ctx.needs_input_grad[0]:
                # d el[c] = Σ_{e: col_e=c} grad[e]: segment-sum over the
                # transposed CSR with the edge permutation
                g_el = e.segment_sum_edges(*_worklist_of(indptr_t)[:4], eperm_t,
                                           grad, ctx.n_src)
            if ctx.needs_input_grad[1]:
                g_er = e.segment_sum_edges(*_worklist_of(indptr)[:4], None,
                                           grad, ctx.n_dst)
            return g_el, g_er, None, None, None, None, None, None
        if ctx.needs_input_grad[0]:
            g_el = torch.zeros(ctx.n_src, grad.shape[1], dtype=grad.dtype,
                               device=grad.device)
            g_el.index_add_(0, indices.long(), grad)
        if ctx.needs_input_grad[1]:
            g_er = torch.zeros(ctx.n_dst, grad.shape[1], dtype=grad.dtype,
                               device=grad.device)
            row = torch.repeat_interleave(
                torch.arange(ctx.n_dst, device=grad.device),
                indptr[1:] - indptr[:-1])
            g_er.index_add_(0, row, grad)
        return g_el, g_er, None, None, None, None, None, None


def sddmm_add(el, er, indptr, indices, indptr_t, indices_t, eperm_t,
              slope=None):
    return _SDDMMAdd.apply(el, er, indptr, indices, indptr_t, indices_t,
                           eperm_t, slope)


class _SegmentSoftmax(Function):
    @staticmethod
    def forward(ctx, logits, indptr):
        alpha = segment_softmax_raw(indptr, logits)
        ctx.save_for_backward(alpha, indptr)
        return alpha

    @staticmethod
    def backward(ctx, grad):
        alpha, indptr = ctx.saved_tensors
        return segment_softmax_bwd_raw(indptr, alpha, grad.contiguous()), None


def segment_softmax(logits, indptr):
    return _SegmentSoftmax.apply(logits, indptr)


import os as _os

# GEMM dispatch (K6). Per-shape A/B on MI355X (profiles/gemm_micro_r02):
# rocBLAS wins the NT forward and NN dx by 1.5-2x on the tall-skinny
# bench shapes; OUR split-K MFMA gemm_tn wins the dW reduction by up to
# 3x (rocBLAS is weak at K >> M=N). "auto" routes each direction to the
# winner; BNSGCN_GEMM=hip forces the hand-written kernels everywhere.
_GEMM_MODE = _os.environ.get("BNSGCN_GEMM", "auto")


class _Linear(Function):
    """XW^T + b; see _GEMM_MODE for the MFMA/rocBLAS routing."""

    @staticmethod
    def forward(ctx, x, weight, bias):
        ctx.save_for_backward(x, weight)
        ctx.has_bias = bias is not None
        if use_hip(x) and _GEMM_MODE == "hip":
            return get_ext().gemm_nt_bias(x, weight, bias)
        return torch.nn.functional.linear(x, weight, bias)

    @staticmethod
    def backward(ctx, grad):
        x, weight = ctx.saved_tensors
        grad = grad.contiguous()
        gx = gw = gb = None
        if use_hip(x):
            e = get_ext()
            if ctx.needs_input_grad[0]:
                gx = e.gemm_nn(grad, weight) if _GEMM_MODE == "hip" \
                    else grad @ weight
            if ctx.needs_input_grad[1]:
                gw = e.gemm_tn(grad, x)     # split-K: ours wins
            if ctx.has_bias and ctx.needs_input_grad[2]:
                return gx, gw, e.syncbn_stats(grad)[0]
        else:
            if ctx.needs_input_grad[0]:
                gx = grad @ weight
            if ctx.needs_input_grad[1]:
                gw = grad.t() @ x
        if ctx.has_bias and ctx.needs_input_grad[2]:
            gb = grad.sum(0)
        return gx, gw, gb


def linear(x, weight, bias=None):
    return _Linear.apply(x, weight, bias)


class _LayerNormF(Function):
    """Row LayerNorm (K7) on the fused gfx950 kernels — torch's ROCm LN
    kernels measured ~27% of the ogbn-products epoch (profiles/
    topk_products_r02.txt). act=True additionally fuses the inter-layer
    ReLU (K9): forward emits relu(ln(x)); backward regenerates the
    activation mask from the saved LN state — no mask, no ReLU kernels."""

    @staticmethod
    def forward(ctx, x, weight, bias, eps, act):
        x = x.contiguous()
        y, mean, rstd = get_ext().ln_fwd(x, weight, bias, eps, int(act))
        ctx.save_for_backward(x, weight, bias, mean, rstd)
        ctx.act = int(act)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, b, mean, rstd = ctx.saved_tensors
        dx, dw, db = get_ext().ln_bwd(x, dy.contiguous(), w, b, mean, rstd,
                                      ctx.act)
        return dx, dw, db, None, None


def layer_norm(x, weight, bias, eps: float = 1e-5, act: bool = False):
    """act=True computes relu(layer_norm(x)) in one fused pair on GPU."""
    if use_hip(x) and x.shape[-1] <= 1024 and x.dtype == torch.float32:
        return _LayerNormF.apply(x, weight, bias, eps, act)
    y = torch.nn.functional.layer_norm(x, (x.shape[-1],), weight, bias, eps)
    return torch.relu(y) if act else y


class _DropoutF(Function):
    """Mask-free dropout (K8): the splitmix64 mask is regenerated from the
    saved seed in backward — one elementwise kernel each way, no mask
    tensor, no torch RNG-state kernels."""

    @staticmethod
    def forward(ctx, x, keep, seed):
        ctx.keep, ctx.seed = keep, seed
        return get_ext().dropout_apply(x.contiguous(), keep, seed)

    @staticmethod
    def backward(ctx, dy):
        return (get_ext().dropout_apply(dy.contiguous(), ctx.keep, ctx.seed),
                None, None)


def dropout(x, p: float, training: bool):
    if not training or p <= 0.0:
        return x
    if use_hip(x) and x.dtype == torch.float32:
        seed = int(torch.randint(0, 2**62, (1,)).item())
        return _DropoutF.apply(x, 1.0 - p, seed)
    return torch.nn.functional.dropout(x, p, training)


def segment_softmax2_raw(indptr1, logits1, indptr2, logits2):
    if use_hip(logits1):
        return get_ext().segment_softmax2(indptr1, logits1, indptr2, logits2,
                                          1.0, 0)[:2]
    return ref.segment_softmax2(indptr1, logits1, indptr2, logits2)


def segment_softmax2_bwd_raw(indptr1, a1, g1, indptr2, a2, g2):
    if use_hip(a1):
        return get_ext().segment_softmax2_backward(indptr1, a1, g1,
                                                   indptr2, a2, g2, 1.0, 0)
    return ref.segment_softmax2_backward(indptr1, a1, g1, indptr2, a2, g2)


class _SegmentSoftmax2(Function):
    """Union softmax over two per-row edge sets (split GAT block), with
    the GAT attention dropout FUSED on the HIP path (p_drop > 0): the
    forward emits the dropped weights alongside the saved pre-drop
    softmax, and the backward regenerates the Philox mask from the saved
    seed — no mask tensor, no separate dropout kernels (reference: DGL
    GATConv attn_drop)."""

    @staticmethod
    def forward(ctx, logits1, logits2, indptr1, indptr2, p_drop):
        keep = 1.0 - float(p_drop or 0.0)
        if use_hip(logits1):
            seed = int(torch.randint(0, 2**62, (1,)).item()) if keep < 1.0 \
                else 0
            a1, a2, da1, da2 = get_ext().segment_softmax2(
                indptr1, logits1, indptr2, logits2, keep, seed)
        else:
            a1, a2 = ref.segment_softmax2(indptr1, logits1, indptr2, logits2)
            seed = 0
            if keep < 1.0:
                m1 = (torch.rand_like(a1) < keep).float()
                m2 = (torch.rand_like(a2) < keep).float()
                da1, da2 = a1 * m1 / keep, a2 * m2 / keep
                ctx.cpu_masks = (m1, m2)
            else:
                da1, da2 = a1, a2
        ctx.keep, ctx.seed = keep, seed
        ctx.save_for_backward(a1, a2, indptr1, indptr2)
        return da1, da2

    @staticmethod
    def backward(ctx, g1, g2):
        a1, a2, indptr1, indptr2 = ctx.saved_tensors
        g1, g2 = g1.contiguous(), g2.contiguous()
        if use_hip(a1):
            d1, d2 = get_ext().segment_softmax2_backward(
                indptr1, a1, g1, indptr2, a2, g2, ctx.keep, ctx.seed)
        else:
            if ctx.keep < 1.0:
                m1, m2 = ctx.cpu_masks
                g1, g2 = g1 * m1 / ctx.keep, g2 * m2 / ctx.keep
            d1, d2 = ref.segment_softmax2_backward(indptr1, a1, g1,
                                                   indptr2, a2, g2)
        return d1, d2, None, None, None


def segment_softmax2(logits1, logits2, indptr1, indptr2, p_drop=0.0):
    return _SegmentSoftmax2.apply(logits1, logits2, indptr1, indptr2, p_drop)


class _AttnProject(Function):
    """Fused GAT attention projections (el, er) = (<z,a_l>, <z,a_r>) per
    (node, head) — one read of z instead of torch's broadcast-mul +
    reduce chains in each direction."""

    @staticmethod
    def forward(ctx, z, al, ar):
        ctx.save_for_backward(z, al, ar)
        if use_hip(z) and z.shape[1] * z.shape[2] % 4 == 0 and z.shape[1] <= 8:
            el, er = get_ext().attn_project(z.contiguous(), al.contiguous(),
                                            ar.contiguous())
            return el, er
        return ref.attn_project(z, al, ar)

    @staticmethod
    def backward(ctx, g_el, g_er):
        z, al, ar = ctx.saved_tensors
        g_el = g_el.contiguous()
        g_er = g_er.contiguous()
        if use_hip(z) and z.shape[1] * z.shape[2] % 4 == 0 and z.shape[1] <= 8:
            dz, dal, dar = get_ext().attn_project_backward(
                z.contiguous(), al.contiguous(), ar.contiguous(), g_el, g_er)
            return dz, dal, dar
        return ref.attn_project_backward(z, al, ar, g_el, g_er)


def attn_project(z, al, ar):
    return _AttnProject.apply(z, al, ar)
