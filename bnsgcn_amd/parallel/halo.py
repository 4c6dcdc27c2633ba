"""Halo exchange + partition-parallel aggregation (the hot path).

Two autograd Functions:

* `PartitionAggregate` — the GCN/SAGE aggregation, FUSED with the halo
  exchange so the RCCL all-to-all overlaps the inner-edge SpMM on both the
  forward and backward passes. The reference's exchange is blocking inside
  forward/backward (helper/feature_buffer.py:93-99,169-182 — its Comm(s)
  is ≈63% of epoch time, README.md:94-95); here the exchange runs on a
  side HIP stream while the inner CSR SpMM proceeds on the main stream,
  and the sampled-halo SpMM accumulates into the same output when the
  collective lands. No [own ‖ peers] concat is ever materialized
  (reference K15 eliminated).

* `HaloExchange` — plain exchange returning packed received rows (used by
  the GAT path, which needs raw source features for attention, and by the
  use_pp precompute pass).

Semantics preserved from the reference: pack scale 1/ratio at send,
1/ratio again at grad scatter (feature_buffer.py:117,129), deterministic
int(p·n) sizes both sides.
"""
from __future__ import annotations

import os

import torch
from torch.autograd import Function

from ..ops.functional import (pack_rows_raw, scatter_add_rows_raw, spmm_sum_raw)
from .comm import all_to_all_rows
from .plan import EpochState, HaloPlan
from ..utils.timer import comm_timer

_comm_stream: torch.cuda.Stream | None = None


def comm_stream() -> torch.cuda.Stream:
    global _comm_stream
    if _comm_stream is None:
        _comm_stream = torch.cuda.Stream()
    return _comm_stream


def _exchange(send: torch.Tensor, recv_counts, send_counts,
              wire_dtype=None) -> torch.Tensor:
    """wire_dtype=torch.bfloat16 halves the bytes on the xGMI links
    (--halo-dtype bf16); compute stays fp32 — the cast happens at the
    wire only (send downcast, recv upcast)."""
    wire = send if wire_dtype is None else send.to(wire_dtype)
    recv = torch.empty(sum(recv_counts), send.shape[1],
                       dtype=wire.dtype, device=send.device)
    all_to_all_rows(recv, wire, recv_counts, send_counts)
    return recv if wire_dtype is None else recv.float()


class _PartitionAggregate(Function):
    """out[v] = dst_scale[v] * ( Σ_{u∈N_in(v)∩inner} src_scale[u]·x[u]
                               + Σ_{r∈sampled halo rows→v} hscale[r]·recv[r] )

    recv = all-to-all of pack(x)·(1/ratio). hscale is the halo-side src
    scale (GCN out-norm of the owning nodes; None for SAGE)."""

    @staticmethod
    def forward(ctx, x, plan: HaloPlan, st: EpochState,
                inner_indptr, inner_indices, inner_t_indptr, inner_t_indices,
                src_scale, dst_scale, use_halo_src_scale: bool):
        # BNSGCN_NO_OVERLAP=1 disables the side-stream overlap (debug /
        # perf A/B); the sequential branch works for GPU tensors too.
        cuda = x.is_cuda and os.environ.get("BNSGCN_NO_OVERLAP") != "1"

        hscale = st.halo_out_norm_inv if use_halo_src_scale else None
        if cuda:
            cs = comm_stream()
            ev = torch.cuda.current_stream().record_event()
            with torch.cuda.stream(cs):
                cs.wait_event(ev)
                send = pack_rows_raw(x, st.pack_idx, st.pack_scale)
                with comm_timer.span("forward", cuda=True):
                    recv = _exchange(send, st.recv_counts, st.send_counts,
                                     plan.wire_dtype)
                ev_done = cs.record_event()
            out = spmm_sum_raw(inner_indptr, inner_indices, x, src_scale, dst_scale)
            torch.cuda.current_stream().wait_event(ev_done)
            recv.record_stream(torch.cuda.current_stream())
            spmm_sum_raw(st.halo_fwd_indptr, st.halo_fwd_indices, recv,
                         src_scale=hscale, dst_scale=dst_scale, out=out)
        else:
            send = pack_rows_raw(x, st.pack_idx, st.pack_scale)
            with comm_timer.span("forward"):
                recv = _exchange(send, st.recv_counts, st.send_counts,
                                 plan.wire_dtype)
            out = spmm_sum_raw(inner_indptr, inner_indices, x, src_scale, dst_scale)
            spmm_sum_raw(st.halo_fwd_indptr, st.halo_fwd_indices, recv,
                         src_scale=hscale, dst_scale=dst_scale, out=out)
        ctx.plan, ctx.st, ctx.hscale = plan, st, hscale
        ctx.save_for_backward(inner_t_indptr, inner_t_indices, src_scale, dst_scale)
        return out

    @staticmethod
    def backward(ctx, g):
        st: EpochState = ctx.st
        inner_t_indptr, inner_t_indices, src_scale, dst_scale = ctx.saved_tensors
        hscale = ctx.hscale
        g = g.contiguous()
        cuda = g.is_cuda and os.environ.get("BNSGCN_NO_OVERLAP") != "1"
        if cuda:
            cs = comm_stream()
            ev = torch.cuda.current_stream().record_event()
            with torch.cuda.stream(cs):
                cs.wait_event(ev)
                # d recv[r] = hscale[r] * Σ_{v∈row r} dst_scale[v]·g[v]
                gr = spmm_sum_raw(st.halo_bwd_indptr, st.halo_bwd_indices, g,
                                  src_scale=dst_scale, dst_scale=hscale)
                with comm_timer.span("backward", cuda=True):
                    back = _exchange(gr, st.send_counts, st.recv_counts,
                                     ctx.plan.wire_dtype)
                ev_done = cs.record_event()
            gx = spmm_sum_raw(inner_t_indptr, inner_t_indices, g,
                              src_scale=dst_scale, dst_scale=src_scale)
            torch.cuda.current_stream().wait_event(ev_done)
            back.record_stream(torch.cuda.current_stream())
            scatter_add_rows_raw(gx, st.pack_idx, back, st.pack_scale)
        else:
            gr = spmm_sum_raw(st.halo_bwd_indptr, st.halo_bwd_indices, g,
                              src_scale=dst_scale, dst_scale=hscale)
            with comm_timer.span("backward"):
                back = _exchange(gr, st.send_counts, st.recv_counts,
                                 ctx.plan.wire_dtype)
            gx = spmm_sum_raw(inner_t_indptr, inner_t_indices, g,
                              src_scale=dst_scale, dst_scale=src_scale)
            scatter_add_rows_raw(gx, st.pack_idx, back, st.pack_scale)
        return (gx,) + (None,) * 9


def partition_aggregate(x, plan: HaloPlan, inner_csrs, src_scale, dst_scale,
                        use_halo_src_scale: bool):
    """inner_csrs = (indptr, indices, indptr_T, indices_T) on device."""
    ip, ix, tip, tix = inner_csrs
    return _PartitionAggregate.apply(x, plan, plan.state, ip, ix, tip, tix,
                                     src_scale, dst_scale, use_halo_src_scale)


class _HaloExchange(Function):
    """recv = all-to-all(pack(x)·scale); backward scatter-adds the returned
    gradient (·scale) into x's grad. Returns the packed received rows in
    peer-major sampled order (aligned with plan.state.hsel)."""

    @staticmethod
    def forward(ctx, x, plan: HaloPlan, st: EpochState):
        send = pack_rows_raw(x, st.pack_idx, st.pack_scale)
        with comm_timer.span("forward", cuda=x.is_cuda):
            recv = _exchange(send, st.recv_counts, st.send_counts,
                             plan.wire_dtype)
        ctx.st = st
        ctx.plan = plan
        ctx.n_inner = x.shape[0]
        return recv

    @staticmethod
    def backward(ctx, g):
        st: EpochState = ctx.st
        g = g.contiguous()
        with comm_timer.span("backward", cuda=g.is_cuda):
            back = _exchange(g, st.send_counts, st.recv_counts,
                             ctx.plan.wire_dtype)
        gx = torch.zeros(ctx.n_inner, g.shape[1], dtype=g.dtype, device=g.device)
        scatter_add_rows_raw(gx, st.pack_idx, back, st.pack_scale)
        return gx, None, None


def halo_exchange(x, plan: HaloPlan):
    return _HaloExchange.apply(x, plan, plan.state)
