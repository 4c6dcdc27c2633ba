"""GraphContext — what a model forward needs to know about the graph.

Two flavors:
* partition context (training): inner CSRs + HaloPlan (+ a full-halo state
  for use_pp precompute / GAT layer 0);
* full-graph context (evaluation / single-process): plain CSRs, no plan.

This replaces the reference's per-epoch DGL heterograph argument
(train.py:392-404, module/layer.py graph arg) with precomputed device
tensors; the per-epoch part lives in HaloPlan.set_epoch.
"""
from __future__ import annotations

import torch

from ..graph.store import Partition
from ..ops.csr_torch import transpose_csr, merge_csr
from ..parallel.halo import partition_aggregate
from ..parallel.plan import EpochState, HaloPlan


def _inv_sqrt(x: torch.Tensor) -> torch.Tensor:
    return 1.0 / torch.sqrt(x.clamp_min(1.0))


def _inv(x: torch.Tensor) -> torch.Tensor:
    return 1.0 / x.clamp_min(1.0)


class GraphContext:
    def __init__(self, indptr, indices, in_deg, out_deg, device,
                 plan: HaloPlan | None = None, need_eperm: bool = True):
        dev = torch.device(device)
        self.plan = plan
        self.indptr = indptr.to(dev)
        self.indices = indices.to(dev)
        tip, tix, t_eperm = transpose_csr(self.indptr, self.indices,
                                          n_cols=self.indptr.numel() - 1
                                          if plan is None else plan.n_inner)
        self.t_indptr, self.t_indices = tip, tix
        # t_eperm[j] = original edge id of the j-th transposed edge (for
        # per-edge payloads through the transpose — GAT attention weights
        # only; int64 per edge = 13.8 GB on papers100M, dropped otherwise)
        self.t_eperm = t_eperm if need_eperm else None
        # training-loss row restriction (set by the runtime): when model
        # forward passes rows=loss_rows for the FINAL layer, aggregation
        # and logits are computed only for labeled rows — identical math
        # (unlabeled logits have zero gradient; only GEMM reduction-order
        # float noise differs), and the
        # [N, n_class] logits tensor shrinks to [n_train_local, n_class]
        # (papers100M: 76 GB -> 0.8 GB)
        self.loss_rows: torch.Tensor | None = None
        self._rows_static: tuple | None = None   # gathered inner CSRs
        self._rows_halo: tuple | None = None     # per-epoch gathered halo
        in_deg = in_deg.to(dev).float()
        out_deg = out_deg.to(dev).float()
        self.in_norm_inv = _inv_sqrt(in_deg)       # GCN dst scale
        self.out_norm_inv = _inv_sqrt(out_deg)     # GCN src scale
        self.in_deg_inv = _inv(in_deg)             # SAGE mean dst scale
        self.n_rows = self.indptr.numel() - 1
        if dev.type == "cuda":
            # pre-build SpMM work lists on the default stream (avoids
            # cross-stream allocation of cached plan tensors)
            from ..ops.functional import _worklist_of
            _worklist_of(self.indptr)
            _worklist_of(self.t_indptr)
        # lazily built per-epoch GAT block (combined inner+halo CSR)
        self._gat_cache: tuple[int, tuple] | None = None
        # full-halo (p=1.0) exchange state for precompute / GAT layer 0
        self._full_state: EpochState | None = None

    # ---------------------------------------------------------------- train
    @classmethod
    def for_partition(cls, part: Partition, plan: HaloPlan, device,
                      need_eperm: bool = True) -> "GraphContext":
        return cls(torch.from_numpy(part.inner_indptr),
                   torch.from_numpy(part.inner_indices),
                   torch.from_numpy(part.in_deg),
                   torch.from_numpy(part.out_deg), device, plan=plan,
                   need_eperm=need_eperm)

    # ----------------------------------------------------------------- eval
    @classmethod
    def for_full_graph(cls, csr_indptr, csr_indices, in_deg, out_deg, device
                       ) -> "GraphContext":
        return cls(csr_indptr, csr_indices, in_deg, out_deg, device, plan=None)

    @property
    def inner_csrs(self):
        return (self.indptr, self.indices, self.t_indptr, self.t_indices)

    def aggregate(self, x: torch.Tensor, mode: str,
                  rows: torch.Tensor | None = None) -> torch.Tensor:
        """mode='gcn': symmetric-normalized sum (reference layer.py:32-38);
        mode='mean': in-degree mean with FULL-graph degrees
        (reference layer.py:85-92 — degrees precomputed before
        partitioning, utils.py:92-93, so the sampled estimator stays
        unbiased after the 1/ratio pack scale).

        rows: optional destination-row restriction (final-layer loss rows):
        output is [len(rows), F], aggregating only into those rows. The
        halo exchange itself is unchanged (senders sample independently of
        receiver labels); only the local aggregation shrinks."""
        if mode == "gcn":
            src, dst, halo_src = self.out_norm_inv, self.in_norm_inv, True
        elif mode == "mean":
            src, dst, halo_src = None, self.in_deg_inv, False
        else:
            raise ValueError(mode)
        if rows is not None:
            dst = dst[rows]
        if self.plan is None:
            from ..ops.functional import spmm_sum
            if rows is None:
                return spmm_sum(x, *self.inner_csrs, src_scale=src,
                                dst_scale=dst)
            ip, ix, tip, tix = self._rows_inner_csrs(rows)
            return spmm_sum(x, ip, ix, tip, tix, src_scale=src, dst_scale=dst)
        if rows is None:
            return partition_aggregate(x, self.plan, self.inner_csrs, src,
                                       dst, halo_src)
        import dataclasses
        st = self.plan.state
        hfip, hfix, hbip, hbix = self._rows_halo_csrs(rows, st)
        st_r = dataclasses.replace(st, halo_fwd_indptr=hfip,
                                   halo_fwd_indices=hfix,
                                   halo_bwd_indptr=hbip,
                                   halo_bwd_indices=hbix)
        ip, ix, tip, tix = self._rows_inner_csrs(rows)
        from ..parallel.halo import _PartitionAggregate
        return _PartitionAggregate.apply(x, self.plan, st_r, ip, ix, tip, tix,
                                         src, dst, halo_src)

    def _rows_inner_csrs(self, rows: torch.Tensor):
        """Row-gathered inner CSR + transpose, cached (rows are static)."""
        from ..ops.csr_torch import gather_rows_csr
        if self._rows_static is None:
            ip, ix = gather_rows_csr(self.indptr, self.indices, rows)
            n_src = self.n_rows if self.plan is None else self.plan.n_inner
            tip, tix, _ = transpose_csr(ip, ix, n_src)
            if self.indptr.is_cuda:
                from ..ops.functional import _worklist_of
                _worklist_of(ip)
                _worklist_of(tip)
            self._rows_static = (ip, ix, tip, tix)
        return self._rows_static

    def _rows_halo_csrs(self, rows: torch.Tensor, st: EpochState):
        """Row-gathered sampled-halo CSR + transpose, cached per epoch
        (optionally pre-built on a side stream by prefetch_rows_halo)."""
        if self._rows_halo is not None and self._rows_halo[0] == st.epoch:
            ev = self._rows_halo[2]
            if ev is not None:      # built on the prefetch stream
                cur = torch.cuda.current_stream()
                cur.wait_event(ev)
                for t in self._rows_halo[1]:
                    t.record_stream(cur)
                    wl = getattr(t, "_bns_worklist", None)
                    if wl is not None:
                        for w in wl:
                            w.record_stream(cur)
                self._rows_halo = (st.epoch, self._rows_halo[1], None)
            return self._rows_halo[1]
        out = self._build_rows_halo(rows, st)
        self._rows_halo = (st.epoch, out, None)
        return out

    def _build_rows_halo(self, rows: torch.Tensor, st: EpochState):
        from ..ops.csr_torch import gather_rows_csr
        hfip, hfix = gather_rows_csr(st.halo_fwd_indptr, st.halo_fwd_indices,
                                     rows)
        R = int(st.halo_bwd_indptr.numel() - 1)
        hbip, hbix, _ = transpose_csr(hfip, hfix, R)
        if hfip.is_cuda:
            from ..ops.functional import _worklist_of
            _worklist_of(hfip)
            _worklist_of(hbip)
        return (hfip, hfix, hbip, hbix)

    def prefetch_rows_halo(self, st: EpochState, stream,
                           gat: bool = False) -> None:
        """Build the restricted-halo CSRs for a FUTURE epoch state on the
        given side stream (called by RankState.prefetch alongside
        HaloPlan.prefetch). gat=True builds the GAT edge-set variant
        (with the transpose edge permutation) instead."""
        if self.loss_rows is None:
            return
        with torch.cuda.stream(stream):
            if gat:
                out = self._build_gat_rows_halo(st)
            else:
                out = self._build_rows_halo(self.loss_rows, st)
            ev = stream.record_event()
        if gat:
            self._gat_rows_halo = (st.epoch, out, ev)
        else:
            self._rows_halo = (st.epoch, out, ev)

    # ------------------------------------------------------------- GAT block
    def gat_block(self):
        """Combined (inner ∪ sampled-halo) dst-indexed CSR with source ids in
        [0, n_inner + R): cols < n_inner are inner sources, cols >= n_inner
        index the packed received rows. Rebuilt per epoch, cached."""
        if self.plan is None:
            ip, ix = self.indptr, self.indices
            tip, tix, eperm = transpose_csr(ip, ix, self.n_rows)
            return ip, ix, tip, tix, eperm, 0
        st = self.plan.state
        if self._gat_cache is not None and self._gat_cache[0] == st.epoch:
            return self._gat_cache[1]
        n_inner = self.plan.n_inner
        ip, ix = merge_csr(self.indptr, self.indices,
                           st.halo_fwd_indptr, st.halo_fwd_indices,
                           col_offset2=n_inner)
        R = sum(st.recv_counts)
        tip, tix, eperm = transpose_csr(ip, ix, n_inner + R)
        block = (ip, ix, tip, tix, eperm, R)
        self._gat_cache = (st.epoch, block)
        return block

    def gat_block_full(self):
        """Combined (inner ∪ FULL halo) block for GAT layer 0 under use_pp:
        every halo row participates, no sampling (reference model.py:118-120
        uses the full merged precompute tensor)."""
        if getattr(self, "_gat_full_cache", None) is not None:
            return self._gat_full_cache
        n_inner = self.plan.n_inner
        fip, fix, _ = transpose_csr(self.plan.halo_indptr,
                                    self.plan.halo_indices, n_inner)
        ip, ix = merge_csr(self.indptr, self.indices, fip, fix,
                           col_offset2=n_inner)
        n_halo = self.plan.halo_indptr.numel() - 1
        tip, tix, eperm = transpose_csr(ip, ix, n_inner + n_halo)
        self._gat_full_cache = (ip, ix, tip, tix, eperm, n_halo)
        return self._gat_full_cache

    def gat_split_halo(self):
        """Per-epoch sampled-halo edge set for the SPLIT GAT block:
        (fwd ip/ix rows=inner dst cols=recv idx, bwd ip/ix, eperm_t).
        Together with the STATIC inner CSRs this replaces the per-epoch
        merged block (no 14M-edge merge/transpose per epoch)."""
        st = self.plan.state
        return (st.halo_fwd_indptr, st.halo_fwd_indices,
                st.halo_bwd_indptr, st.halo_bwd_indices, st.halo_eperm_t)

    def gat_rows_inner(self):
        """Final-layer loss-row restriction for GAT: row-gathered INNER
        edge set (ip, ix, tip, tix, eperm_t) — static, cached."""
        if getattr(self, "_gat_rows_inner", None) is None:
            from ..ops.csr_torch import gather_rows_csr
            rows = self.loss_rows
            ip, ix = gather_rows_csr(self.indptr, self.indices, rows)
            # transpose_csr's eperm maps t-edge -> fwd-edge, which is
            # exactly the wperm the backward spmm_edge consumes
            tip, tix, eperm_t = transpose_csr(ip, ix, self.plan.n_inner)
            if ip.is_cuda:
                from ..ops.functional import _worklist_of
                _worklist_of(ip)
                _worklist_of(tip)
            self._gat_rows_inner = (ip, ix, tip, tix, eperm_t)
        return self._gat_rows_inner

    def gat_rows_halo(self, st: EpochState):
        """Row-gathered sampled-halo edge set for the restricted final
        GAT layer, cached per epoch (optionally pre-built on the prefetch
        stream — see prefetch_rows_halo)."""
        c = getattr(self, "_gat_rows_halo", None)
        if c is not None and c[0] == st.epoch:
            ev = c[2]
            if ev is not None:        # built on the prefetch stream
                cur = torch.cuda.current_stream()
                cur.wait_event(ev)
                for t in c[1]:
                    t.record_stream(cur)
                    wl = getattr(t, "_bns_worklist", None)
                    if wl is not None:
                        for w in wl:
                            w.record_stream(cur)
                self._gat_rows_halo = (st.epoch, c[1], None)
            return c[1]
        out = self._build_gat_rows_halo(st)
        self._gat_rows_halo = (st.epoch, out, None)
        return out

    def _build_gat_rows_halo(self, st: EpochState):
        from ..ops.csr_torch import gather_rows_csr
        rows = self.loss_rows
        hip_, hix_ = gather_rows_csr(st.halo_fwd_indptr,
                                     st.halo_fwd_indices, rows)
        R = int(st.halo_bwd_indptr.numel() - 1)
        hbip, hbix, eperm_t = transpose_csr(hip_, hix_, R)
        if hip_.is_cuda:
            from ..ops.functional import _worklist_of
            _worklist_of(hip_)
            _worklist_of(hbip)
        return (hip_, hix_, hbip, hbix, eperm_t)

    def gat_split_full(self):
        """Static FULL-halo edge set (GAT layer 0 under use_pp)."""
        if getattr(self, "_gat_split_full_cache", None) is None:
            fip, fix, eperm_f = transpose_csr(self.plan.halo_indptr,
                                              self.plan.halo_indices,
                                              self.plan.n_inner)
            eperm_t = torch.empty_like(eperm_f)
            eperm_t[eperm_f] = torch.arange(eperm_f.numel(),
                                            device=eperm_f.device)
            self._gat_split_full_cache = (fip, fix, self.plan.halo_indptr,
                                          self.plan.halo_indices, eperm_t)
        return self._gat_split_full_cache

    # -------------------------------------------------- full-halo exchange
    def full_state(self) -> EpochState:
        """Exchange plan at sampling rate 1.0 (use_pp precompute, reference
        train.py:170-211, and GAT layer 0 under use_pp)."""
        assert self.plan is not None
        if self._full_state is None:
            full = HaloPlan.__new__(HaloPlan)
            full.__dict__.update(self.plan.__dict__)
            full.rate = 1.0
            full.unit_ratio = True
            full.send_size = list(full.n_out)
            full.recv_size = list(full.n_in)
            full._state = None
            full._static = True
            self._full_state = full.set_epoch(0)
        return self._full_state
