"""HaloPlan — per-job static layout + per-epoch BNS sampling state.

Replaces the reference's Buffer-size bookkeeping + per-epoch graph rebuild
(reference: train.py:90-131 get_pos/get_send_size/get_recv_size,
train.py:225-281 select_node/construct_graph, helper/feature_buffer.py
init_buffer). Differences by design (MI355X-first):

* sampling uses the shared Philox counter-RNG (ops/philox.py), so sender
  and receiver derive identical per-epoch samples locally and the NODE-id
  all-to-all (reference C3, train.py:389) is eliminated;
* the per-epoch sampled-halo graph is two CSR gathers over the static halo
  CSR (ops/csr_torch.py), not a DGL heterograph rebuild;
* send/recv sizes keep the reference's deterministic `int(p·n)` contract
  (train.py:107-131) and the 1/ratio unbiasing with ratio = s/n
  (train.py:118, feature_buffer.py:117,129) applied at pack AND at
  grad-scatter — the gloo semantics, which are the correct estimator (the
  reference's mpi path forgets the backward rescale).
"""
from __future__ import annotations

from dataclasses import dataclass

import numpy as np
import torch

from ..graph.store import Partition
from ..ops.csr_torch import gather_rows_csr, transpose_csr
from ..ops.philox import sample_boundary


@dataclass
class EpochState:
    epoch: int
    send_counts: list        # rows I send to each peer (len P)
    recv_counts: list        # rows I receive from each peer (len P)
    pack_idx: torch.Tensor   # [S] inner-local rows to pack, peer-major
    pack_scale: torch.Tensor | None  # [S] 1/ratio per row (None if all 1)
    hsel: torch.Tensor       # [R] halo rows selected, peer-major (= recv order)
    # sampled halo block, fwd: rows = inner dst, cols = recv-row index
    halo_fwd_indptr: torch.Tensor = None
    halo_fwd_indices: torch.Tensor = None
    # bwd: rows = recv-row index, cols = inner dst
    halo_bwd_indptr: torch.Tensor = None
    halo_bwd_indices: torch.Tensor = None
    # weighted-transpose edge map: halo_eperm_t[j] = fwd-edge id of the
    # j-th bwd edge (for per-edge payloads through the transpose)
    halo_eperm_t: torch.Tensor = None
    halo_out_norm_inv: torch.Tensor = None   # [R] 1/sqrt(out_deg) of recv rows
    halo_in_deg: torch.Tensor = None         # [R] full in-degree of recv rows


def _state_tensors(st: EpochState):
    """Every device tensor in an EpochState, incl. cached worklists (for
    cross-stream record_stream when the state was built by prefetch)."""
    out = []
    for f in ("pack_idx", "pack_scale", "hsel", "halo_fwd_indptr",
              "halo_fwd_indices", "halo_bwd_indptr", "halo_bwd_indices",
              "halo_eperm_t", "halo_out_norm_inv", "halo_in_deg"):
        t = getattr(st, f)
        if t is not None:
            out.append(t)
            wl = getattr(t, "_bns_worklist", None)
            if wl is not None:
                out.extend(wl)
    return out


class HaloPlan:
    """One per training job per rank. `set_epoch(e)` refreshes sampling."""

    def __init__(self, part: Partition, sampling_rate: float, seed: int,
                 device: torch.device | str, unit_ratio: bool = False,
                 wire_dtype: torch.dtype | None = None):
        # wire_dtype=torch.bfloat16 (--halo-dtype bf16): halo payloads are
        # downcast at the wire, halving per-link xGMI bytes; compute and
        # the 1/ratio estimator stay fp32 (an OPTION — default full fp32)
        self.wire_dtype = wire_dtype
        self.device = torch.device(device)
        self.rank = part.rank
        self.n_parts = part.n_parts
        self.n_inner = part.n_inner
        self.rate = float(sampling_rate)
        self.seed = int(seed)
        # unit_ratio=True disables the 1/ratio rescale (GAT — the reference's
        # own TODO says attention inputs must not be ratio-scaled,
        # train.py:117 / SURVEY.md §2.5.3).
        self.unit_ratio = bool(unit_ratio)

        dev = self.device
        self.n_out = [len(b) for b in part.boundary]              # send-side
        sl = part.halo_peer_slices()
        self.n_in = [s.stop - s.start for s in sl]                # recv-side
        self.halo_start = [s.start for s in sl]
        self.boundary = [torch.from_numpy(np.ascontiguousarray(b)).long().to(dev)
                         for b in part.boundary]
        self.halo_indptr = torch.from_numpy(part.halo_indptr).to(dev)
        self.halo_indices = torch.from_numpy(part.halo_indices).to(dev)
        self.halo_out_norm_inv_full = torch.from_numpy(
            1.0 / np.sqrt(np.maximum(part.halo_out_deg, 1))).float().to(dev)
        self.halo_in_deg_full = torch.from_numpy(
            part.halo_in_deg.astype(np.float32)).to(dev)

        # deterministic size contract int(p*n) — both sides agree
        self.send_size = [int(self.rate * n) for n in self.n_out]
        self.recv_size = [int(self.rate * n) for n in self.n_in]
        self.ratio = [ (self.send_size[j] / self.n_out[j]) if self.n_out[j] else 0.0
                       for j in range(self.n_parts)]
        self._state: EpochState | None = None
        self._static = self.rate >= 1.0 or self.rate <= 0.0

    def _sample(self, n: int, s: int, epoch: int, src: int, dst: int) -> torch.Tensor:
        """Sorted s-subset of [0,n) — device-side Philox + torch stable sort
        on GPU (reference K16 was CPU np.random.choice),
        numpy Philox on CPU; both bitwise-identical."""
        dev = self.device
        if s >= n:
            return torch.arange(n, dtype=torch.long, device=dev)
        if dev.type == "cuda":
            from ..ops._ext import get_ext, has_ext
            if has_ext():
                with torch.cuda.device(dev):
                    keys = get_ext().philox_keys(n, self.seed, epoch, src, dst)
                order = torch.argsort(keys, stable=True)[:s]
                return torch.sort(order)[0]
        return torch.from_numpy(
            sample_boundary(n, s, self.seed, epoch, src, dst)).to(dev)

    # ------------------------------------------------------------------
    def prefetch(self, epoch: int) -> None:
        """Build epoch `epoch`'s sampling state on a SIDE stream so the
        gathers/sorts/transposes overlap the CURRENT epoch's backward on
        the GPU (the plan is a pure function of seed+epoch — independent
        of training state). `set_epoch(epoch)` then just installs it.
        No-op on CPU or for static rates."""
        import os
        if self._static or self.device.type != "cuda" \
                or os.environ.get("BNSGCN_NO_PREFETCH") == "1":
            return
        import torch as _t
        if not hasattr(self, "_prefetch_stream"):
            self._prefetch_stream = _t.cuda.Stream()
        s = self._prefetch_stream
        ev = _t.cuda.current_stream().record_event()
        with _t.cuda.stream(s):
            s.wait_event(ev)        # boundary/halo tensors are settled
            st = self._build(epoch)
            done = s.record_event()
        self._next = (epoch, st, done)

    def set_epoch(self, epoch: int) -> EpochState:
        if self._static and self._state is not None:
            return self._state
        nxt = getattr(self, "_next", None)
        if nxt is not None and nxt[0] == epoch:
            self._next = None
            _, st, done = nxt
            import torch as _t
            cur = _t.cuda.current_stream()
            cur.wait_event(done)
            for t in _state_tensors(st):
                t.record_stream(cur)
            self._state = st
            return st
        st = self._build(epoch)
        self._state = st
        return st

    def _build(self, epoch: int) -> EpochState:
        dev = self.device
        me = self.rank
        pack_parts, scale_parts, hsel_parts = [], [], []
        for j in range(self.n_parts):
            if j == me:
                continue
            # outgoing sample (me -> j)
            s, n = self.send_size[j], self.n_out[j]
            if s > 0:
                pos = self._sample(n, s, epoch, me, j)
                pack_parts.append(self.boundary[j][pos])
                if not self.unit_ratio:
                    scale_parts.append(torch.full((s,), n / s, dtype=torch.float32,
                                                  device=self.device))
            # incoming sample (j -> me)
            r, m = self.recv_size[j], self.n_in[j]
            if r > 0:
                pos = self._sample(m, r, epoch, j, me)
                hsel_parts.append(self.halo_start[j] + pos)

        pack_idx = (torch.cat(pack_parts) if pack_parts
                    else torch.zeros(0, dtype=torch.long, device=dev))
        pack_scale = (torch.cat(scale_parts) if scale_parts else None)
        if self.unit_ratio:
            pack_scale = None
        hsel = (torch.cat(hsel_parts) if hsel_parts
                else torch.zeros(0, dtype=torch.long, device=dev))

        bwd_ip, bwd_ix = gather_rows_csr(self.halo_indptr, self.halo_indices, hsel)
        fwd_ip, fwd_ix, eperm_f = transpose_csr(bwd_ip, bwd_ix, self.n_inner)
        # eperm_f[k] = bwd edge of fwd edge k; invert for bwd -> fwd
        eperm_t = torch.empty_like(eperm_f)
        eperm_t[eperm_f] = torch.arange(eperm_f.numel(), device=eperm_f.device)
        if dev.type == "cuda":
            from ..ops.functional import _worklist_of
            _worklist_of(bwd_ip)
            _worklist_of(fwd_ip)

        st = EpochState(
            epoch=epoch,
            send_counts=list(self.send_size), recv_counts=list(self.recv_size),
            pack_idx=pack_idx, pack_scale=pack_scale, hsel=hsel,
            halo_fwd_indptr=fwd_ip, halo_fwd_indices=fwd_ix,
            halo_bwd_indptr=bwd_ip, halo_bwd_indices=bwd_ix,
            halo_eperm_t=eperm_t,
            halo_out_norm_inv=self.halo_out_norm_inv_full[hsel],
            halo_in_deg=self.halo_in_deg_full[hsel],
        )
        st.send_counts[me] = 0
        st.recv_counts[me] = 0
        return st

    @property
    def state(self) -> EpochState:
        assert self._state is not None, "call set_epoch() first"
        return self._state

    def total_recv(self) -> int:
        return sum(self.state.recv_counts)
