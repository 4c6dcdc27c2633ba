"""Bucketed gradient all-reduce (data-parallel weight sync), overlapped
with backward.

Reference counterpart: helper/reducer.py — one process group PER PARAMETER
plus a CPU thread pool and pinned mirrors (a gloo workaround, SURVEY.md
C8), attached via per-param hooks so reduction overlaps the rest of
backward (reference train.py:337-338). The MI355X design keeps the
overlap but collapses the machinery: parameter gradients live as views
into one flat fp32 arena carved into fixed buckets; a
post-accumulate-grad hook on every parameter launches a bucket's async
all-reduce the moment its last parameter's gradient lands, so the
collective rides the communicator stream under the remaining backward
compute. Buckets launch in DESCENDING index order (backward completes
late-layer params — the tail of the arena — first), and the launch
sequence is kept identical across ranks by only issuing a bucket when all
higher-indexed buckets have been issued; stragglers are flushed by
`reduce()` after backward, so the reduced values are bit-identical to a
pure post-backward reduction. Scaled by 1/n_train at `synchronize()`,
matching the reference estimator (loss sum-reduced, grads divided by the
global train-node count, reducer.py:34, train.py:358-361). Bucket size
defaults to 16 MiB: on the 7-link xGMI clique each bucket's per-peer
share still saturates a link while keeping several buckets in flight.
"""
from __future__ import annotations

import os

import torch
import torch.distributed as dist


class GradReducer:
    def __init__(self, model: torch.nn.Module, n_train_global: int,
                 bucket_bytes: int = 16 << 20, overlap: bool | None = None):
        if overlap is None:   # BNSGCN_NO_REDUCE_OVERLAP=1: perf A/B switch
            overlap = os.environ.get("BNSGCN_NO_REDUCE_OVERLAP") != "1"
        self.params = [p for p in model.parameters() if p.requires_grad]
        self.n_train = float(n_train_global)
        total = sum(p.numel() for p in self.params)
        device = self.params[0].device if self.params else torch.device("cpu")
        self.flat = torch.zeros(total, dtype=torch.float32, device=device)
        # carve p.grad views out of the arena — autograd accumulates in place
        bucket_elems = max(1, bucket_bytes // 4)
        off = 0
        self.buckets: list[torch.Tensor] = []
        param_range: list[tuple[int, int]] = []
        for p in self.params:
            n = p.numel()
            p.grad = self.flat[off:off + n].view_as(p)
            param_range.append((off, off + n))
            off += n
        for lo in range(0, total, bucket_elems):
            self.buckets.append(self.flat[lo:lo + bucket_elems])
        nb = len(self.buckets)
        # how many params overlap each bucket (a param may span several)
        self._bucket_need = [0] * nb
        self._param_buckets: list[list[int]] = []
        for lo, hi in param_range:
            bs = list(range(lo // bucket_elems,
                            (max(hi - 1, lo)) // bucket_elems + 1))
            self._param_buckets.append(bs)
            for b in bs:
                self._bucket_need[b] += 1
        self._pending = list(self._bucket_need)
        self._launched = [False] * nb
        self._next = nb - 1          # highest-index bucket not yet launched
        self._works = []
        self.overlap = overlap
        if overlap:
            for i, p in enumerate(self.params):
                p.register_post_accumulate_grad_hook(
                    self._make_hook(self._param_buckets[i]))

    def _make_hook(self, bucket_ids: list[int]):
        def hook(_param):
            for b in bucket_ids:
                self._pending[b] -= 1
            self._launch_ready()
        return hook

    def _launch_ready(self):
        """Issue, in fixed descending order, every bucket whose params have
        all accumulated. The order gate keeps the RCCL collective sequence
        identical on every rank regardless of hook timing."""
        if not dist.is_initialized() or dist.get_world_size() == 1:
            return
        while self._next >= 0 and self._pending[self._next] == 0 \
                and not self._launched[self._next]:
            self._launched[self._next] = True
            self._works.append(
                dist.all_reduce(self.buckets[self._next], async_op=True))
            self._next -= 1

    def zero_grad(self):
        self.flat.zero_()
        self._pending = list(self._bucket_need)
        self._launched = [False] * len(self.buckets)
        self._next = len(self.buckets) - 1
        self._works = []

    def reduce(self):
        """Flush every not-yet-launched bucket (call after backward). With
        overlap on, most buckets are already in flight; params unused this
        step never fire hooks, so their buckets launch here."""
        if not dist.is_initialized() or dist.get_world_size() == 1:
            return
        for b in range(len(self.buckets) - 1, -1, -1):
            if not self._launched[b]:
                self._launched[b] = True
                self._works.append(
                    dist.all_reduce(self.buckets[b], async_op=True))
        self._next = -1

    def synchronize(self):
        """Wait for reductions and apply the 1/n_train scale."""
        for w in self._works:
            w.wait()
        self._works = []
        self.flat.div_(self.n_train)
