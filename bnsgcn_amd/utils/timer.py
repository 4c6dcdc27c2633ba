"""Comm/compute timers (reference: helper/timer/comm_timer.py).

The reference uses wall clocks, valid only because its gloo transfers block
the CPU. Our collectives are stream-async, so GPU spans
are measured with HIP events recorded on the issuing stream and summed at
epoch end; CPU spans use perf_counter.
"""
from __future__ import annotations

import time
from contextlib import contextmanager

import torch


class CommTimer:
    def __init__(self):
        self._cpu: dict[str, float] = {}
        self._events: list[tuple[str, torch.cuda.Event, torch.cuda.Event]] = []

    @contextmanager
    def span(self, name: str, cuda: bool = False):
        if cuda:
            s = torch.cuda.Event(enable_timing=True)
            e = torch.cuda.Event(enable_timing=True)
            s.record()
            yield
            e.record()
            self._events.append((name, s, e))
        else:
            t0 = time.perf_counter()
            yield
            self._cpu[name] = self._cpu.get(name, 0.0) + time.perf_counter() - t0

    def tot_time(self) -> float:
        """Total seconds across all spans (synchronizes pending GPU events)."""
        tot = sum(self._cpu.values())
        if self._events:
            torch.cuda.synchronize()
            for _, s, e in self._events:
                tot += s.elapsed_time(e) / 1e3
        return tot

    def clear(self):
        self._cpu.clear()
        self._events.clear()


comm_timer = CommTimer()
