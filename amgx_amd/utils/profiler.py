"""Tracing/profiling (reference include/amgx_timer.h nvtxRange RAII +
per-level Profile.tic/toc, src/cycles/fixed_cycle.cu:97-150, src/profile.cu).

MI355X design: rocTX ranges via torch.cuda.nvtx (which drives roctracer on
ROCm, so rocprofv3 --marker-trace shows the phases) plus a host-side phase
accumulator that works identically on CPU."""

from __future__ import annotations

import contextlib
import time
from collections import defaultdict
from typing import Dict, Tuple

import torch

_HAS_NVTX = hasattr(torch.cuda, "nvtx")


@contextlib.contextmanager
def trace_range(name: str):
    """rocTX/NVTX range (reference nvtxRange, include/amgx_timer.h:15-42).
    No-op overhead when CUDA/HIP is unavailable."""
    pushed = False
    if _HAS_NVTX and torch.cuda.is_available():
        torch.cuda.nvtx.range_push(name)
        pushed = True
    try:
        yield
    finally:
        if pushed:
            torch.cuda.nvtx.range_pop()


class PhaseProfiler:
    """Accumulating tic/toc phase timer (reference AMG_Level Profile
    'Smoother'/'restrictRes'/... markers). ``enabled=False`` makes every
    call a cheap no-op; GPU phases are bracketed with stream sync only when
    ``sync`` is requested (accurate timing at the cost of overlap)."""

    def __init__(self, enabled: bool = False, sync: bool = False):
        self.enabled = enabled
        self.sync = sync and torch.cuda.is_available()
        self.acc: Dict[str, Tuple[int, float]] = defaultdict(
            lambda: (0, 0.0))
        self._open: Dict[str, float] = {}

    def tic(self, name: str):
        if not self.enabled:
            return
        if self.sync:
            torch.cuda.synchronize()
        self._open[name] = time.perf_counter()

    def toc(self, name: str):
        if not self.enabled or name not in self._open:
            return
        if self.sync:
            torch.cuda.synchronize()
        dt = time.perf_counter() - self._open.pop(name)
        c, t = self.acc[name]
        self.acc[name] = (c + 1, t + dt)

    @contextlib.contextmanager
    def phase(self, name: str):
        self.tic(name)
        with trace_range(name):
            yield
        self.toc(name)

    def report(self) -> str:
        lines = ["Phase profile:",
                 "  %-24s %8s %12s %12s" % ("phase", "calls", "total_s",
                                            "avg_ms")]
        for name in sorted(self.acc):
            c, t = self.acc[name]
            lines.append("  %-24s %8d %12.6f %12.4f"
                         % (name, c, t, 1e3 * t / max(c, 1)))
        return "\n".join(lines)
