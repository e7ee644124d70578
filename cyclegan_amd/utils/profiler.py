"""Lightweight training-loop observability (SURVEY §5 tracing row).

- ``StepTimer``: wall-clock per-step timing with device sync options.
- ``HipEventTimer``: hipEvent-based GPU interval timing (e.g. around the
  gradient all-reduce) without host syncs until ``elapsed()`` is read.
- rocprofv3 recipe for kernel-level profiling (no code needed):

      cd /tmp && export TMPDIR=/tmp
      rocprofv3 --kernel-trace --stats --output-format csv -d out -o kt -- \
          python bench.py --steps 6 --warmup 2
      # per-kernel totals in out/kt_kernel_stats.csv
      # PMC counters (own run): rocprofv3 --pmc SQ_LDS_BANK_CONFLICT,...
"""

from __future__ import annotations

import time
from collections import defaultdict
from typing import Dict, List, Optional

import torch


class StepTimer:
    """Accumulates wall-clock intervals per tag; optional CUDA sync."""

    def __init__(self, sync: bool = False):
        self.sync = sync and torch.cuda.is_available()
        self.totals: Dict[str, float] = defaultdict(float)
        self.counts: Dict[str, int] = defaultdict(int)
        self._t0: Dict[str, float] = {}

    def start(self, tag: str):
        if self.sync:
            torch.cuda.synchronize()
        self._t0[tag] = time.perf_counter()

    def stop(self, tag: str):
        if self.sync:
            torch.cuda.synchronize()
        self.totals[tag] += time.perf_counter() - self._t0.pop(tag)
        self.counts[tag] += 1

    def summary(self) -> Dict[str, float]:
        return {k: self.totals[k] / max(1, self.counts[k]) for k in self.totals}

    def reset(self):
        self.totals.clear()
        self.counts.clear()


class HipEventTimer:
    """GPU interval timing via events; read elapsed() after a sync point."""

    def __init__(self):
        self._pairs: List[tuple] = []

    def begin(self) -> Optional[tuple]:
        if not torch.cuda.is_available():
            return None
        s = torch.cuda.Event(enable_timing=True)
        e = torch.cuda.Event(enable_timing=True)
        s.record()
        self._pairs.append((s, e))
        return (s, e)

    def end(self):
        if self._pairs:
            self._pairs[-1][1].record()

    def elapsed_ms(self) -> float:
        """Total ms across recorded intervals (syncs)."""
        if not self._pairs:
            return 0.0
        torch.cuda.synchronize()
        total = sum(s.elapsed_time(e) for s, e in self._pairs)
        self._pairs.clear()
        return total
