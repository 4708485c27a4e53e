"""Per-stage tracing: hipEvent-pair timing on the compute stream.

The reference has no tracing beyond wall-clock result counting
(test/test.py:28-36). SURVEY.md §5 requires per-stage timing via hipEvent
pairs; `EventTimer` brackets each stage forward with `torch.cuda.Event`
(hipEvent under ROCm) pairs on the current stream and accumulates device
time without host syncs until `total_ms()` is read. On CPU it degrades to
perf_counter. Kernel-level visibility stays with rocprofv3 (all hot
kernels are in-repo and named; see profiles/README.md).
"""

import time
from typing import List

import torch


class EventTimer:
    """Bounded ring of (start, stop) event pairs; ~zero overhead on the
    hot path (two hipEventRecord per item, no sync until harvest)."""

    def __init__(self, device: torch.device, depth: int = 64):
        self.is_cuda = device.type == "cuda"
        self.depth = depth
        self._n = 0
        self._ms = 0.0
        self._t0 = 0.0
        if self.is_cuda:
            self._pairs: List = [
                (torch.cuda.Event(enable_timing=True),
                 torch.cuda.Event(enable_timing=True))
                for _ in range(depth)]
            self._pending: List[int] = []

    def start(self):
        if self.is_cuda:
            if len(self._pending) == self.depth:
                self._harvest(1)
            i = self._n % self.depth
            self._pairs[i][0].record()
        else:
            self._t0 = time.perf_counter()

    def stop(self):
        if self.is_cuda:
            i = self._n % self.depth
            self._pairs[i][1].record()
            self._pending.append(i)
        else:
            self._ms += (time.perf_counter() - self._t0) * 1e3
        self._n += 1

    def _harvest(self, at_least: int = 0):
        while self._pending:
            i = self._pending[0]
            s, e = self._pairs[i]
            if not at_least and not e.query():
                break
            e.synchronize()
            self._ms += s.elapsed_time(e)
            self._pending.pop(0)
            at_least = max(0, at_least - 1)

    @property
    def count(self) -> int:
        return self._n

    def total_ms(self) -> float:
        if self.is_cuda:
            self._harvest(len(self._pending))
        return self._ms

    def mean_ms(self) -> float:
        return self.total_ms() / max(self._n, 1)
