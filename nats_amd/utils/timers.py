"""Wall-clock step timing (the reference's ud_start/ud pair, nats.py:1400-
1411) plus a simple scoped timer for profiling sections."""

import time
from collections import defaultdict


class StepTimer:
    """Accumulates per-section wall times; device-synchronising on CUDA so
    the numbers mean what they say."""

    def __init__(self, sync_cuda=True):
        self.totals = defaultdict(float)
        self.counts = defaultdict(int)
        self.sync_cuda = sync_cuda
        self._starts = {}

    def _sync(self):
        if self.sync_cuda:
            import torch
            if torch.cuda.is_available():
                torch.cuda.synchronize()

    def start(self, name):
        self._sync()
        self._starts[name] = time.perf_counter()

    def stop(self, name):
        self._sync()
        dt = time.perf_counter() - self._starts.pop(name)
        self.totals[name] += dt
        self.counts[name] += 1
        return dt

    def report(self):
        lines = []
        for name in sorted(self.totals):
            n = self.counts[name]
            lines.append("%-24s %8.3f ms/call x %d" % (
                name, 1e3 * self.totals[name] / max(n, 1), n))
        return "\n".join(lines)
