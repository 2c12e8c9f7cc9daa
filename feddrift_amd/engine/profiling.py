"""Per-phase wall-clock profiling for the round loop.

The reference's only tracing is an 'aggregate time cost' log line
(FedAvgEnsAggregatorSoftCluster.py:138,193). Here every round phase
(plan / train / aggregate / cluster / test) is timed; summaries feed the
metrics logger and the rounds/sec reports. GPU-side per-kernel numbers
come from rocprofv3 (see profiles/README.md).
"""

from __future__ import annotations

import time
from collections import defaultdict
from contextlib import contextmanager
from typing import Dict


class PhaseTimer:
    def __init__(self, sync_fn=None):
        self.total: Dict[str, float] = defaultdict(float)
        self.count: Dict[str, int] = defaultdict(int)
        self.sync_fn = sync_fn

    @contextmanager
    def phase(self, name: str):
        t0 = time.perf_counter()
        try:
            yield
        finally:
            if self.sync_fn is not None:
                self.sync_fn()
            self.total[name] += time.perf_counter() - t0
            self.count[name] += 1

    def summary(self) -> Dict[str, Dict[str, float]]:
        return {k: {"total_s": round(self.total[k], 4),
                    "calls": self.count[k],
                    "avg_ms": round(self.total[k] / self.count[k] * 1e3, 4)}
                for k in self.total}
