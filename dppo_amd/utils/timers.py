"""Per-phase wall-clock timers (rollout / gae / update /
allreduce / opt).  The reference prints only total elapsed time
(reference main.py:52,65); the rebuild tracks each phase so rocprof
numbers can be attributed.
"""

from __future__ import annotations

import time
from collections import defaultdict
from contextlib import contextmanager
from typing import Dict


class PhaseTimers:
    def __init__(self, cuda_sync: bool = False):
        self.totals: Dict[str, float] = defaultdict(float)
        self.counts: Dict[str, int] = defaultdict(int)
        self.cuda_sync = cuda_sync

    @contextmanager
    def phase(self, name: str):
        if self.cuda_sync:
            import torch

            if torch.cuda.is_available():
                torch.cuda.synchronize()
        t0 = time.perf_counter()
        try:
            yield
        finally:
            if self.cuda_sync:
                import torch

                if torch.cuda.is_available():
                    torch.cuda.synchronize()
            self.totals[name] += time.perf_counter() - t0
            self.counts[name] += 1

    def summary(self) -> Dict[str, float]:
        return dict(self.totals)

    def reset(self) -> None:
        self.totals.clear()
        self.counts.clear()
