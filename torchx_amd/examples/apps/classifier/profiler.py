"""Stage-duration profiler for the classifier example (counterpart of the
reference's SimpleLoggingProfiler, torchx/examples/apps/lightning/
profiler.py:25-56 — stage wall times, summarized at the end and offered
to the experiment tracker as run metadata)."""

from __future__ import annotations

import logging
import time
from collections import defaultdict
from contextlib import contextmanager
from typing import Dict, Iterator

log = logging.getLogger(__name__)


class StageProfiler:
    def __init__(self) -> None:
        self.totals: Dict[str, float] = defaultdict(float)
        self.counts: Dict[str, int] = defaultdict(int)

    @contextmanager
    def stage(self, name: str) -> Iterator[None]:
        t0 = time.perf_counter()
        try:
            yield
        finally:
            dt = time.perf_counter() - t0
            self.totals[name] += dt
            self.counts[name] += 1

    def summary(self) -> Dict[str, float]:
        return {f"{k}_seconds": round(v, 4) for k, v in self.totals.items()}

    def report(self) -> None:
        for name in sorted(self.totals):
            log.info("stage %-12s total %8.3fs over %d calls", name,
                     self.totals[name], self.counts[name])
