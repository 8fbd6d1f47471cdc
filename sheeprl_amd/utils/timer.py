"""Wall-clock section timer.

Parity with the reference's ``timer`` context/decorator (sheeprl/utils/timer.py:16-83):
class-level dict of accumulated seconds keyed by section name, globally
disable-able, drained at log time.  Extended for the MI355X build with optional
rocprof/roctx-style markers: when running under rocprofv3 the named ranges show
up in the trace (we emit them with ``torch.cuda.nvtx`` which maps to roctx on
ROCm builds).
"""

from __future__ import annotations

import time
from typing import Dict

import torch

_HAS_NVTX = torch.cuda.is_available()


class timer:
    disabled: bool = False
    timers: Dict[str, float] = {}
    counts: Dict[str, int] = {}

    def __init__(self, name: str) -> None:
        self.name = name

    def __enter__(self) -> "timer":
        if not timer.disabled:
            self._start = time.perf_counter()
            if _HAS_NVTX:
                torch.cuda.nvtx.range_push(self.name)
        return self

    def __exit__(self, *exc) -> None:
        if not timer.disabled:
            if _HAS_NVTX:
                torch.cuda.nvtx.range_pop()
            dt = time.perf_counter() - self._start
            timer.timers[self.name] = timer.timers.get(self.name, 0.0) + dt
            timer.counts[self.name] = timer.counts.get(self.name, 0) + 1

    @classmethod
    def compute(cls) -> Dict[str, float]:
        return dict(cls.timers)

    @classmethod
    def reset(cls) -> None:
        cls.timers.clear()
        cls.counts.clear()
