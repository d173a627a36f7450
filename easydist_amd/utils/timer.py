"""EDTimer: warmed event-based step timer.

Capability parity with reference ``easydist/utils/timer.py`` (EDTimer
23-130: CUDA-event / CPU variants with warmup + trials). On ROCm,
torch.cuda.Event is a HIP event.
"""
from __future__ import annotations

import time
from typing import Callable, Optional

import torch


class EDTimer:
    def __init__(self, func: Callable, trials: int = 10, warmup: int = 3,
                 in_ms: bool = True, device: Optional[str] = None):
        self.func = func
        self.trials = trials
        self.warmup = warmup
        self.in_ms = in_ms
        self.use_cuda = (device or
                         ("cuda" if torch.cuda.is_available() else "cpu")
                         ).startswith("cuda")

    def time(self) -> float:
        for _ in range(self.warmup):
            self.func()
        if self.use_cuda:
            torch.cuda.synchronize()
            start = torch.cuda.Event(enable_timing=True)
            end = torch.cuda.Event(enable_timing=True)
            start.record()
            for _ in range(self.trials):
                self.func()
            end.record()
            end.synchronize()
            ms = start.elapsed_time(end) / self.trials
        else:
            t0 = time.perf_counter()
            for _ in range(self.trials):
                self.func()
            ms = (time.perf_counter() - t0) * 1000.0 / self.trials
        return ms if self.in_ms else ms / 1000.0
