"""Tracing / verbose logging.

Parity with the reference's observability: wall-clock
step timing macros + DEBUG-gated printf of every RPC. Here:
  * `trace_range(name)` — rocTX/NVTX range context (visible in rocprofv3
    `--marker-trace` timelines) + optional wall-clock accumulation
  * `debug_log(...)` — env-gated (LCTR_DEBUG=1) verbose logging used by
    the PS/collective layers
  * `StepTimer` — per-step rolling wall-clock stats.
"""

from __future__ import annotations

import os
import time
from contextlib import contextmanager

import torch

DEBUG = os.environ.get("LCTR_DEBUG", "0") not in ("", "0", "false")


def debug_log(*args) -> None:
    if DEBUG:
        print("[lctr]", *args, flush=True)


@contextmanager
def trace_range(name: str):
    use_nvtx = torch.cuda.is_available()
    if use_nvtx:
        torch.cuda.nvtx.range_push(name)  # rocTX on ROCm
    t0 = time.perf_counter()
    try:
        yield
    finally:
        if use_nvtx:
            torch.cuda.nvtx.range_pop()
        debug_log(f"{name}: {(time.perf_counter() - t0) * 1e3:.3f} ms")


class StepTimer:
    def __init__(self, window: int = 100):
        self.window = window
        self.times: list[float] = []
        self._t0 = None

    def start(self):
        self._t0 = time.perf_counter()

    def stop(self) -> float:
        dt = time.perf_counter() - self._t0
        self.times.append(dt)
        if len(self.times) > self.window:
            self.times.pop(0)
        return dt

    def mean_ms(self) -> float:
        return sum(self.times) / max(1, len(self.times)) * 1e3
