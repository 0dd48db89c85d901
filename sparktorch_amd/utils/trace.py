"""Tracing and per-iteration metrics.

The reference has no profiling support at all — only ``verbose`` prints per
iteration (reference distributed.py:200-203, hogwild.py:133-134).  Here every
engine can
* wrap hot-loop phases in rocprof-visible ranges (``trace_range`` — on ROCm,
  ``torch.cuda.nvtx`` emits roctx markers that show up in
  ``rocprofv3 --kernel-trace`` timelines), and
* record per-iteration wall times and losses in a lightweight
  :class:`StepMetrics` whose summary is printed at ``verbose`` level and
  returned to the driver for observability.
"""

from __future__ import annotations

import time
from contextlib import contextmanager
from typing import Dict, List, Optional

import torch

_NVTX_OK: Optional[bool] = None


def _nvtx_available() -> bool:
    global _NVTX_OK
    if _NVTX_OK is None:
        try:
            _NVTX_OK = torch.cuda.is_available() and hasattr(torch.cuda, "nvtx")
        except Exception:
            _NVTX_OK = False
    return _NVTX_OK


@contextmanager
def trace_range(name: str):
    """rocprof-visible marker range (no-op on CPU-only hosts)."""
    pushed = False
    if _nvtx_available():
        try:
            torch.cuda.nvtx.range_push(name)
            pushed = True
        except Exception:
            pass
    try:
        yield
    finally:
        if pushed:
            torch.cuda.nvtx.range_pop()


class StepMetrics:
    """Per-iteration wall-time/loss counters with O(1) memory.

    Keeps running aggregates plus the most recent ``window`` samples so both
    "whole run" and "steady state" rates can be reported (warmup iterations
    dominate naive averages on a GPU: allocator, autotune, graph capture).
    """

    def __init__(self, window: int = 50):
        self.window = window
        self.count = 0
        self.total_s = 0.0
        self.last_ms: List[float] = []
        self.last_losses: List[float] = []
        self._t0: Optional[float] = None

    def start(self) -> None:
        self._t0 = time.perf_counter()

    def stop(self, loss: Optional[float] = None) -> float:
        dt = time.perf_counter() - (self._t0 if self._t0 is not None else time.perf_counter())
        self._t0 = None
        self.count += 1
        self.total_s += dt
        self.last_ms.append(dt * 1000.0)
        if len(self.last_ms) > self.window:
            self.last_ms.pop(0)
        if loss is not None:
            self.last_losses.append(float(loss))
            if len(self.last_losses) > self.window:
                self.last_losses.pop(0)
        return dt

    @contextmanager
    def step(self):
        self.start()
        try:
            yield
        finally:
            self.stop()

    def summary(self) -> Dict[str, float]:
        steady = self.last_ms[len(self.last_ms) // 2 :] or [0.0]
        return {
            "iters": self.count,
            "total_s": round(self.total_s, 4),
            "avg_ms": round(self.total_s / self.count * 1000.0, 4) if self.count else 0.0,
            "steady_ms": round(sum(steady) / len(steady), 4),
            "last_loss": self.last_losses[-1] if self.last_losses else float("nan"),
        }

    def __repr__(self) -> str:  # pragma: no cover - cosmetic
        s = self.summary()
        return "StepMetrics(iters=%d, avg=%.3fms, steady=%.3fms, last_loss=%.5f)" % (
            s["iters"], s["avg_ms"], s["steady_ms"], s["last_loss"],
        )
