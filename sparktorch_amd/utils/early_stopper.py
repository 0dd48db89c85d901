"""Early stopping.  Parity with reference sparktorch/early_stopper.py:8-56.

Semantics preserved exactly:
  * ``patience == 0`` disables early stopping (step always returns False).
  * NaN metric -> immediate stop.
  * ``min_delta`` may be absolute or a percentage of the previous best.
  * mode 'min' (loss-like) or 'max' (score-like).
"""

from __future__ import annotations

import math


class EarlyStopping(object):
    def __init__(self, mode: str = "min", min_delta: float = 0.0, patience: int = 10, percentage: bool = False):
        self.mode = mode
        self.min_delta = min_delta
        self.patience = patience
        self.percentage = percentage
        self.best = None
        self.num_bad_epochs = 0
        if patience == 0:
            self.is_better = lambda a, b: True
            self.step = lambda a: False  # type: ignore[assignment]
        else:
            self._init_is_better(mode, min_delta, percentage)

    def step(self, metric: float) -> bool:
        if metric != metric or (isinstance(metric, float) and math.isnan(metric)):
            return True  # NaN => stop (reference early_stopper.py:28-29)
        if self.best is None:
            self.best = metric
            return False
        if self.is_better(metric, self.best):
            self.num_bad_epochs = 0
            self.best = metric
        else:
            self.num_bad_epochs += 1
        return self.num_bad_epochs >= self.patience

    def _init_is_better(self, mode: str, min_delta: float, percentage: bool) -> None:
        if mode not in {"min", "max"}:
            raise ValueError("mode " + mode + " is unknown!")
        if not percentage:
            if mode == "min":
                self.is_better = lambda a, best: a < best - min_delta
            else:
                self.is_better = lambda a, best: a > best + min_delta
        else:
            if mode == "min":
                self.is_better = lambda a, best: a < best - (best * min_delta / 100)
            else:
                self.is_better = lambda a, best: a > best + (best * min_delta / 100)
