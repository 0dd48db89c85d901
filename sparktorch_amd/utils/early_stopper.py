"""Early stopping with the reference's exact observable semantics
(reference sparktorch/early_stopper.py:8-56, itself adapted from a public
gist):

  * ``patience == 0`` disables stopping entirely — ``step`` always returns
    False, even for NaN, and an invalid ``mode`` is not rejected.
  * NaN metric -> immediate stop (when enabled).
  * ``min_delta`` is an absolute margin, or a percentage of the current
    best when ``percentage=True`` (computed from the signed best, matching
    the reference).
  * mode ``'min'`` for loss-like metrics, ``'max'`` for score-like.

Independent design: instead of the gist's per-mode comparison-lambda
factory, one ``is_better`` method computes the signed improvement margin
directly.
"""

from __future__ import annotations

class EarlyStopping:
    def __init__(
        self,
        mode: str = "min",
        min_delta: float = 0.0,
        patience: int = 10,
        percentage: bool = False,
    ):
        self.mode = mode
        self.min_delta = min_delta
        self.patience = patience
        self.percentage = percentage
        self.best: float | None = None
        self.num_bad_epochs = 0
        if patience != 0 and mode not in ("min", "max"):
            raise ValueError("mode " + mode + " is unknown!")

    def is_better(self, metric: float, best: float) -> bool:
        """Did ``metric`` improve on ``best`` by more than the margin?"""
        margin = (best * self.min_delta / 100.0) if self.percentage else self.min_delta
        if self.mode == "min":
            return metric < best - margin
        return metric > best + margin

    def step(self, metric: float) -> bool:
        """Record one epoch's metric; True means training should stop."""
        if self.patience == 0:
            return False
        if metric != metric:  # NaN of any float flavor
            return True
        if self.best is None:
            self.best = metric
            return False
        if self.is_better(metric, self.best):
            self.best = metric
            self.num_bad_epochs = 0
        else:
            self.num_bad_epochs += 1
        return self.num_bad_epochs >= self.patience
