"""Fused optimizers over flat buckets.

Replaces the reference's CPU ``optimizer.step()`` (distributed.py:198,
server.py:139) with ONE HIP kernel launch per flat bucket per step
(multi-element grid-stride kernel, grad 1/world scaling folded in — the
divide at reference distributed.py:181 costs an extra pass there).

On CUDA(ROCm) devices the `_sparkhip` extension is REQUIRED (fail-loud); on
CPU a numerically identical torch implementation runs so the same trainer code
is testable in the no-GPU sandbox.
"""

from __future__ import annotations

import math
from typing import List, Optional, Tuple

import torch

from sparktorch_amd import ops
from sparktorch_amd.parallel.buckets import FlatBuckets


class _FusedOptimizerBase:
    def __init__(self, buckets: FlatBuckets):
        self.buckets = buckets
        self.step_count = 0

    def zero_grad(self) -> None:
        self.buckets.zero_grad()

    @property
    def _on_gpu(self) -> bool:
        pairs = self.buckets.flat_pairs()
        return bool(pairs) and pairs[0][0].is_cuda

    def state_dict(self):
        return {"step_count": self.step_count}


class FusedAdam(_FusedOptimizerBase):
    """Adam/AdamW on flat buckets; bitwise-matches torch.optim.Adam math."""

    def __init__(
        self,
        buckets: FlatBuckets,
        lr: float = 1e-3,
        betas: Tuple[float, float] = (0.9, 0.999),
        eps: float = 1e-8,
        weight_decay: float = 0.0,
        adamw: bool = False,
    ):
        super().__init__(buckets)
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.adamw = adamw
        self.exp_avg: List[torch.Tensor] = []
        self.exp_avg_sq: List[torch.Tensor] = []
        self.step_dev: Optional[torch.Tensor] = None  # device step counter
        for flat_p, _ in buckets.flat_pairs():
            self.exp_avg.append(torch.zeros_like(flat_p))
            self.exp_avg_sq.append(torch.zeros_like(flat_p))
            if self.step_dev is None and flat_p.is_cuda:
                # bias corrections are computed in-kernel from this counter,
                # so a hipGraph-captured step stays correct across replays
                self.step_dev = torch.zeros(1, dtype=torch.int32, device=flat_p.device)

    @torch.no_grad()
    def step(self, grad_scale: float = 1.0) -> None:
        self.step_count += 1
        t = self.step_count
        bc1 = 1.0 - self.beta1**t
        bc2 = 1.0 - self.beta2**t
        stepped_dev = False
        for i, (p, g) in enumerate(self.buckets.flat_pairs()):
            if p.is_cuda:
                if not stepped_dev and self.step_dev is not None:
                    ops.ext().increment_i32(self.step_dev)
                    stepped_dev = True
                ops.ext().fused_adam(
                    p,
                    g,
                    self.exp_avg[i],
                    self.exp_avg_sq[i],
                    self.lr,
                    self.beta1,
                    self.beta2,
                    self.eps,
                    self.weight_decay,
                    bc1,
                    bc2,
                    grad_scale,
                    self.adamw,
                    self.step_dev,
                )
            else:
                m, v = self.exp_avg[i], self.exp_avg_sq[i]
                if grad_scale != 1.0:
                    g = g * grad_scale
                if self.weight_decay != 0.0:
                    if self.adamw:
                        p.mul_(1.0 - self.lr * self.weight_decay)
                    else:
                        g = g.add(p, alpha=self.weight_decay)
                m.mul_(self.beta1).add_(g, alpha=1.0 - self.beta1)
                v.mul_(self.beta2).addcmul_(g, g, value=1.0 - self.beta2)
                denom = (v / bc2).sqrt_().add_(self.eps)
                p.addcdiv_(m, denom, value=-self.lr / bc1)


class FusedSGD(_FusedOptimizerBase):
    def __init__(
        self,
        buckets: FlatBuckets,
        lr: float = 0.01,
        momentum: float = 0.0,
        weight_decay: float = 0.0,
        nesterov: bool = False,
        dampening: float = 0.0,
    ):
        super().__init__(buckets)
        self.lr = lr
        self.momentum = momentum
        self.weight_decay = weight_decay
        self.nesterov = nesterov
        self.dampening = dampening
        self.momentum_buf: List[Optional[torch.Tensor]] = [
            torch.zeros_like(p) if momentum != 0.0 else None for p, _ in buckets.flat_pairs()
        ]

    @torch.no_grad()
    def step(self, grad_scale: float = 1.0) -> None:
        self.step_count += 1
        first = self.step_count == 1
        for i, (p, g) in enumerate(self.buckets.flat_pairs()):
            if p.is_cuda:
                ops.ext().fused_sgd(
                    p,
                    g,
                    self.momentum_buf[i] if self.momentum_buf[i] is not None else g,
                    self.lr,
                    self.momentum,
                    self.weight_decay,
                    self.dampening,
                    grad_scale,
                    self.nesterov,
                    first,
                    self.momentum_buf[i] is not None,
                )
            else:
                if grad_scale != 1.0:
                    g = g * grad_scale
                if self.weight_decay != 0.0:
                    g = g.add(p, alpha=self.weight_decay)
                buf = self.momentum_buf[i]
                if buf is not None:
                    if first:
                        buf.copy_(g)
                    else:
                        buf.mul_(self.momentum).add_(g, alpha=1.0 - self.dampening)
                    g = g.add(buf, alpha=self.momentum) if self.nesterov else buf
                p.add_(g, alpha=-self.lr)


def fused_optimizer_for(torch_opt: torch.optim.Optimizer, buckets: FlatBuckets):
    """Map a hydrated torch optimizer onto its fused flat-bucket equivalent.

    Returns None when no fused mapping exists (trainer then keeps the torch
    optimizer, re-bound to the flattened params).
    """
    # NB: match by class NAME, not isinstance — dill round-trips torch classes
    # by value, so a deserialized Adam is a distinct class object.
    d = torch_opt.defaults
    name = type(torch_opt).__name__
    if name == "AdamW":
        return FusedAdam(
            buckets,
            lr=d.get("lr", 1e-3),
            betas=d.get("betas", (0.9, 0.999)),
            eps=d.get("eps", 1e-8),
            weight_decay=d.get("weight_decay", 0.0),
            adamw=True,
        )
    if name == "Adam":
        return FusedAdam(
            buckets,
            lr=d.get("lr", 1e-3),
            betas=d.get("betas", (0.9, 0.999)),
            eps=d.get("eps", 1e-8),
            weight_decay=d.get("weight_decay", 0.0),
            adamw=False,
        )
    if name == "SGD":
        return FusedSGD(
            buckets,
            lr=d.get("lr", 0.01),
            momentum=d.get("momentum", 0.0),
            weight_decay=d.get("weight_decay", 0.0),
            nesterov=d.get("nesterov", False),
            dampening=d.get("dampening", 0.0),
        )
    return None
