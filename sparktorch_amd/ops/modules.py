"""nn.Module wrappers over the native kernels + the model converter.

``convert_model_for_mi355x`` swaps every ``nn.Linear`` in a user's serialized
model for :class:`HipLinear` (ADOPTING the same Parameter objects, so flat
buckets / optimizers see identical tensors) and maps the criterion onto its
fused equivalent.  This is how models from the reference's zoo
(tests/simple_net.py, examples/*.py) run on the hand-written MFMA path
without user changes.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from sparktorch_amd.ops.functional import hip_cross_entropy, hip_linear, hip_mse


class HipLinear(nn.Module):
    """Drop-in nn.Linear on the MFMA path, with optional fused ReLU."""

    def __init__(self, in_features: int, out_features: int, bias: bool = True,
                 activation: Optional[str] = None):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        self.bias = nn.Parameter(torch.zeros(out_features)) if bias else None
        self.activation = activation
        nn.init.kaiming_uniform_(self.weight, a=5 ** 0.5)

    @classmethod
    def from_linear(cls, lin: nn.Linear, activation: Optional[str] = None) -> "HipLinear":
        m = cls.__new__(cls)
        nn.Module.__init__(m)
        m.in_features = lin.in_features
        m.out_features = lin.out_features
        m.weight = lin.weight  # SAME Parameter object
        m.bias = lin.bias
        m.activation = activation
        return m

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return hip_linear(x, self.weight, self.bias, relu=self.activation == "relu")

    def extra_repr(self) -> str:
        return "in=%d, out=%d, act=%s" % (self.in_features, self.out_features, self.activation)


class HipCrossEntropy(nn.Module):
    def forward(self, logits, target):
        return hip_cross_entropy(logits, target)


class HipMSE(nn.Module):
    def forward(self, pred, target):
        return hip_mse(pred, target)


class MnistMLPFused(nn.Module):
    """784-256-256-10 MLP with fused linear+relu — the bench flagship.
    state_dict-compatible with models.mnist.MnistMLP."""

    def __init__(self, in_dim: int = 784, hidden: int = 256, classes: int = 10):
        super().__init__()
        self.fc1 = HipLinear(in_dim, hidden, activation="relu")
        self.fc2 = HipLinear(hidden, hidden, activation="relu")
        self.fc3 = HipLinear(hidden, classes, activation=None)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.is_cuda and x.dtype != torch.bfloat16:
            x = x.to(torch.bfloat16)
        return self.fc3(self.fc2(self.fc1(x)))


def convert_model_for_mi355x(model: nn.Module) -> nn.Module:
    """Swap nn.Linear -> HipLinear in place (same Parameters)."""
    for name, child in model.named_children():
        if isinstance(child, nn.Linear):
            setattr(model, name, HipLinear.from_linear(child))
        else:
            convert_model_for_mi355x(child)
    return model


def convert_criterion_for_mi355x(criterion):
    name = type(criterion).__name__
    if name == "CrossEntropyLoss":
        return HipCrossEntropy()
    if name == "MSELoss":
        return HipMSE()
    return criterion
