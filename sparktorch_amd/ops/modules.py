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

from sparktorch_amd.ops.functional import (
    hip_add_relu,
    hip_batch_norm2d,
    hip_batch_norm2d_nhwc,
    hip_conv2d,
    hip_conv2d_nhwc,
    hip_cross_entropy,
    hip_dropout,
    hip_global_avg_pool,
    hip_global_avg_pool_nhwc,
    hip_linear,
    hip_max_pool2d,
    hip_max_pool2d_nhwc,
    hip_relu,
    hip_mse,
)


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


class HipConv2d(nn.Module):
    """Drop-in nn.Conv2d on the implicit-GEMM MFMA path (stride/padding;
    dilation=1, groups=1), optional fused ReLU."""

    def __init__(self, in_ch: int, out_ch: int, kernel_size: int, stride: int = 1,
                 padding: int = 0, bias: bool = True, activation: Optional[str] = None,
                 layout: str = "nchw"):
        super().__init__()
        ks = (kernel_size, kernel_size) if isinstance(kernel_size, int) else kernel_size
        self.stride = (stride, stride) if isinstance(stride, int) else stride
        self.padding = (padding, padding) if isinstance(padding, int) else padding
        self.weight = nn.Parameter(torch.empty(out_ch, in_ch, *ks))
        self.bias = nn.Parameter(torch.zeros(out_ch)) if bias else None
        self.activation = activation
        self.layout = layout
        nn.init.kaiming_uniform_(self.weight, a=5 ** 0.5)

    @classmethod
    def from_conv2d(cls, conv: nn.Conv2d, activation: Optional[str] = None) -> "HipConv2d":
        if (conv.dilation != (1, 1)) or (conv.groups != 1) or conv.padding_mode != "zeros":
            raise ValueError("HipConv2d supports dilation=1, groups=1, zero padding")
        m = cls.__new__(cls)
        nn.Module.__init__(m)
        m.stride = conv.stride
        m.padding = conv.padding
        m.weight = conv.weight  # SAME Parameter objects
        m.bias = conv.bias
        m.activation = activation
        m.layout = "nchw"
        return m

    def forward(self, x):
        fn = hip_conv2d_nhwc if self.layout == "nhwc" else hip_conv2d
        return fn(
            x, self.weight, self.bias, self.stride, self.padding, relu=self.activation == "relu"
        )


class HipMaxPool2d(nn.Module):
    def __init__(self, kernel_size: int, stride: Optional[int] = None, padding: int = 0,
                 layout: str = "nchw"):
        super().__init__()
        self.kernel_size = kernel_size
        self.stride = kernel_size if stride is None else stride
        self.padding = padding
        self.layout = layout

    def forward(self, x):
        fn = hip_max_pool2d_nhwc if self.layout == "nhwc" else hip_max_pool2d
        return fn(x, self.kernel_size, self.stride, self.padding)


class HipBatchNorm2d(nn.Module):
    """Drop-in nn.BatchNorm2d on the native kernels (affine, running stats),
    with optional fused ReLU.  state_dict-compatible with nn.BatchNorm2d."""

    def __init__(self, num_features: int, eps: float = 1e-5, momentum: float = 0.1,
                 activation: Optional[str] = None, layout: str = "nchw"):
        super().__init__()
        self.num_features = num_features
        self.eps = eps
        self.momentum = momentum
        self.activation = activation
        self.layout = layout
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.register_buffer("num_batches_tracked", torch.tensor(0, dtype=torch.long))

    @classmethod
    def from_batchnorm(cls, bn: nn.BatchNorm2d, activation: Optional[str] = None) -> "HipBatchNorm2d":
        if not bn.affine or not bn.track_running_stats:
            raise ValueError("HipBatchNorm2d needs affine=True, track_running_stats=True")
        m = cls.__new__(cls)
        nn.Module.__init__(m)
        m.num_features = bn.num_features
        m.eps = bn.eps
        m.momentum = bn.momentum if bn.momentum is not None else 0.1
        m.activation = activation
        m.layout = "nchw"
        m.weight = bn.weight  # SAME Parameter objects
        m.bias = bn.bias
        m.register_buffer("running_mean", bn.running_mean)
        m.register_buffer("running_var", bn.running_var)
        m.register_buffer("num_batches_tracked", bn.num_batches_tracked)
        return m

    def forward(self, x):
        if self.training:
            self.num_batches_tracked += 1
        fn = hip_batch_norm2d_nhwc if self.layout == "nhwc" else hip_batch_norm2d
        return fn(
            x, self.weight, self.bias, self.running_mean, self.running_var,
            self.training, self.momentum, self.eps, relu=self.activation == "relu",
        )

    def extra_repr(self) -> str:
        return "%d, eps=%g, act=%s" % (self.num_features, self.eps, self.activation)


class HipGlobalAvgPool(nn.Module):
    """Global average pool as one reduction kernel.

    ``keepdim=False`` (default) fuses the usual
    ``adaptive_avg_pool2d(x, 1).flatten(1)`` idiom and returns [B, C];
    ``keepdim=True`` preserves AdaptiveAvgPool2d(1)'s [B, C, 1, 1] shape for
    models that consume the 4D output (SE blocks, later convs, views)."""

    def __init__(self, layout: str = "nchw", keepdim: bool = False):
        super().__init__()
        self.layout = layout
        self.keepdim = keepdim

    def forward(self, x):
        fn = hip_global_avg_pool_nhwc if self.layout == "nhwc" else hip_global_avg_pool
        y = fn(x)
        if self.keepdim:
            return y.view(y.shape[0], y.shape[1], 1, 1)
        return y


class HipDropout(nn.Module):
    def __init__(self, p: float = 0.5, channel_wise: bool = False, layout: str = "nchw"):
        super().__init__()
        self.p = p
        self.channel_wise = channel_wise
        self.layout = layout

    def forward(self, x):
        return hip_dropout(x, self.p, training=self.training,
                           channel_wise=self.channel_wise, layout=self.layout)


class MnistCNNFused(nn.Module):
    """The reference example CNN (examples/cnn_network.py:6-24) on the fully
    native path, channels-last end to end: NHWC implicit-GEMM convs (conv1
    with fused ReLU; conv2's ReLU runs after the pool — see forward), NHWC
    maxpool, counter-based NHWC Dropout2d.  The fc layer
    expects the reference's NCHW flatten order, so its (tiny) weight is
    viewed in NHWC order per forward instead of permuting the (huge)
    activation.  state_dict-compatible with models.mnist.MnistCNN."""

    def __init__(self):
        super().__init__()
        self.conv1 = HipConv2d(1, 16, kernel_size=5, activation="relu", layout="nhwc")
        # conv2's ReLU commutes with the monotone maxpool: applying it AFTER
        # the pool (on the 4x smaller tensor) gives bit-equal activations
        # while the backward mask pass shrinks 4x
        self.conv2 = HipConv2d(16, 32, kernel_size=3, layout="nhwc")
        self.dropout = HipDropout(p=0.25, channel_wise=True, layout="nhwc")
        self.fc = HipLinear(3872, 10)

    def forward(self, x):
        if x.is_cuda and x.dtype != torch.bfloat16:
            x = x.to(torch.bfloat16)
        x = x.view(-1, 1, 28, 28).permute(0, 2, 3, 1).contiguous()  # NHWC entry
        x = self.conv1(x)
        x = self.conv2(x)
        x = hip_relu(hip_max_pool2d_nhwc(x, 2))
        x = self.dropout(x)
        B, H, W, C = x.shape
        x = x.reshape(B, H * W * C)
        # fc weight reordered [10, C*H*W] -> [10, H*W*C] to match NHWC flatten
        w = self.fc.weight.view(-1, C, H, W).permute(0, 2, 3, 1).reshape(-1, H * W * C)
        return hip_linear(x, w.contiguous(), self.fc.bias)


class FusedBasicBlock(nn.Module):
    """ResNet basic block on the native path: implicit-GEMM convs, BN with
    fused ReLU, and a fused residual add+relu join.  state_dict-compatible
    with models.resnet.BasicBlock."""

    def __init__(self, in_ch: int, out_ch: int, stride: int = 1, layout: str = "nhwc"):
        super().__init__()
        self.conv1 = HipConv2d(in_ch, out_ch, 3, stride=stride, padding=1, bias=False,
                               layout=layout)
        self.bn1 = HipBatchNorm2d(out_ch, activation="relu", layout=layout)
        self.conv2 = HipConv2d(out_ch, out_ch, 3, stride=1, padding=1, bias=False,
                               layout=layout)
        self.bn2 = HipBatchNorm2d(out_ch, layout=layout)
        self.down = None
        if stride != 1 or in_ch != out_ch:
            self.down = nn.Sequential(
                HipConv2d(in_ch, out_ch, 1, stride=stride, bias=False, layout=layout),
                HipBatchNorm2d(out_ch, layout=layout),
            )

    def forward(self, x):
        identity = x if self.down is None else self.down(x)
        out = self.bn1(self.conv1(x))          # BN+ReLU fused
        out = self.conv2(out)
        out = self.bn2(out)
        return hip_add_relu(out, identity)     # residual join fused


class ResNet18Fused(nn.Module):
    """ResNet-18 (BASELINE config 4: synthetic 3x224x224 Vector rows) fully
    on the hand-written CDNA4 kernels, channels-last (NHWC) end to end:
    7x7/s2 stem conv, overlapping 3x3/s2/p1 maxpool, 8 basic blocks,
    global-avg-pool reduction, MFMA fc.  NHWC makes every im2col/col2im run
    CI-contiguous (shortx8 vector moves) and the implicit-GEMM output
    [B*HO*WO, CO] IS the activation — no layout permutes anywhere (the NCHW
    gathers were 54%% of the step in the first rocprof capture, profiles/).
    state_dict-compatible with models.resnet.ResNet18 (weights [CO,CI,KH,KW])."""

    def __init__(self, num_classes: int = 1000, in_ch: int = 3, layout: str = "nhwc"):
        super().__init__()
        self.layout = layout
        self.conv1 = HipConv2d(in_ch, 64, 7, stride=2, padding=3, bias=False, layout=layout)
        self.bn1 = HipBatchNorm2d(64, activation="relu", layout=layout)
        self.maxpool = HipMaxPool2d(3, stride=2, padding=1, layout=layout)
        layers = []
        cfg = [(64, 1), (128, 2), (256, 2), (512, 2)]
        ch = 64
        for out_ch, stride in cfg:
            layers.append(FusedBasicBlock(ch, out_ch, stride, layout=layout))
            layers.append(FusedBasicBlock(out_ch, out_ch, 1, layout=layout))
            ch = out_ch
        self.layers = nn.Sequential(*layers)
        self.gap = HipGlobalAvgPool(layout=layout)
        self.fc = HipLinear(512, num_classes)

    def forward(self, x):
        if x.is_cuda and x.dtype != torch.bfloat16:
            x = x.to(torch.bfloat16)
        if x.dim() == 2:  # flattened Vector rows, like the CNN unflatten idiom
            x = x.view(-1, 3, 224, 224)
        if self.layout == "nhwc":  # one entry permute; everything after is NHWC
            x = x.permute(0, 2, 3, 1).contiguous()
        x = self.bn1(self.conv1(x))
        x = self.maxpool(x)
        x = self.layers(x)
        return self.fc(self.gap(x))


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
    """Swap torch modules for their native-kernel equivalents in place
    (adopting the SAME Parameter objects, so optimizers/buckets are
    unaffected).  Modules with unsupported configs are left as-is (they run
    through torch-ROCm eager, e.g. grouped/dilated convs)."""
    for name, child in model.named_children():
        cname = type(child).__name__
        if cname in ("Linear",) or isinstance(child, nn.Linear):
            setattr(model, name, HipLinear.from_linear(child))
        elif cname in ("Conv2d",) or isinstance(child, nn.Conv2d):
            try:
                setattr(model, name, HipConv2d.from_conv2d(child))
            except (ValueError, AttributeError):
                pass
        elif cname == "MaxPool2d" or isinstance(child, nn.MaxPool2d):
            ks, st, pd = child.kernel_size, child.stride, child.padding
            if isinstance(ks, int) and isinstance(st, (int, type(None))) and isinstance(pd, int) \
                    and child.dilation == 1 and not child.ceil_mode:
                setattr(model, name, HipMaxPool2d(ks, stride=st, padding=pd))
        elif cname == "BatchNorm2d" or isinstance(child, nn.BatchNorm2d):
            try:
                setattr(model, name, HipBatchNorm2d.from_batchnorm(child))
            except (ValueError, AttributeError):
                pass
        elif cname == "AdaptiveAvgPool2d" and child.output_size in (1, (1, 1)):
            # keepdim: the generic converter must preserve AdaptiveAvgPool2d's
            # [B,C,1,1] output rank — fused models opt into the flattened
            # variant explicitly
            setattr(model, name, HipGlobalAvgPool(keepdim=True))
        elif cname == "Dropout2d":
            setattr(model, name, HipDropout(p=child.p, channel_wise=True))
        elif cname == "Dropout":
            setattr(model, name, HipDropout(p=child.p, channel_wise=False))
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
