"""Numerics for the implicit-GEMM stride-1 conv vs plain fp32 torch.

Every op compares against an fp32 torch reference computed on the
bf16-quantized inputs the kernels actually see.  Shapes cover the ResNet
3x3 p1 layers, the MNIST-CNN 3x3 p0 CI=16 layer (K % 64 != 0 exercises the
tail stage), a 5x5 case, and partial M/N tiles.
"""

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():  # collected but skipped off-GPU
    pytest.skip("needs MI355X", allow_module_level=True)

from sparktorch_amd import ops
from sparktorch_amd.ops.functional import _ConvImplicitNHWCFn, hip_conv2d_nhwc

DEV = "cuda:0"


def bf(x):
    return x.to(torch.bfloat16)


def _nhwc(x_nchw):
    return x_nchw.permute(0, 2, 3, 1).contiguous()


# B, CI, CO, H, W, KH, pad
SHAPES = [
    (2, 64, 64, 56, 56, 3, 1),    # resnet l1
    (2, 128, 128, 28, 28, 3, 1),  # resnet l2
    (1, 64, 128, 14, 14, 3, 1),   # partial M tile
    (3, 64, 64, 7, 7, 3, 1),      # M = 147: clamp path
    (4, 16, 32, 24, 24, 3, 0),    # MNIST conv2 (K = 144: tail stage, p0)
    (2, 16, 32, 20, 20, 5, 2),    # 5x5 p2 (K = 400: tail)
    (2, 64, 64, 16, 16, 1, 0),    # 1x1 as implicit GEMM
]


def test_pad_nhwc_rings():
    x = torch.randn(2, 5, 6, 64, device=DEV).to(torch.bfloat16).contiguous()
    for P in (1, 2):
        xP = ops.ext().pad_nhwc(x, P)
        assert xP.shape == (2, 5 + 2 * P, 6 + 2 * P, 64)
        assert torch.equal(xP[:, P:-P, P:-P, :], x)
        assert xP[:, :P].abs().sum().item() == 0
        assert xP[:, :, -P:].abs().sum().item() == 0


def test_flip_w2d():
    CO, CI, KH = 64, 64, 3
    w = torch.randn(CO, CI, KH, KH, device=DEV)
    w2d = w.permute(0, 2, 3, 1).reshape(CO, KH * KH * CI).to(torch.bfloat16).contiguous()
    wf = ops.ext().flip_w2d(w2d, CI, KH * KH)
    wf4 = wf.view(CI, KH, KH, CO)
    w4 = w2d.view(CO, KH, KH, CI)
    for kh in range(KH):
        for kw in range(KH):
            assert torch.equal(
                wf4[:, kh, kw, :], w4[:, KH - 1 - kh, KH - 1 - kw, :].t().contiguous()
            )


@pytest.mark.parametrize("B,CI,CO,H,W,KH,pad", SHAPES)
def test_implicit_fwd_matches_torch(B, CI, CO, H, W, KH, pad):
    torch.manual_seed(0)
    x = torch.randn(B, CI, H, W, device=DEV)
    w = torch.randn(CO, CI, KH, KH, device=DEV) * 0.1
    y = hip_conv2d_nhwc(_nhwc(bf(x)), w, None, stride=(1, 1), padding=(pad, pad), relu=False)
    ref = F.conv2d(bf(x).float(), bf(w).float(), None, stride=1, padding=pad)
    ref = _nhwc(ref)
    assert y.shape == ref.shape
    err = (y.float() - ref).abs().max().item()
    scale = ref.abs().max().item() + 1e-6
    assert err / scale < 0.03, (err, scale)


def test_implicit_fwd_relu_bias():
    torch.manual_seed(1)
    B, CI, CO, H, W = 2, 64, 64, 14, 14
    x = torch.randn(B, CI, H, W, device=DEV)
    w = torch.randn(CO, CI, 3, 3, device=DEV) * 0.1
    b = torch.randn(CO, device=DEV)
    y = hip_conv2d_nhwc(_nhwc(bf(x)), w, b, stride=(1, 1), padding=(1, 1), relu=True)
    ref = F.relu(F.conv2d(bf(x).float(), bf(w).float(), b, stride=1, padding=1))
    err = (y.float() - _nhwc(ref)).abs().max().item()
    assert err < 0.05, err


@pytest.mark.parametrize("B,CI,CO,H,W,KH,pad", SHAPES[:3] + SHAPES[4:6])
def test_implicit_backward_matches_torch(B, CI, CO, H, W, KH, pad):
    torch.manual_seed(2)
    x32 = torch.randn(B, CI, H, W, device=DEV)
    w32 = torch.randn(CO, CI, KH, KH, device=DEV) * 0.1

    xh = _nhwc(bf(x32)).requires_grad_(True)
    wh = w32.clone().requires_grad_(True)
    y = _ConvImplicitNHWCFn.apply(xh, wh, None, (1, 1), (pad, pad), False)
    gy = torch.randn_like(y.float()) * 0.1
    y.backward(bf(gy))

    xr = bf(x32).float().requires_grad_(True)
    wr = bf(w32).float().requires_grad_(True)
    yr = F.conv2d(xr, wr, None, stride=1, padding=pad)
    # gy is NHWC (same layout as y); the torch reference is NCHW
    yr.backward(bf(gy).float().permute(0, 3, 1, 2))

    dw_err = (wh.grad - wr.grad).abs().max().item()
    dw_scale = wr.grad.abs().max().item() + 1e-6
    assert dw_err / dw_scale < 0.03, (dw_err, dw_scale)

    dx_ref = _nhwc(xr.grad)
    dx_err = (xh.grad.float() - dx_ref).abs().max().item()
    dx_scale = dx_ref.abs().max().item() + 1e-6
    assert dx_err / dx_scale < 0.03, (dx_err, dx_scale)


def test_implicit_wgrad_slab_matches_atomic():
    torch.manual_seed(3)
    B, CI, CO, H, W = 2, 64, 128, 28, 28
    x = _nhwc(torch.randn(B, CI, H, W, device=DEV).to(torch.bfloat16))
    dz = torch.randn(B * H * W, CO, device=DEV).to(torch.bfloat16).contiguous()
    xP = ops.ext().pad_nhwc(x, 1)
    a = ops.ext().conv_implicit_wgrad(dz, xP, 3, 3, 8, False)
    b = ops.ext().conv_implicit_wgrad(dz, xP, 3, 3, 8, True)
    assert torch.allclose(a, b, atol=1e-3, rtol=1e-4)


def test_resnet_block_uses_implicit_and_trains():
    """A resnet-style stack through SyncTrainer still trains (the module
    converter path picks the implicit conv for its 3x3 s1 convs)."""
    import torch.nn as nn

    from sparktorch_amd.parallel.sync import SyncTrainer

    torch.manual_seed(4)

    class Blocky(nn.Module):
        def __init__(self):
            super().__init__()
            self.conv1 = nn.Conv2d(64, 64, 3, padding=1, bias=False)
            self.bn1 = nn.BatchNorm2d(64)
            self.conv2 = nn.Conv2d(64, 64, 3, padding=1, bias=False)
            self.bn2 = nn.BatchNorm2d(64)
            self.fc = nn.Linear(64, 10)

        def forward(self, x):
            x = x.view(-1, 64, 8, 8)
            x = torch.relu(self.bn1(self.conv1(x)))
            x = torch.relu(self.bn2(self.conv2(x)))
            x = x.mean(dim=(2, 3))
            return self.fc(x)

    model = Blocky()
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    tr = SyncTrainer(model, nn.CrossEntropyLoss(), opt, device=DEV, world_size=1)
    x = torch.randn(256, 64 * 8 * 8, device=DEV).to(torch.bfloat16)
    yl = torch.randint(0, 10, (256,), device=DEV)
    losses = [tr.train_step(x, yl) for _ in range(8)]
    assert losses[-1] < losses[0]


def test_mnist_cnn_fused_trains():
    """The fused MNIST CNN (conv2 now on the implicit path) still trains."""
    import torch.nn as nn

    from sparktorch_amd.ops.modules import MnistCNNFused
    from sparktorch_amd.parallel.sync import SyncTrainer

    torch.manual_seed(5)
    model = MnistCNNFused()
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    tr = SyncTrainer(model, nn.CrossEntropyLoss(), opt, device=DEV, world_size=1)
    x = torch.randn(4096, 784, device=DEV).to(torch.bfloat16)
    yl = torch.randint(0, 10, (4096,), device=DEV)
    losses = [tr.train_step(x, yl) for _ in range(8)]
    assert losses[-1] < losses[0]


def test_stem_s2d_matches_torch():
    """7x7 s2 p3 CI=3 stem via space-to-depth vs fp32 torch (fwd + wgrad)."""
    from sparktorch_amd.ops.functional import _ConvStemS2DFn

    torch.manual_seed(7)
    B, H = 2, 32  # any even spatial works; bench uses 224
    x32 = torch.randn(B, 3, H, H, device=DEV)
    w32 = torch.randn(64, 3, 7, 7, device=DEV) * 0.1

    xh = _nhwc(bf(x32))
    wh = w32.clone().requires_grad_(True)
    y = _ConvStemS2DFn.apply(xh, wh, None, False)
    ref = F.conv2d(bf(x32).float(), bf(w32).float(), None, stride=2, padding=3)
    ref = _nhwc(ref)
    assert y.shape == ref.shape
    err = (y.float() - ref).abs().max().item()
    scale = ref.abs().max().item() + 1e-6
    assert err / scale < 0.03, (err, scale)

    gy = torch.randn_like(y.float()) * 0.1
    y.backward(bf(gy))
    wr = bf(w32).float().requires_grad_(True)
    yr = F.conv2d(bf(x32).float(), wr, None, stride=2, padding=3)
    yr.backward(bf(gy).float().permute(0, 3, 1, 2))
    dw_err = (wh.grad - wr.grad).abs().max().item()
    dw_scale = wr.grad.abs().max().item() + 1e-6
    assert dw_err / dw_scale < 0.03, (dw_err, dw_scale)


@pytest.mark.parametrize("B,CI,CO,H,W,KH,pad", [
    (2, 64, 128, 28, 28, 3, 1),   # resnet l2 first conv (s2 3x3)
    (2, 64, 128, 28, 28, 1, 0),   # resnet downsample (s2 1x1)
    (2, 16, 32, 24, 24, 3, 1),    # s2 with K tail
])
def test_implicit_s2_fwd_bwd_matches_torch(B, CI, CO, H, W, KH, pad):
    torch.manual_seed(9)
    x32 = torch.randn(B, CI, H, W, device=DEV)
    w32 = torch.randn(CO, CI, KH, KH, device=DEV) * 0.1

    xh = _nhwc(bf(x32)).requires_grad_(True)
    wh = w32.clone().requires_grad_(True)
    y = _ConvImplicitNHWCFn.apply(xh, wh, None, (2, 2), (pad, pad), False)
    xr = bf(x32).float().requires_grad_(True)
    wr = bf(w32).float().requires_grad_(True)
    yr = F.conv2d(xr, wr, None, stride=2, padding=pad)
    assert y.shape == _nhwc(yr).shape
    err = (y.float() - _nhwc(yr)).abs().max().item()
    scale = yr.abs().max().item() + 1e-6
    assert err / scale < 0.03, (err, scale)

    gy = torch.randn_like(y.float()) * 0.1
    y.backward(bf(gy))
    yr.backward(bf(gy).float().permute(0, 3, 1, 2))
    dw_err = (wh.grad - wr.grad).abs().max().item()
    assert dw_err / (wr.grad.abs().max().item() + 1e-6) < 0.03, dw_err
    dx_ref = _nhwc(xr.grad)
    dx_err = (xh.grad.float() - dx_ref).abs().max().item()
    assert dx_err / (dx_ref.abs().max().item() + 1e-6) < 0.03, dx_err
