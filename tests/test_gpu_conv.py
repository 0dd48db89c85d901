"""Numerics for the conv-family kernels vs torch fp32 references."""

import numpy as np
import pytest
import torch
import torch.nn as nn
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs MI355X", allow_module_level=True)

from sparktorch_amd import ops

DEV = "cuda:0"


def bf(x):
    return x.to(torch.bfloat16)


def ref_of(x):
    return bf(x).float()


@pytest.mark.parametrize("B,CI,H,W,KH,s,p", [(4, 3, 12, 12, 3, 1, 0), (2, 1, 28, 28, 5, 1, 0),
                                             (3, 4, 16, 16, 3, 2, 1)])
def test_im2col_matches_unfold(B, CI, H, W, KH, s, p):
    torch.manual_seed(0)
    x = torch.randn(B, CI, H, W, device=DEV)
    col = ops.ext().im2col(bf(x).contiguous(), KH, KH, s, s, p, p)
    ref = F.unfold(ref_of(x), KH, padding=p, stride=s)  # [B, K, L]
    ref = ref.permute(0, 2, 1).reshape(col.shape)
    assert torch.allclose(col.float(), ref, atol=1e-2)


@pytest.mark.parametrize("relu", [False, True])
def test_conv2d_fwd_bwd_vs_torch(relu):
    from sparktorch_amd.ops.functional import hip_conv2d

    torch.manual_seed(1)
    B, CI, CO, H, W, K = 8, 3, 16, 14, 14, 3
    x = torch.randn(B, CI, H, W, device=DEV)
    w = (torch.randn(CO, CI, K, K, device=DEV) * 0.1)
    b = torch.randn(CO, device=DEV)

    xb = bf(x).requires_grad_(True)
    wr = w.clone().requires_grad_(True)
    br = b.clone().requires_grad_(True)
    y = hip_conv2d(xb, wr, br, (1, 1), (0, 0), relu=relu)
    gout = torch.randn_like(y.float())
    y.backward(bf(gout))

    x2 = ref_of(x).requires_grad_(True)
    w2 = w.clone().requires_grad_(True)
    b2 = b.clone().requires_grad_(True)
    y2 = F.conv2d(x2, bf(w2).float(), b2)
    if relu:
        y2 = F.relu(y2)
    y2.backward(bf(gout).float())

    s = y2.abs().max().item() + 1e-6
    assert (y.float() - y2).abs().max().item() / s < 0.03
    assert (wr.grad - w2.grad).abs().max().item() / (w2.grad.abs().max() + 1e-6) < 0.03
    assert (br.grad - b2.grad).abs().max().item() / (b2.grad.abs().max() + 1e-6) < 0.03
    assert (xb.grad.float() - x2.grad).abs().max().item() / (x2.grad.abs().max() + 1e-6) < 0.03


def test_maxpool_fwd_bwd():
    from sparktorch_amd.ops.functional import hip_max_pool2d

    torch.manual_seed(2)
    x = torch.randn(4, 8, 22, 22, device=DEV)
    xb = bf(x).requires_grad_(True)
    y = hip_max_pool2d(xb, 2)
    gout = torch.randn_like(y.float())
    y.backward(bf(gout))

    x2 = ref_of(x).requires_grad_(True)
    y2 = F.max_pool2d(x2, 2)
    y2.backward(bf(gout).float())
    assert torch.allclose(y.float(), y2, atol=1e-2)
    assert torch.allclose(xb.grad.float(), x2.grad, atol=1e-2)


def test_dropout_statistics_and_backward_consistency():
    from sparktorch_amd.ops.functional import _DropoutFn

    torch.manual_seed(3)
    x = torch.ones(100_000, device=DEV, dtype=torch.bfloat16)
    p = 0.25
    out = _DropoutFn.apply(x, p, 12345, 1, 0)
    kept = (out != 0).float().mean().item()
    assert abs(kept - 0.75) < 0.02
    # kept values scaled by 1/(1-p)
    assert abs(out.float().max().item() - 1.0 / 0.75) < 0.01
    # backward must reproduce the SAME mask
    out2 = _DropoutFn.apply(x, p, 12345, 1, 0)
    assert torch.equal(out, out2)


def test_dropout2d_channelwise():
    from sparktorch_amd.ops.functional import _DropoutFn

    x = torch.ones(8, 64, 11, 11, device=DEV, dtype=torch.bfloat16)
    out = _DropoutFn.apply(x, 0.5, 99, 11 * 11, 0)
    per_channel = out.float().sum(dim=(2, 3)).flatten()
    # each channel entirely zero or entirely kept
    assert ((per_channel == 0) | (per_channel > 100)).all()


def test_dropout2d_channelwise_nhwc():
    from sparktorch_amd.ops.functional import _DropoutFn

    x = torch.ones(8, 11, 11, 64, device=DEV, dtype=torch.bfloat16)
    out = _DropoutFn.apply(x, 0.5, 99, 11 * 11 * 64, 64)
    per_channel = out.float().sum(dim=(1, 2)).flatten()  # NHWC: sum over H,W
    # each (b, c) unit entirely zero or entirely kept
    assert ((per_channel == 0) | (per_channel > 100)).all()


def test_cnn_fused_end_to_end_train():
    from sparktorch_amd.ops.modules import MnistCNNFused
    from sparktorch_amd.parallel.sync import SyncTrainer

    torch.manual_seed(4)
    model = MnistCNNFused()
    trainer = SyncTrainer(
        model, nn.CrossEntropyLoss(), torch.optim.Adam(model.parameters(), lr=1e-3),
        device=DEV, world_size=1,
    )
    x = torch.randn(512, 784, device=DEV, dtype=torch.bfloat16)
    y = torch.randint(0, 10, (512,), device=DEV)
    losses = [trainer.train_step(x, y) for _ in range(15)]
    assert np.isfinite(losses).all()
    assert min(losses[-5:]) < losses[0]


def test_converted_reference_cnn_runs_native():
    """The reference CNN (eager modules) through the converter trains on the
    native path."""
    from sparktorch_amd.models.mnist import MnistCNN
    from sparktorch_amd.ops.modules import HipConv2d, HipLinear, convert_model_for_mi355x
    from sparktorch_amd.parallel.sync import SyncTrainer

    model = MnistCNN()
    trainer = SyncTrainer(
        model, nn.CrossEntropyLoss(), torch.optim.Adam(model.parameters(), lr=1e-3),
        device=DEV, world_size=1,
    )
    assert isinstance(trainer.model.conv1, HipConv2d)
    assert isinstance(trainer.model.fc, HipLinear)
    x = torch.randn(256, 784, device=DEV, dtype=torch.bfloat16)
    y = torch.randint(0, 10, (256,), device=DEV)
    l0 = trainer.train_step(x, y)
    assert np.isfinite(l0)
