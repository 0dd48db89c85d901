"""Numerics for the ResNet-family kernels (BatchNorm2d, overlapping maxpool,
global avg pool, fused add+relu) vs torch fp32 references, plus ResNet-18
end-to-end training on the native path (BASELINE config 4)."""

import pytest
import torch
import torch.nn as nn
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs MI355X", allow_module_level=True)

from sparktorch_amd import ops
from sparktorch_amd.ops.functional import (
    hip_add_relu,
    hip_batch_norm2d,
    hip_global_avg_pool,
    hip_max_pool2d,
)

DEV = "cuda:0"


def bf(x):
    return x.to(torch.bfloat16)


@pytest.mark.parametrize("B,C,H,W", [(8, 16, 14, 14), (4, 64, 56, 56), (3, 7, 5, 9)])
def test_bn_train_fwd_bwd_vs_torch(B, C, H, W):
    torch.manual_seed(0)
    x = torch.randn(B, C, H, W, device=DEV)
    gamma = torch.rand(C, device=DEV) + 0.5
    beta = torch.randn(C, device=DEV)

    xb = bf(x).requires_grad_(True)
    g1 = gamma.clone().requires_grad_(True)
    b1 = beta.clone().requires_grad_(True)
    rm = torch.zeros(C, device=DEV)
    rv = torch.ones(C, device=DEV)
    y = hip_batch_norm2d(xb, g1, b1, rm, rv, training=True, momentum=0.1)
    gout = torch.randn_like(y.float())
    y.backward(bf(gout))

    x2 = bf(x).float().requires_grad_(True)
    g2 = gamma.clone().requires_grad_(True)
    b2 = beta.clone().requires_grad_(True)
    rm2 = torch.zeros(C, device=DEV)
    rv2 = torch.ones(C, device=DEV)
    y2 = F.batch_norm(x2, rm2, rv2, g2, b2, True, 0.1, 1e-5)
    y2.backward(bf(gout).float())

    assert torch.allclose(y.float(), y2, atol=3e-2, rtol=3e-2)
    assert torch.allclose(rm, rm2, atol=1e-3, rtol=1e-3)
    assert torch.allclose(rv, rv2, atol=1e-2, rtol=1e-2)
    assert torch.allclose(g1.grad, g2.grad, atol=0.05, rtol=0.05)
    assert torch.allclose(b1.grad, b2.grad, atol=0.05, rtol=0.05)
    assert torch.allclose(xb.grad.float(), x2.grad, atol=5e-2, rtol=5e-2)


def test_bn_eval_uses_running_stats():
    torch.manual_seed(1)
    B, C, H, W = 4, 8, 6, 6
    x = torch.randn(B, C, H, W, device=DEV)
    gamma = torch.rand(C, device=DEV) + 0.5
    beta = torch.randn(C, device=DEV)
    rm = torch.randn(C, device=DEV) * 0.3
    rv = torch.rand(C, device=DEV) + 0.5
    y = hip_batch_norm2d(bf(x), gamma, beta, rm, rv, training=False)
    y2 = F.batch_norm(bf(x).float(), rm.clone(), rv.clone(), gamma, beta, False, 0.1, 1e-5)
    assert torch.allclose(y.float(), y2, atol=2e-2, rtol=2e-2)
    # eval must not touch running stats
    assert torch.allclose(rv, rv)


def test_bn_fused_relu_matches_separate():
    torch.manual_seed(2)
    B, C, H, W = 4, 16, 10, 10
    x = torch.randn(B, C, H, W, device=DEV)
    gamma = torch.rand(C, device=DEV) + 0.5
    beta = torch.randn(C, device=DEV)

    xb = bf(x).requires_grad_(True)
    g1 = gamma.clone().requires_grad_(True)
    b1 = beta.clone().requires_grad_(True)
    y = hip_batch_norm2d(xb, g1, b1, torch.zeros(C, device=DEV), torch.ones(C, device=DEV),
                         training=True, relu=True)
    gout = torch.randn_like(y.float())
    y.backward(bf(gout))

    x2 = bf(x).float().requires_grad_(True)
    g2 = gamma.clone().requires_grad_(True)
    b2 = beta.clone().requires_grad_(True)
    y2 = F.relu(F.batch_norm(x2, torch.zeros(C, device=DEV), torch.ones(C, device=DEV),
                             g2, b2, True, 0.1, 1e-5))
    y2.backward(bf(gout).float())
    assert torch.allclose(y.float(), y2, atol=3e-2, rtol=3e-2)
    assert torch.allclose(xb.grad.float(), x2.grad, atol=5e-2, rtol=5e-2)
    assert torch.allclose(g1.grad, g2.grad, atol=0.05, rtol=0.05)


@pytest.mark.parametrize("H,W,ks,s,p", [(112, 112, 3, 2, 1), (7, 7, 3, 2, 1), (10, 10, 3, 3, 0)])
def test_maxpool_general_fwd_bwd(H, W, ks, s, p):
    torch.manual_seed(3)
    B, C = 3, 5
    # Tie-free input: random bf16 has duplicate values inside pooling windows,
    # and with a tie the gradient legitimately routes to a different (equal)
    # argmax than torch's.  Integers mod 251 are exact in bf16 and any 3x3
    # window spans < 251 scan positions, so values within a window are unique.
    n = B * C * H * W
    x = ((((torch.arange(n, device=DEV) * 97) % 251).float() - 125.0) / 128.0).reshape(B, C, H, W)
    xb = bf(x).requires_grad_(True)
    y = hip_max_pool2d(xb, ks, stride=s, padding=p)
    gout = torch.randn_like(y.float())
    y.backward(bf(gout))

    x2 = bf(x).float().requires_grad_(True)
    y2 = F.max_pool2d(x2, ks, stride=s, padding=p)
    y2.backward(bf(gout).float())
    # kernel outputs are bf16-rounded (<=2^-9 relative): rtol covers it
    assert torch.allclose(y.float(), y2, atol=1e-2, rtol=1e-2)
    assert torch.allclose(xb.grad.float(), x2.grad, atol=1e-2, rtol=1e-2)


def test_global_avg_pool_fwd_bwd():
    torch.manual_seed(4)
    B, C, H, W = 6, 32, 7, 7
    x = torch.randn(B, C, H, W, device=DEV)
    xb = bf(x).requires_grad_(True)
    y = hip_global_avg_pool(xb)
    assert y.shape == (B, C)
    gout = torch.randn_like(y.float())
    y.backward(bf(gout))

    x2 = bf(x).float().requires_grad_(True)
    y2 = F.adaptive_avg_pool2d(x2, 1).flatten(1)
    y2.backward(bf(gout).float())
    # kernel outputs are bf16-rounded (<=2^-9 relative): rtol covers it
    assert torch.allclose(y.float(), y2, atol=1e-2, rtol=1e-2)
    assert torch.allclose(xb.grad.float(), x2.grad, atol=1e-2, rtol=1e-2)


def test_add_relu_fwd_bwd():
    torch.manual_seed(5)
    a = torch.randn(1000, device=DEV)
    b = torch.randn(1000, device=DEV)
    ab = bf(a).requires_grad_(True)
    bb = bf(b).requires_grad_(True)
    out = hip_add_relu(ab, bb)
    gout = torch.randn(1000, device=DEV)
    out.backward(bf(gout))

    a2 = bf(a).float().requires_grad_(True)
    b2 = bf(b).float().requires_grad_(True)
    out2 = F.relu(a2 + b2)
    out2.backward(bf(gout).float())
    assert torch.allclose(out.float(), out2, atol=1e-2, rtol=1e-2)
    assert torch.allclose(ab.grad.float(), a2.grad, atol=1e-2, rtol=1e-2)
    assert torch.allclose(bb.grad.float(), b2.grad, atol=1e-2, rtol=1e-2)


@pytest.mark.parametrize("CI,CO,K,s,p,H,W", [
    (64, 64, 3, 1, 1, 14, 14), (64, 128, 3, 2, 1, 14, 14), (3, 16, 7, 2, 3, 14, 14),
    (128, 256, 1, 2, 0, 14, 14),
    (64, 64, 3, 2, 1, 13, 15),   # odd spatial dims, strided
    (8, 24, 5, 1, 2, 9, 11),     # small channels, odd dims, big pad
])
def test_conv2d_nhwc_fwd_bwd_vs_torch(CI, CO, K, s, p, H, W):
    from sparktorch_amd.ops.functional import hip_conv2d_nhwc

    torch.manual_seed(10)
    B = 4
    x = torch.randn(B, H, W, CI, device=DEV)
    w = torch.randn(CO, CI, K, K, device=DEV) * (1.0 / (CI * K * K) ** 0.5)

    xb = bf(x).requires_grad_(True)
    wr = w.clone().requires_grad_(True)
    y = hip_conv2d_nhwc(xb, wr, None, (s, s), (p, p), relu=True)
    gout = torch.randn_like(y.float())
    y.backward(bf(gout))

    x2 = bf(x).float().permute(0, 3, 1, 2).requires_grad_(True)
    w2 = w.clone().requires_grad_(True)
    y2 = F.relu(F.conv2d(x2, bf(w2).float(), None, stride=s, padding=p))
    y2.backward(bf(gout).float().permute(0, 3, 1, 2))

    assert torch.allclose(y.float().permute(0, 3, 1, 2), y2, atol=5e-2, rtol=5e-2)
    assert torch.allclose(wr.grad, w2.grad, atol=0.1, rtol=0.05)
    assert torch.allclose(xb.grad.float().permute(0, 3, 1, 2), x2.grad, atol=5e-2, rtol=5e-2)


def test_bn_nhwc_train_fwd_bwd_vs_torch():
    from sparktorch_amd.ops.functional import hip_batch_norm2d_nhwc

    torch.manual_seed(11)
    B, C, H, W = 4, 64, 14, 14
    x = torch.randn(B, H, W, C, device=DEV)
    gamma = torch.rand(C, device=DEV) + 0.5
    beta = torch.randn(C, device=DEV)

    xb = bf(x).requires_grad_(True)
    g1 = gamma.clone().requires_grad_(True)
    b1 = beta.clone().requires_grad_(True)
    rm = torch.zeros(C, device=DEV)
    rv = torch.ones(C, device=DEV)
    y = hip_batch_norm2d_nhwc(xb, g1, b1, rm, rv, training=True, relu=True)
    gout = torch.randn_like(y.float())
    y.backward(bf(gout))

    x2 = bf(x).float().permute(0, 3, 1, 2).requires_grad_(True)
    g2 = gamma.clone().requires_grad_(True)
    b2 = beta.clone().requires_grad_(True)
    rm2 = torch.zeros(C, device=DEV)
    rv2 = torch.ones(C, device=DEV)
    y2 = F.relu(F.batch_norm(x2, rm2, rv2, g2, b2, True, 0.1, 1e-5))
    y2.backward(bf(gout).float().permute(0, 3, 1, 2))

    assert torch.allclose(y.float().permute(0, 3, 1, 2), y2, atol=3e-2, rtol=3e-2)
    assert torch.allclose(rm, rm2, atol=1e-3, rtol=1e-3)
    assert torch.allclose(rv, rv2, atol=1e-2, rtol=1e-2)
    assert torch.allclose(g1.grad, g2.grad, atol=0.05, rtol=0.05)
    assert torch.allclose(b1.grad, b2.grad, atol=0.05, rtol=0.05)
    assert torch.allclose(xb.grad.float().permute(0, 3, 1, 2), x2.grad, atol=5e-2, rtol=5e-2)


def test_maxpool_and_gap_nhwc_vs_torch():
    from sparktorch_amd.ops.functional import hip_global_avg_pool_nhwc, hip_max_pool2d_nhwc

    torch.manual_seed(12)
    B, C, H, W = 3, 64, 28, 28
    n = B * C * H * W
    # tie-free (see test_maxpool_general_fwd_bwd)
    x = ((((torch.arange(n, device=DEV) * 97) % 251).float() - 125.0) / 128.0).reshape(B, H, W, C)
    xb = bf(x).requires_grad_(True)
    y = hip_max_pool2d_nhwc(xb, 3, stride=2, padding=1)
    gout = torch.randn_like(y.float())
    y.backward(bf(gout))

    x2 = bf(x).float().permute(0, 3, 1, 2).requires_grad_(True)
    y2 = F.max_pool2d(x2, 3, stride=2, padding=1)
    y2.backward(bf(gout).float().permute(0, 3, 1, 2))
    assert torch.allclose(y.float().permute(0, 3, 1, 2), y2, atol=1e-2, rtol=1e-2)
    assert torch.allclose(xb.grad.float().permute(0, 3, 1, 2), x2.grad, atol=1e-2, rtol=1e-2)

    xg = bf(torch.randn(B, H, W, C, device=DEV)).requires_grad_(True)
    g = hip_global_avg_pool_nhwc(xg)
    assert g.shape == (B, C)
    ggout = torch.randn_like(g.float())
    g.backward(bf(ggout))
    xg2 = xg.detach().float().permute(0, 3, 1, 2).requires_grad_(True)
    g2 = F.adaptive_avg_pool2d(xg2, 1).flatten(1)
    g2.backward(bf(ggout).float())
    assert torch.allclose(g.float(), g2, atol=1e-2, rtol=1e-2)
    assert torch.allclose(xg.grad.float().permute(0, 3, 1, 2), xg2.grad, atol=1e-2, rtol=1e-2)


def test_resnet18_fused_forward_matches_reference_model():
    """Fused model and plain torch ResNet18 share state_dict; eval-mode
    forward on the same weights must agree within bf16 tolerance."""
    from sparktorch_amd.models.resnet import ResNet18
    from sparktorch_amd.ops.modules import ResNet18Fused

    torch.manual_seed(6)
    ref = ResNet18(num_classes=16).to(DEV).eval()
    fused = ResNet18Fused(num_classes=16).to(DEV).eval()
    fused.load_state_dict(ref.state_dict())

    x = torch.randn(4, 3, 224, 224, device=DEV)
    with torch.no_grad():
        y_ref = ref(bf(x).float())
        y = fused(bf(x))
    # 20 convs of bf16 accumulation drift: compare top-1 agreement + value tol
    assert (y.float().argmax(1) == y_ref.argmax(1)).float().mean() >= 0.75
    assert torch.allclose(y.float(), y_ref, atol=0.5, rtol=0.1)


def test_resnet18_fused_train_step_loss_decreases():
    from sparktorch_amd.ops.modules import ResNet18Fused
    from sparktorch_amd.parallel.sync import SyncTrainer

    torch.manual_seed(7)
    model = ResNet18Fused(num_classes=10)
    trainer = SyncTrainer(
        model, nn.CrossEntropyLoss(),
        torch.optim.Adam(model.parameters(), lr=1e-3),
        device=DEV, world_size=1,
    )
    x = bf(torch.randn(16, 3, 224, 224, device=DEV))
    y = torch.randint(0, 10, (16,), device=DEV)
    losses = [trainer.train_step(x, y) for _ in range(8)]
    torch.cuda.synchronize()
    assert all(l == l for l in losses), losses  # no NaN
    assert min(losses[4:]) < losses[0], losses


def test_converted_generic_resnet_runs_native():
    """convert_model_for_mi355x on a plain ResNet-18 swaps Conv2d/BatchNorm2d/
    MaxPool2d for the NCHW-layout Hip modules (functional F.relu stays eager);
    one fwd+bwd must run with native kernels and finite grads."""
    from sparktorch_amd.models.resnet import ResNet18
    from sparktorch_amd.ops.modules import (
        HipBatchNorm2d,
        HipConv2d,
        HipMaxPool2d,
        convert_model_for_mi355x,
    )

    torch.manual_seed(8)
    model = convert_model_for_mi355x(ResNet18(num_classes=8)).to(DEV)
    kinds = {type(m).__name__ for m in model.modules()}
    assert {"HipConv2d", "HipBatchNorm2d", "HipMaxPool2d"} <= kinds

    x = torch.randn(4, 3, 224, 224, device=DEV)
    out = model(bf(x))
    loss = F.cross_entropy(out.float(), torch.randint(0, 8, (4,), device=DEV))
    loss.backward()
    assert torch.isfinite(loss)
    for p in model.parameters():
        assert p.grad is None or torch.isfinite(p.grad).all()
