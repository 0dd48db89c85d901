"""Seeded random-shape fuzz over the conv/linear autograd stack vs fp32
torch references — catches boundary-path regressions the hand-picked shape
tables miss (partial tiles, odd spatial dims, K-padding, dispatch-arm
crossovers)."""

import random

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs MI355X", allow_module_level=True)

DEV = "cuda:0"


def bf(x):
    return x.to(torch.bfloat16)


@pytest.mark.parametrize("seed", range(6))
def test_fuzz_conv_nhwc(seed):
    from sparktorch_amd.ops.functional import hip_conv2d_nhwc

    rng = random.Random(1000 + seed)
    B = rng.choice([1, 2, 3, 5])
    CI = rng.choice([3, 8, 16, 24, 64])
    CO = rng.choice([8, 16, 48, 64, 96])
    K = rng.choice([1, 3, 5, 7])
    s = rng.choice([1, 2])
    p = rng.randint(0, K // 2)
    H = rng.randint(max(K, 6), 21)
    W = rng.randint(max(K, 6), 21)
    torch.manual_seed(seed)

    x = torch.randn(B, H, W, CI, device=DEV)
    w = torch.randn(CO, CI, K, K, device=DEV) * (1.0 / (CI * K * K) ** 0.5)
    b = torch.randn(CO, device=DEV)
    relu = bool(seed % 2)

    xb = bf(x).requires_grad_(True)
    wr = w.clone().requires_grad_(True)
    br = b.clone().requires_grad_(True)
    y = hip_conv2d_nhwc(xb, wr, br, (s, s), (p, p), relu=relu)
    gout = torch.randn_like(y.float())
    y.backward(bf(gout))

    x2 = bf(x).float().permute(0, 3, 1, 2).requires_grad_(True)
    w2 = w.clone().requires_grad_(True)
    b2 = b.clone().requires_grad_(True)
    y2 = F.conv2d(x2, bf(w2).float(), b2, stride=s, padding=p)
    if relu:
        y2 = F.relu(y2)
    y2.backward(bf(gout).float().permute(0, 3, 1, 2))

    shape = (B, CI, CO, K, s, p, H, W, relu)
    assert torch.allclose(y.float().permute(0, 3, 1, 2), y2, atol=8e-2, rtol=8e-2), shape
    assert torch.allclose(wr.grad, w2.grad, atol=0.15, rtol=0.08), shape
    assert torch.allclose(br.grad, b2.grad, atol=0.15, rtol=0.08), shape
    assert torch.allclose(xb.grad.float().permute(0, 3, 1, 2), x2.grad, atol=8e-2, rtol=8e-2), shape


@pytest.mark.parametrize("seed", range(6))
def test_fuzz_linear(seed):
    from sparktorch_amd.ops.functional import hip_linear

    rng = random.Random(2000 + seed)
    M = rng.choice([1, 7, 65, 130, 513, 2000])
    K = rng.choice([8, 24, 100, 256, 777])
    N = rng.choice([1, 10, 64, 120, 300])
    torch.manual_seed(seed)
    relu = bool(seed % 2)

    x = torch.randn(M, K, device=DEV)
    w = torch.randn(N, K, device=DEV) * (1.0 / K ** 0.5)
    b = torch.randn(N, device=DEV)

    xb = bf(x).requires_grad_(True)
    wr = w.clone().requires_grad_(True)
    br = b.clone().requires_grad_(True)
    y = hip_linear(xb, wr, br, relu=relu)
    gout = torch.randn_like(y.float())
    y.backward(bf(gout))

    x2 = bf(x).float().requires_grad_(True)
    w2 = w.clone().requires_grad_(True)
    b2 = b.clone().requires_grad_(True)
    y2 = F.linear(x2, bf(w2).float(), b2)
    if relu:
        y2 = F.relu(y2)
    y2.backward(bf(gout).float())

    shape = (M, K, N, relu)
    assert torch.allclose(y.float(), y2, atol=8e-2, rtol=8e-2), shape
    assert torch.allclose(wr.grad, w2.grad, atol=0.15, rtol=0.08), shape
    assert torch.allclose(br.grad, b2.grad, atol=0.15, rtol=0.08), shape
    assert torch.allclose(xb.grad.float(), x2.grad, atol=8e-2, rtol=8e-2), shape
