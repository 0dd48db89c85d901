"""Boundary-shape sweep for the MFMA GEMM: exercises every dispatch arm
(2x2 / 4x1-narrow / 2x4-wide tiles, glds vs register staging, swizzled
fallbacks, split-K, epilogue LDS-transpose vs scalar edge path) against plain
fp32 torch references."""

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs MI355X", allow_module_level=True)

from sparktorch_amd import ops

DEV = "cuda:0"


def bf(x):
    return x.to(torch.bfloat16)


SHAPES = [
    # (M, N, K) — tile boundaries, narrow/wide arms, odd K (no glds), tails
    (1, 10, 8),
    (127, 64, 25),          # partial M-tile, narrow-N candidate, odd K
    (129, 10, 147),         # M just over one tile, tiny N
    (300, 64, 576),         # narrow arm with M tail
    (512, 640, 64),         # wide-N arm (N>=256), single K slab
    (1000, 200, 72),        # unaligned N (no vector epilogue), M tail
    (4096, 256, 784),       # glds fwd, wide wgrad
    (777, 384, 152),        # everything misaligned except K
]


@pytest.mark.parametrize("M,N,K", SHAPES)
def test_linear_fwd_dgrad_wgrad_shapes(M, N, K):
    torch.manual_seed(M * 31 + N * 7 + K)
    x = bf(torch.randn(M, K, device=DEV)).contiguous()
    w = (torch.randn(N, K, device=DEV) * (1.0 / K ** 0.5)).contiguous()
    b = torch.randn(N, device=DEV).contiguous()
    dz = bf(torch.randn(M, N, device=DEV)).contiguous()

    y = ops.ext().linear_fwd(x, w, b, False)
    y_ref = x.float() @ bf(w).float().t() + b
    assert torch.allclose(y.float(), y_ref, atol=0.1, rtol=0.05), (M, N, K, "fwd")

    dx = ops.ext().linear_dgrad(dz, w)
    dx_ref = dz.float() @ bf(w).float()
    assert torch.allclose(dx.float(), dx_ref, atol=0.1, rtol=0.05), (M, N, K, "dgrad")

    for sk in (1, 4):
        dw = ops.ext().linear_wgrad(dz, x, sk)
        dw_ref = dz.float().t() @ x.float()
        assert torch.allclose(dw, dw_ref, atol=0.5, rtol=0.05), (M, N, K, sk, "wgrad")


@pytest.mark.parametrize("M,N,K", [(129, 64, 152), (4096, 128, 64)])
def test_linear_fwd_bf16_weights_glds_b(M, N, K):
    """bf16 k-contiguous B takes the dual-DMA (A+B glds) arm."""
    torch.manual_seed(0)
    x = bf(torch.randn(M, K, device=DEV)).contiguous()
    wb = bf(torch.randn(N, K, device=DEV) * (1.0 / K ** 0.5)).contiguous()
    y = ops.ext().linear_fwd(x, wb, None, False)
    y_ref = x.float() @ wb.float().t()
    assert torch.allclose(y.float(), y_ref, atol=0.1, rtol=0.05)
