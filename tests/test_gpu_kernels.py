"""Numerics tests for the hand-written gfx950 kernels vs plain-torch fp32
references (computed on bf16-quantized inputs, since the kernels run bf16
MFMA with fp32 accumulation)."""

import numpy as np
import pytest
import torch
import torch.nn as nn
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():  # collected but skipped off-GPU
    pytest.skip("needs MI355X", allow_module_level=True)

from sparktorch_amd import ops

DEV = "cuda:0"


def bf(x):
    return x.to(torch.bfloat16)


def ref_of(x):
    """fp32 value of the bf16-quantized tensor (what the kernel actually sees)."""
    return bf(x).float()


@pytest.mark.parametrize("M,K,N", [(128, 128, 128), (257, 123, 77), (1024, 784, 256), (64, 10, 3)])
def test_linear_fwd(M, K, N):
    torch.manual_seed(0)
    x = torch.randn(M, K, device=DEV)
    w = torch.randn(N, K, device=DEV) * 0.1
    b = torch.randn(N, device=DEV)
    y = ops.ext().linear_fwd(bf(x).contiguous(), w.contiguous(), b, False)
    ref = ref_of(x) @ ref_of(w).t() + b
    err = (y.float() - ref).abs().max().item()
    scale = ref.abs().max().item() + 1e-6
    assert err / scale < 0.02, err


def test_linear_fwd_relu():
    torch.manual_seed(1)
    x = torch.randn(300, 200, device=DEV)
    w = torch.randn(100, 200, device=DEV) * 0.1
    b = torch.randn(100, device=DEV)
    y = ops.ext().linear_fwd(bf(x).contiguous(), w.contiguous(), b, True)
    ref = F.relu(ref_of(x) @ ref_of(w).t() + b)
    assert (y.float() - ref).abs().max().item() < 0.05


def test_linear_fwd_no_bias():
    x = torch.randn(130, 64, device=DEV)
    w = torch.randn(32, 64, device=DEV)
    y = ops.ext().linear_fwd(bf(x).contiguous(), w.contiguous(), None, False)
    ref = ref_of(x) @ ref_of(w).t()
    assert (y.float() - ref).abs().max().item() / (ref.abs().max() + 1e-6) < 0.02


def test_linear_dgrad():
    torch.manual_seed(2)
    dz = torch.randn(257, 96, device=DEV)
    w = torch.randn(96, 200, device=DEV) * 0.1
    dx = ops.ext().linear_dgrad(bf(dz).contiguous(), w.contiguous())
    ref = ref_of(dz) @ ref_of(w)
    assert (dx.float() - ref).abs().max().item() / (ref.abs().max() + 1e-6) < 0.02


@pytest.mark.parametrize("splitk", [1, 4, 16])
def test_linear_wgrad_splitk(splitk):
    torch.manual_seed(3)
    B, N, K = 4096, 64, 96
    dz = torch.randn(B, N, device=DEV) * 0.03
    x = torch.randn(B, K, device=DEV)
    dw = ops.ext().linear_wgrad(bf(dz).contiguous(), bf(x).contiguous(), splitk)
    ref = ref_of(dz).t() @ ref_of(x)
    rel = (dw - ref).abs().max().item() / (ref.abs().max().item() + 1e-6)
    assert rel < 0.02, rel


@pytest.mark.parametrize("K", [96, 128, 784])  # K%128==0 exercises the extra tile
def test_linear_wgrad_bias_fused(K):
    torch.manual_seed(30)
    B, N = 4096, 64
    dz = torch.randn(B, N, device=DEV) * 0.03
    x = torch.randn(B, K, device=DEV)
    dw = torch.zeros(N, K, device=DEV)
    db = torch.zeros(N, device=DEV)
    ops.ext().linear_wgrad_bias_into(bf(dz).contiguous(), bf(x).contiguous(), dw, db, 8)
    ref_dw = ref_of(dz).t() @ ref_of(x)
    ref_db = ref_of(dz).sum(0)
    assert (dw - ref_dw).abs().max().item() / (ref_dw.abs().max().item() + 1e-6) < 0.02
    assert (db - ref_db).abs().max().item() / (ref_db.abs().max().item() + 1e-6) < 0.02
    # accumulate semantics: second call doubles
    ops.ext().linear_wgrad_bias_into(bf(dz).contiguous(), bf(x).contiguous(), dw, db, 8)
    assert (dw - 2 * ref_dw).abs().max().item() / (2 * ref_dw.abs().max().item() + 1e-6) < 0.02


def test_bias_grad():
    dz = torch.randn(5000, 37, device=DEV)
    db = ops.ext().bias_grad(bf(dz).contiguous())
    ref = ref_of(dz).sum(0)
    assert torch.allclose(db, ref, rtol=1e-2, atol=1e-2)


def test_matmul_transposes():
    torch.manual_seed(4)
    a = torch.randn(100, 60, device=DEV)
    b = torch.randn(60, 80, device=DEV)
    # asymmetric operands catch row/col swaps (guide §3)
    c = ops.ext().matmul_bf16(bf(a).contiguous(), bf(b).contiguous(), False, False)
    ref = ref_of(a) @ ref_of(b)
    assert (c - ref).abs().max() / ref.abs().max() < 0.02

    c2 = ops.ext().matmul_bf16(bf(a.t()).contiguous(), bf(b).contiguous(), True, False)
    assert (c2 - ref).abs().max() / ref.abs().max() < 0.02

    c3 = ops.ext().matmul_bf16(bf(a).contiguous(), bf(b.t()).contiguous(), False, True)
    assert (c3 - ref).abs().max() / ref.abs().max() < 0.02


def test_relu_bwd():
    dy = torch.randn(1000, device=DEV)
    y = torch.randn(1000, device=DEV)
    dz = ops.ext().relu_bwd(bf(dy).contiguous(), bf(y).contiguous())
    ref = bf(dy).float() * (bf(y).float() > 0)
    assert torch.allclose(dz.float(), ref, atol=1e-2)


def test_ce_fused_vs_torch():
    torch.manual_seed(5)
    B, C = 1037, 10
    logits = torch.randn(B, C, device=DEV)
    tgt = torch.randint(0, C, (B,), device=DEV)
    loss, dlogits = ops.ext().ce_fused(bf(logits).contiguous(), tgt)

    lref = bf(logits).float().detach().requires_grad_(True)
    ref_loss = F.cross_entropy(lref, tgt)
    ref_loss.backward()
    assert abs(loss.item() - ref_loss.item()) < 2e-3
    assert (dlogits.float() - lref.grad).abs().max().item() < 1e-3


def test_mse_fused_vs_torch():
    pred = torch.randn(513, 7, device=DEV)
    tgt = torch.randn(513, 7, device=DEV)
    loss, dpred = ops.ext().mse_fused(bf(pred).contiguous(), bf(tgt).contiguous())
    p = bf(pred).float().detach().requires_grad_(True)
    ref = F.mse_loss(p, bf(tgt).float())
    ref.backward()
    assert abs(loss.item() - ref.item()) < 1e-2
    assert (dpred.float() - p.grad).abs().max().item() < 1e-3


def test_fused_adam_matches_torch_gpu():
    torch.manual_seed(6)
    n = 100_003
    p0 = torch.randn(n, device=DEV)
    g = torch.randn(n, device=DEV)

    p_ref = p0.clone().requires_grad_(True)
    opt = torch.optim.Adam([p_ref], lr=0.01)
    p_hip = p0.clone()
    m = torch.zeros(n, device=DEV)
    v = torch.zeros(n, device=DEV)
    for t in range(1, 4):
        p_ref.grad = g.clone()
        opt.step()
        bc1 = 1 - 0.9**t
        bc2 = 1 - 0.999**t
        ops.ext().fused_adam(p_hip, g, m, v, 0.01, 0.9, 0.999, 1e-8, 0.0, bc1, bc2, 1.0, False,
                             None)
    assert (p_hip - p_ref).abs().max().item() < 1e-5


def test_cast_kernels():
    d = torch.randn(10_001, device=DEV, dtype=torch.float64)
    f = ops.ext().cast_f64_f32(d)
    assert torch.allclose(f, d.float())
    b16 = ops.ext().cast_f32_bf16(f)
    assert (b16.float() - f).abs().max().item() < 0.01 * f.abs().max().item()


def test_hip_linear_autograd_full_layer():
    """Whole fused layer fwd+bwd vs torch fp32 on bf16-quantized data."""
    from sparktorch_amd.ops.functional import hip_linear

    torch.manual_seed(7)
    x = torch.randn(512, 64, device=DEV)
    w = (torch.randn(32, 64, device=DEV) * 0.1).requires_grad_(True)
    b = torch.randn(32, device=DEV).requires_grad_(True)
    xb = bf(x).requires_grad_(True)

    y = hip_linear(xb, w, b, relu=True)
    gout = torch.randn_like(y.float())
    y.backward(bf(gout))

    xr = ref_of(x).requires_grad_(True)
    wr = w.detach().clone().requires_grad_(True)
    br = b.detach().clone().requires_grad_(True)
    yr = F.relu(xr @ ref_of(wr).t() + br)
    yr.backward(bf(gout).float())

    assert (y.float() - yr).abs().max().item() < 0.05
    assert (w.grad - wr.grad).abs().max().item() / (wr.grad.abs().max() + 1e-6) < 0.03
    assert (b.grad - br.grad).abs().max().item() / (br.grad.abs().max() + 1e-6) < 0.03
    assert (xb.grad.float() - xr.grad).abs().max().item() / (xr.grad.abs().max() + 1e-6) < 0.03


def test_sync_trainer_gpu_end_to_end():
    from sparktorch_amd.ops.modules import MnistMLPFused
    from sparktorch_amd.parallel.sync import SyncTrainer

    torch.manual_seed(8)
    model = MnistMLPFused()
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    trainer = SyncTrainer(model, nn.CrossEntropyLoss(), opt, device=DEV, world_size=1)
    x = torch.randn(4096, 784, device=DEV, dtype=torch.bfloat16)
    y = torch.randint(0, 10, (4096,), device=DEV)
    losses = [trainer.train_step(x, y) for _ in range(20)]
    assert losses[-1] < losses[0]
    assert np.isfinite(losses).all()


def test_native_extension_is_loaded_not_fallback():
    """Fail loudly if the HIP extension would be bypassed on a GPU box."""
    assert ops.available(), "_sparkhip must be importable on the GPU box"
    import sparktorch_amd.ops._sparkhip as ext

    assert "sparktorch_amd" in ext.__file__  # in-tree .so, not site-packages


def test_graphed_inference_parity():
    from sparktorch_amd.ops.graph import GraphedForward
    from sparktorch_amd.models.mnist import MnistMLP

    torch.manual_seed(9)
    model = MnistMLP().to(DEV).eval()
    gf = GraphedForward(model, device=DEV, batch_size=256)
    x = torch.randn(100, 784)
    with torch.no_grad():
        ref = model(x.to(DEV))
    out = gf(x)
    assert torch.allclose(out, ref, atol=1e-4)


def test_small_wgrad_path_matches_gemm():
    """CO*K<=1024 wgrad takes the batched outer-product kernel (M>=65536
    triggers it); verify against the fp32 reference."""
    torch.manual_seed(5)
    M, CO, K = 65536, 16, 25
    dz = torch.randn(M, CO, device=DEV).to(torch.bfloat16).contiguous()
    x = torch.randn(M, K, device=DEV).to(torch.bfloat16).contiguous()
    dw = ops.ext().linear_wgrad(dz, x, 8)
    ref = dz.float().t() @ x.float()
    assert torch.allclose(dw, ref, atol=0.5, rtol=1e-2)
    # _into variant accumulates
    acc = torch.ones(CO, K, device=DEV)
    ops.ext().linear_wgrad_into(dz, x, acc, 8)
    assert torch.allclose(acc, ref + 1.0, atol=0.5, rtol=1e-2)


def test_hipgraph_train_step_matches_eager():
    """compileMode='hipgraph' captures zero+fwd+bwd+fused-step and replays;
    identical seeds must give the same loss trajectory as the eager trainer
    (the reference's torch.compile support is broken — distributed.py:117-118;
    this is the working equivalent)."""
    from sparktorch_amd.ops.modules import MnistMLPFused
    from sparktorch_amd.parallel.sync import SyncTrainer

    x = torch.randn(4096, 784, device=DEV).to(torch.bfloat16)
    y = torch.randint(0, 10, (4096,), device=DEV)

    losses = {}
    for mode in (None, "hipgraph"):
        torch.manual_seed(42)
        model = MnistMLPFused()
        tr = SyncTrainer(
            model, nn.CrossEntropyLoss(), torch.optim.Adam(model.parameters(), lr=1e-3),
            device=DEV, world_size=1, compile_mode=mode,
        )
        losses[mode] = [tr.train_step(x, y) for _ in range(6)]
    torch.cuda.synchronize()
    for a, b in zip(losses[None], losses["hipgraph"]):
        assert abs(a - b) < 5e-3 + 0.01 * abs(a), (losses[None], losses["hipgraph"])


def test_hipgraph_with_batchnorm_buffers_restored():
    """hipGraph whole-step capture on a BN model: the 2 real warmup steps
    must not leak BatchNorm running-stat mutations into the first replay
    (warmup snapshot/restore covers module buffers — ADVICE round-1 fix).
    The graphed trajectory must match an eager trainer started from the
    same init."""
    import copy

    from sparktorch_amd.parallel.sync import SyncTrainer

    def build():
        torch.manual_seed(31)
        m = nn.Sequential(
            nn.Conv2d(16, 16, 3, padding=1, bias=False),
            nn.BatchNorm2d(16),
            nn.ReLU(),
            nn.Flatten(),
            nn.Linear(16 * 8 * 8, 10),
        )

        class Wrap(nn.Module):
            def __init__(self):
                super().__init__()
                self.m = m

            def forward(self, x):
                return self.m(x.view(-1, 16, 8, 8))

        return Wrap()

    x = torch.randn(512, 16 * 8 * 8, device=DEV).to(torch.bfloat16)
    y = torch.randint(0, 10, (512,), device=DEV)

    eager = SyncTrainer(build(), nn.CrossEntropyLoss(),
                        torch.optim.Adam(build().parameters(), lr=1e-3),
                        device=DEV, world_size=1)
    # NB: optimizer param groups rebuilt inside SyncTrainer; the ctor above
    # only supplies defaults
    graphed = SyncTrainer(build(), nn.CrossEntropyLoss(),
                          torch.optim.Adam(build().parameters(), lr=1e-3),
                          device=DEV, world_size=1, compile_mode="hipgraph")

    le = [eager.train_step(x, y) for _ in range(6)]
    lg = [graphed.train_step(x, y) for _ in range(6)]
    for i, (a, b) in enumerate(zip(le, lg)):
        assert abs(a - b) / max(abs(a), 1e-3) < 0.05, (i, le, lg)
