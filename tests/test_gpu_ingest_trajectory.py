"""GPU tests for the device-side Vector ingest path and full-model
trajectory parity of the native bf16 trainers vs eager fp32 torch.

Trajectory parity (VERDICT round-1 item 5): per-op numerics tests bound each
kernel's error, but a systematic drift (wrong grad scale, missed cast, stale
buffer) only shows up over a multi-step trajectory — so train the SAME model
from the SAME init on the SAME data for 20 steps, native bf16 vs eager fp32,
and require the loss curves to track.
"""

import numpy as np
import pytest
import torch
import torch.nn as nn

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():  # collected but skipped off-GPU
    pytest.skip("needs MI355X", allow_module_level=True)

from sparktorch_amd import ops
from sparktorch_amd.parallel.sync import SyncTrainer
from sparktorch_amd.utils.data import handle_features, handle_features_device
from sparktorch_amd.utils.serialize import DataObj

DEV = "cuda:0"


# ---------------------------------------------------------------------------
# device ingest
# ---------------------------------------------------------------------------


def test_cast_f64_bf16_matches_two_step():
    torch.manual_seed(0)
    for n in (17, 4096, 100_001):
        x = torch.randn(n, device=DEV, dtype=torch.float64) * 3.0
        got = ops.ext().cast_f64_bf16(x.contiguous())
        ref = x.float().to(torch.bfloat16)  # same two-step rounding semantics
        assert got.dtype == torch.bfloat16
        assert torch.equal(got.view(torch.int16), ref.view(torch.int16))


def _rows(n, dim, seed=3, labels=True):
    rng = np.random.default_rng(seed)
    out = []
    for i in range(n):
        out.append(DataObj(rng.standard_normal(dim),  # fp64, like DenseVector.toArray
                           float(i % 7) if labels else None, None, None))
    return out


def test_handle_features_device_matches_cpu_pack():
    rows = _rows(513, 37)
    np.random.seed(42)
    dev = handle_features_device(rows, 0.0, device=DEV)
    cpu = handle_features(rows, 0.0)
    assert dev.x_train.is_cuda and dev.x_train.dtype == torch.bfloat16
    ref_x = cpu.x_train.to(torch.bfloat16)
    assert torch.equal(dev.x_train.cpu().view(torch.int16), ref_x.view(torch.int16))
    assert torch.equal(dev.y_train.cpu(), cpu.y_train)


def test_handle_features_device_validation_split():
    rows = _rows(400, 16)
    np.random.seed(7)
    dev = handle_features_device(rows, 0.25, device=DEV)
    np.random.seed(7)
    cpu = handle_features(rows, 0.25)
    assert dev.x_train.shape[0] == cpu.x_train.shape[0] == 300
    assert dev.x_val.shape[0] == cpu.x_val.shape[0] == 100
    assert torch.equal(dev.x_val.cpu().view(torch.int16),
                       cpu.x_val.to(torch.bfloat16).view(torch.int16))
    assert torch.equal(dev.y_val.cpu(), cpu.y_val)


def test_handle_features_device_autoencoder_and_empty():
    d = handle_features_device(_rows(64, 10, labels=False), 0.0, device=DEV)
    assert d.y_train is None and d.x_train.shape == (64, 10)
    e = handle_features_device([], 0.0, device=DEV)
    assert e.x_train is None


def test_handle_features_device_fp32_rows_skip_cast():
    rng = np.random.default_rng(5)
    rows = [DataObj(rng.standard_normal(8).astype(np.float32), 1.0, None, None)
            for _ in range(32)]
    d = handle_features_device(rows, 0.0, device=DEV)
    assert d.x_train.dtype == torch.bfloat16 and d.x_train.shape == (32, 8)


# ---------------------------------------------------------------------------
# full-model trajectory parity vs eager fp32 (20 steps)
# ---------------------------------------------------------------------------


def _train_curve(model, x, y, steps, device, criterion=None, native=True):
    crit = criterion if criterion is not None else nn.CrossEntropyLoss()
    if native:
        opt = torch.optim.Adam(model.parameters(), lr=1e-3)
        tr = SyncTrainer(model, crit, opt, device=device, world_size=1)
        return [tr.train_step(x, y) for _ in range(steps)]
    model = model.to(device)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    losses = []
    for _ in range(steps):
        opt.zero_grad(set_to_none=True)
        loss = crit(model(x), y)
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    return losses


def _assert_curves_track(native, eager, rel=0.15, label=""):
    assert len(native) == len(eager)
    assert native[-1] < native[0], "%s native loss did not decrease: %r" % (label, native)
    assert eager[-1] < eager[0], "%s eager loss did not decrease" % label
    for i, (a, b) in enumerate(zip(native, eager)):
        denom = max(abs(b), 1e-3)
        assert abs(a - b) / denom < rel, (
            "%s step %d: native %.5f vs fp32 %.5f (rel %.3f)\nnative=%r\neager=%r"
            % (label, i, a, b, abs(a - b) / denom, native, eager)
        )


def test_trajectory_parity_mlp_20_steps():
    from sparktorch_amd.models.mnist import MnistMLP

    torch.manual_seed(11)
    x32 = torch.randn(8192, 784, device=DEV)
    y = torch.randint(0, 10, (8192,), device=DEV)

    torch.manual_seed(5)
    m_native = MnistMLP()
    torch.manual_seed(5)
    m_eager = MnistMLP()

    native = _train_curve(m_native, x32.to(torch.bfloat16), y, 20, DEV, native=True)
    eager = _train_curve(m_eager, x32, y, 20, DEV, native=False)
    _assert_curves_track(native, eager, rel=0.15, label="mlp")


def test_trajectory_parity_cnn_20_steps():
    from sparktorch_amd.models.mnist import MnistCNN

    torch.manual_seed(12)
    x32 = torch.randn(1024, 784, device=DEV)
    y = torch.randint(0, 10, (1024,), device=DEV)

    torch.manual_seed(6)
    m_native = MnistCNN()
    torch.manual_seed(6)
    m_eager = MnistCNN()
    # dropout masks are RNG-backend-specific (native counter-based vs torch) —
    # disable it so the curves are comparable deterministically
    m_native.dropout.p = 0.0
    m_eager.dropout.p = 0.0

    # SyncTrainer converts Conv/Linear/pool/dropout to the native kernels
    native = _train_curve(m_native, x32.to(torch.bfloat16), y, 20, DEV, native=True)
    eager = _train_curve(m_eager, x32, y, 20, DEV, native=False)
    _assert_curves_track(native, eager, rel=0.25, label="cnn")


def test_trajectory_parity_resnet18_10_steps():
    from sparktorch_amd.models.resnet import ResNet18

    torch.manual_seed(13)
    x32 = torch.randn(64, 3 * 224 * 224, device=DEV)
    y = torch.randint(0, 1000, (64,), device=DEV)

    torch.manual_seed(7)
    m_native = ResNet18()
    torch.manual_seed(7)
    m_eager = ResNet18()

    native = _train_curve(m_native, x32.to(torch.bfloat16), y, 10, DEV, native=True)
    eager = _train_curve(m_eager, x32, y, 10, DEV, native=False)
    # BN-heavy model at batch 64: bf16 stats noise compounds faster
    _assert_curves_track(native, eager, rel=0.35, label="resnet18")
