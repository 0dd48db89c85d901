"""FlatBuckets + SyncTrainer single-process correctness (world_size=1, CPU)."""

import numpy as np
import pytest
import torch
import torch.nn as nn

from sparktorch_amd.models.simple_net import Net
from sparktorch_amd.models.mnist import MnistMLP
from sparktorch_amd.parallel.buckets import FlatBuckets
from sparktorch_amd.parallel.sync import SyncTrainer
from sparktorch_amd.ops.optim import FusedAdam, FusedSGD, fused_optimizer_for


def test_flat_buckets_views_preserve_values():
    model = Net()
    before = {k: v.clone() for k, v in model.state_dict().items()}
    fb = FlatBuckets(list(model.parameters()), world_size=1)
    after = model.state_dict()
    for k in before:
        assert torch.equal(before[k], after[k])
    # grads are views into flat storage
    x = torch.randn(8, 10)
    model(x).sum().backward()
    total = sum(int(b.flat_grad.abs().sum() > 0) for b in fb.buckets)
    assert total >= 1


def test_flat_buckets_grad_accumulation_matches_eager():
    torch.manual_seed(0)
    model_a = Net()
    model_b = Net()
    model_b.load_state_dict(model_a.state_dict())

    fb = FlatBuckets(list(model_b.parameters()), world_size=1)
    x = torch.randn(16, 10)
    y = torch.randn(16, 1)

    crit = nn.MSELoss()
    crit(model_a(x), y).backward()
    crit(model_b(x), y).backward()

    for pa, pb in zip(model_a.parameters(), model_b.parameters()):
        assert torch.allclose(pa.grad, pb.grad, atol=1e-6)


@pytest.mark.parametrize("opt_cls,fused_cls", [(torch.optim.Adam, FusedAdam), (torch.optim.SGD, FusedSGD)])
def test_fused_optimizer_matches_torch(opt_cls, fused_cls):
    torch.manual_seed(1)
    ref = Net()
    fused_model = Net()
    fused_model.load_state_dict(ref.state_dict())

    kwargs = {"lr": 0.01} if opt_cls is torch.optim.SGD else {"lr": 0.01}
    ref_opt = opt_cls(ref.parameters(), **kwargs)

    fb = FlatBuckets(list(fused_model.parameters()), world_size=1)
    fused_opt = fused_optimizer_for(opt_cls(fused_model.parameters(), **kwargs), fb)
    assert isinstance(fused_opt, fused_cls)

    crit = nn.MSELoss()
    x = torch.randn(32, 10)
    y = torch.randn(32, 1)
    for _ in range(5):
        ref_opt.zero_grad()
        crit(ref(x), y).backward()
        ref_opt.step()

        fb.zero_grad()
        crit(fused_model(x), y).backward()
        fb.finalize(average=False)
        fused_opt.step(grad_scale=1.0)

    for pa, pb in zip(ref.parameters(), fused_model.parameters()):
        assert torch.allclose(pa, pb, atol=1e-5), (pa - pb).abs().max()


def test_fused_sgd_momentum_matches_torch():
    torch.manual_seed(2)
    ref = MnistMLP(in_dim=16, hidden=8, classes=3)
    fm = MnistMLP(in_dim=16, hidden=8, classes=3)
    fm.load_state_dict(ref.state_dict())

    ref_opt = torch.optim.SGD(ref.parameters(), lr=0.05, momentum=0.9, weight_decay=1e-4)
    fb = FlatBuckets(list(fm.parameters()), world_size=1)
    fused_opt = fused_optimizer_for(
        torch.optim.SGD(fm.parameters(), lr=0.05, momentum=0.9, weight_decay=1e-4), fb
    )

    crit = nn.CrossEntropyLoss()
    x = torch.randn(64, 16)
    y = torch.randint(0, 3, (64,))
    for _ in range(6):
        ref_opt.zero_grad()
        crit(ref(x), y).backward()
        ref_opt.step()

        fb.zero_grad()
        crit(fm(x), y).backward()
        fb.finalize(average=False)
        fused_opt.step()

    for pa, pb in zip(ref.parameters(), fm.parameters()):
        assert torch.allclose(pa, pb, atol=1e-5)


def test_sync_trainer_reduces_loss():
    torch.manual_seed(3)
    model = MnistMLP(in_dim=10, hidden=32, classes=2)
    opt = torch.optim.Adam(model.parameters(), lr=0.01)
    trainer = SyncTrainer(model, nn.CrossEntropyLoss(), opt, device="cpu", world_size=1)

    x = torch.randn(256, 10)
    y = (x.sum(dim=1) > 0).long()
    first = trainer.train_step(x, y)
    for _ in range(30):
        last = trainer.train_step(x, y)
    assert last < first


def test_sync_trainer_long_label_retry():
    """CrossEntropy with float labels must hit the long-label path."""
    model = Net()
    # Net outputs 1 value; use MSE with float targets through the same helper
    from sparktorch_amd.parallel.sync import compute_loss

    crit = nn.CrossEntropyLoss()
    pred = torch.randn(4, 3)
    y = torch.tensor([0.0, 1.0, 2.0, 1.0])
    loss = compute_loss(crit, pred, y)
    assert loss.dim() == 0

    mse = nn.MSELoss()
    loss2 = compute_loss(mse, torch.randn(4, 1), torch.randn(4, 1))
    assert loss2.dim() == 0


def test_bucket_split_multiple_buckets():
    model = MnistMLP()  # ~270k params
    fb = FlatBuckets(list(model.parameters()), bucket_cap_mb=0.25, world_size=1)
    assert len(fb.buckets) > 1
    assert sum(b.numel() for b in fb.buckets) == sum(p.numel() for p in model.parameters())
