"""GPU end-to-end tests: estimator fit/transform on MI355X, multi-rank
collectives with GPU tensors, hogwild GPU worker."""

import numpy as np
import pytest
import torch
import torch.nn as nn

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs MI355X", allow_module_level=True)

from sparktorch_amd import SparkTorch, serialize_torch_obj
from sparktorch_amd.compat.local import LocalDataFrame, free_port
from sparktorch_amd.models.mnist import MnistMLP


def _df(n=400, dim=20, parts=2):
    rng = np.random.RandomState(0)
    feats = rng.normal(0, 1, (n, dim)).astype(np.float64)
    labels = (feats.sum(axis=1) > 0).astype(np.float64)
    return LocalDataFrame.from_arrays(feats, list(labels), num_partitions=parts)


def test_estimator_fit_transform_on_gpu_single_partition():
    df = _df(parts=1)
    obj = serialize_torch_obj(
        MnistMLP(in_dim=20, hidden=32, classes=2), nn.CrossEntropyLoss(), torch.optim.Adam, lr=0.01
    )
    est = SparkTorch(
        inputCol="features",
        labelCol="label",
        predictionCol="predicted",
        torchObj=obj,
        iters=10,
        mode="synchronous",
        device="cuda:0",
    )
    model = est.fit(df)
    out = model.transform(df)
    preds = {r["predicted"] for r in out.collect()}
    assert preds.issubset({0.0, 1.0})


def test_sync_two_ranks_one_gpu_gloo():
    """world_size=2 on one GPU over gloo: exercises the full bucket/direct-
    grad/all-reduce/consensus path with GPU tensors (RCCL needs one rank per
    GPU; gloo stands in so the logic is testable on a 1-GPU box)."""
    from sparktorch_amd.parallel.sync import train_distributed
    from sparktorch_amd.utils.data import handle_data

    df = _df(parts=2)
    obj = serialize_torch_obj(
        MnistMLP(in_dim=20, hidden=16, classes=2), nn.CrossEntropyLoss(), torch.optim.Adam, lr=0.01
    )
    rdd = df.rdd.mapPartitions(handle_data("features", "label"))
    state = train_distributed(
        rdd, obj, iters=3, device="cuda:0", backend="gloo", early_stop_patience=2,
        validation_pct=0.25,
    )
    for v in state.values():
        assert torch.isfinite(v).all()


def test_hogwild_gpu_worker():
    df = _df(parts=2)
    obj = serialize_torch_obj(
        MnistMLP(in_dim=20, hidden=16, classes=2), nn.MSELoss(), torch.optim.Adam, lr=0.01
    )
    est = SparkTorch(
        inputCol="features",
        labelCol="label",
        predictionCol="predicted",
        torchObj=obj,
        iters=3,
        mode="hogwild",
        port=free_port(),
        device="cuda:0",
        acquireLock=True,
    )
    model = est.fit(df)
    assert model.transform(df).count() == 400


def test_rccl_world1_init_allreduce_and_train():
    """RCCL (torch 'nccl' on ROCm) smoke on one GPU: init, all-reduce, a
    bucketed train step with the process group live.  One rank per GPU is
    RCCL's model, so world_size=1 is what a 1-GPU box can genuinely run;
    the multi-rank logic is covered by the gloo world_size=2 tests."""
    import os

    import torch.distributed as dist

    from sparktorch_amd.compat.local import free_port
    from sparktorch_amd.parallel.sync import SyncTrainer

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ["MASTER_PORT"] = str(free_port())
    dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        t = torch.arange(1024, device="cuda:0", dtype=torch.float32)
        dist.all_reduce(t)
        assert float(t[-1]) == 1023.0

        torch.manual_seed(0)
        model = MnistMLP(in_dim=20, hidden=16, classes=2)
        trainer = SyncTrainer(
            model, nn.CrossEntropyLoss(), torch.optim.Adam(model.parameters(), lr=1e-2),
            device="cuda:0", world_size=1,
        )
        x = torch.randn(256, 20, device="cuda:0", dtype=torch.bfloat16)
        y = torch.randint(0, 2, (256,), device="cuda:0")
        l0 = trainer.train_step(x, y)
        for _ in range(4):
            l1 = trainer.train_step(x, y)
        assert l1 == l1 and l1 <= l0 * 1.5
    finally:
        dist.destroy_process_group()


def test_estimator_resnet_fit_transform_on_gpu():
    """Full-stack GPU fit of a (tiny-classes) ResNet-18 through the
    estimator: dill serialization of the 11M-param net, converter swap to
    the native NHWC-capable modules, sync training, batched transform."""
    import numpy as np

    from sparktorch_amd.models.resnet import ResNet18

    rng = np.random.RandomState(7)
    feats = rng.standard_normal((24, 3 * 224 * 224)).astype(np.float64) * 0.1
    labels = list(rng.randint(0, 4, 24).astype(np.float64))
    df = LocalDataFrame.from_arrays(feats, labels, num_partitions=1)

    obj = serialize_torch_obj(ResNet18(num_classes=4), nn.CrossEntropyLoss(),
                              torch.optim.Adam, lr=1e-3)
    model = SparkTorch(
        inputCol="features", labelCol="label", predictionCol="predicted",
        torchObj=obj, iters=2, miniBatch=8, device="cuda:0", mode="synchronous",
    ).fit(df)
    out = model.transform(df).collect()
    assert len(out) == 24
    assert all(r["predicted"] in (0.0, 1.0, 2.0, 3.0) for r in out)


def test_arbitrary_module_transformer_trains():
    """Parity guarantee beyond the model zoo: the reference accepts ANY
    nn.Module (serialize_torch_obj is generic).  Modules the converter has
    no native kernel for (LayerNorm, MultiheadAttention, GELU) must run
    through torch-ROCm eager inside the same trainer, mixed with native
    HipLinear layers, and train."""
    import torch.nn as nn

    from sparktorch_amd.parallel.sync import SyncTrainer

    torch.manual_seed(21)

    class TinyTransformer(nn.Module):
        def __init__(self, d=64, nhead=4, seq=16, classes=10):
            super().__init__()
            self.seq, self.d = seq, d
            self.inp = nn.Linear(49, d)          # -> HipLinear via converter
            self.attn = nn.MultiheadAttention(d, nhead, batch_first=True)
            self.ln1 = nn.LayerNorm(d)
            self.ff = nn.Sequential(nn.Linear(d, 4 * d), nn.GELU(), nn.Linear(4 * d, d))
            self.ln2 = nn.LayerNorm(d)
            self.head = nn.Linear(d, classes)

        def forward(self, x):
            x = x.view(-1, self.seq, 49).float()
            # native HipLinear layers emit bf16 activations; eager fp32
            # modules (attention/LayerNorm) take an explicit .float() at the
            # boundary — standard mixed-precision practice
            h = self.inp(x).float()
            a, _ = self.attn(h, h, h, need_weights=False)
            h = self.ln1(h + a)
            h = self.ln2(h + self.ff(h).float())
            return self.head(h.mean(dim=1))

    model = TinyTransformer()
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    tr = SyncTrainer(model, nn.CrossEntropyLoss(), opt, device="cuda:0", world_size=1)

    x = torch.randn(512, 16 * 49, device="cuda:0").to(torch.bfloat16)
    y = torch.randint(0, 10, (512,), device="cuda:0")
    losses = [tr.train_step(x, y) for _ in range(12)]
    # steady descent (measured ~0.012/step at this lr); monotone overall
    assert losses[-1] < losses[0] - 0.05, losses
    assert losses[-1] < min(losses[:3]), losses

    # the dense layers really did convert to the native path
    from sparktorch_amd.ops.modules import HipLinear

    assert isinstance(tr.model.inp, HipLinear)
    assert isinstance(tr.model.head, HipLinear)
