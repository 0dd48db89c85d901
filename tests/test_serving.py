"""Serving layer: SparkTorchModel-compatible scoring over HTTP."""

import numpy as np
import pytest
import torch
import torch.nn as nn

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient

from sparktorch_amd.inference import create_spark_torch_model
from sparktorch_amd.serving import InferenceServer


def _net(out=3):
    torch.manual_seed(0)
    return nn.Sequential(nn.Linear(10, 16), nn.ReLU(), nn.Linear(16, out))


def test_predict_matches_transformer_semantics():
    net = _net(out=3)
    srv = InferenceServer(net, device="cpu", batch_size=16)
    x = np.random.default_rng(0).normal(size=(40, 10)).astype(np.float32)

    preds = srv.predict(x.tolist())
    with torch.no_grad():
        ref = net(torch.from_numpy(x)).argmax(dim=1).double().tolist()
    assert preds == ref  # argmax semantics, batching invisible

    vecs = srv.predict(x[:3].tolist(), vector_out=True)
    with torch.no_grad():
        refv = net(torch.from_numpy(x[:3])).tolist()
    assert np.allclose(vecs, refv, atol=1e-6)


def test_accepts_stage_and_modstr():
    net = _net(out=1)
    stage = create_spark_torch_model(net, "features", "p")
    x = [[0.1] * 10, [0.2] * 10]
    a = InferenceServer(stage, device="cpu").predict(x)
    b = InferenceServer(stage.getOrDefault(stage.modStr), device="cpu").predict(x)
    with torch.no_grad():
        ref = net(torch.tensor(x)).reshape(-1).tolist()
    assert np.allclose(a, ref, atol=1e-6) and a == b


def test_http_endpoints():
    srv = InferenceServer(_net(out=3), device="cpu", batch_size=8)
    client = TestClient(srv.app())

    assert client.get("/health").json()["status"] == "ok"

    x = [[0.0] * 10, [1.0] * 10]
    r = client.post("/predict", json={"instances": x})
    assert r.status_code == 200
    preds = r.json()["predictions"]
    assert len(preds) == 2 and all(isinstance(p, float) for p in preds)

    r = client.post("/predict", json={"instances": x, "vector_out": True})
    assert r.status_code == 200
    assert len(r.json()["predictions"][0]) == 3

    assert client.post("/predict", json={"instances": []}).status_code == 400
    assert client.post("/predict", json={"instances": [[1.0, 2.0]]}).status_code == 400

    info = client.get("/info").json()
    assert info["hipgraph"] is False and info["n_served"] == 4


@pytest.mark.gpu
def test_serving_hipgraph_gpu():
    if not torch.cuda.is_available():
        pytest.skip("needs MI355X")
    srv = InferenceServer(_net(out=3), device="cuda:0", batch_size=64)
    assert srv._runner is not None  # hipGraph path active
    x = np.random.default_rng(1).normal(size=(200, 10)).astype(np.float32)
    preds = srv.predict(x.tolist())
    with torch.no_grad():
        ref = _net(out=3)(torch.from_numpy(x)).argmax(dim=1).double().tolist()
    # bf16 graphed forward vs fp32 eager: argmax agrees on all but
    # near-ties; require 95% agreement
    agree = sum(1 for a, b in zip(preds, ref) if a == b) / len(ref)
    assert agree > 0.95, agree


def test_predict_thread_safe():
    """Concurrent /predict calls must serialize on the scoring engine
    (the GPU path replays a hipGraph with static buffers)."""
    import concurrent.futures

    net = _net(out=3)
    srv = InferenceServer(net, device="cpu", batch_size=32)
    x = np.random.default_rng(2).normal(size=(64, 10)).astype(np.float32)
    with torch.no_grad():
        ref = net(torch.from_numpy(x)).argmax(dim=1).double().tolist()
    with concurrent.futures.ThreadPoolExecutor(max_workers=8) as ex:
        futs = [ex.submit(srv.predict, x.tolist()) for _ in range(16)]
        for f in futs:
            assert f.result() == ref
    assert srv._n_served == 16 * 64


def test_load_stage_roundtrip(tmp_path):
    """Saved carrier stage -> serving loader -> same predictions."""
    from sparktorch_amd.serving import InferenceServer, load_stage

    net = _net(out=1)
    stage = create_spark_torch_model(net, "features", "p")
    path = str(tmp_path / "stage")
    stage.write().overwrite().save(path)
    srv = InferenceServer(load_stage(path), device="cpu")
    x = [[0.3] * 10]
    with torch.no_grad():
        ref = float(net(torch.tensor(x)).reshape(-1)[0])
    assert abs(srv.predict(x)[0] - ref) < 1e-6
