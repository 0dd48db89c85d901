import json

import numpy as np
import pytest
import torch
import torch.nn as nn

from sparktorch_amd import serialize_torch_obj, serialize_torch_obj_lazy
from sparktorch_amd.models.simple_net import Net, NetworkWithParameters
from sparktorch_amd.utils.codec import (
    b64_to_obj,
    obj_to_b64,
    obj_to_stopwords,
    stopwords_to_obj,
)
from sparktorch_amd.utils.serialize import load_base_torch, load_torch_model


def test_b64_roundtrip():
    obj = {"a": 1, "t": torch.ones(3)}
    back = b64_to_obj(obj_to_b64(obj))
    assert back["a"] == 1
    assert torch.equal(back["t"], torch.ones(3))


def test_stopwords_codec_roundtrip():
    obj = {"x": list(range(100))}
    sw = obj_to_stopwords(obj)
    assert sw[-1] == "4c1740b00d3c4ff6806a1402321572cb"
    assert all(tok == "" or 0 <= int(tok) < 256 for tok in sw[0].split(","))
    assert stopwords_to_obj(sw) == obj


def test_serialize_torch_obj_envelope():
    model = Net()
    s = serialize_torch_obj(model, nn.MSELoss(), torch.optim.Adam, lr=0.001)
    d = json.loads(s)
    assert set(d.keys()) == {"torch_obj", "shapes"}
    # shapes match parameters: fc1 w/b, fc2 w/b
    assert d["shapes"] == [[20, 10], [20], [1, 20], [1]]


def test_load_torch_model_eager():
    model = Net()
    s = serialize_torch_obj(model, nn.MSELoss(), torch.optim.Adam, lr=0.005)
    loaded = load_torch_model(s, from_json=True)
    assert isinstance(loaded.model, Net)
    assert isinstance(loaded.criterion, nn.MSELoss)
    # dill round-trips torch classes by value, so compare by name
    assert type(loaded.optimizer).__name__ == "Adam"
    assert loaded.optimizer.defaults["lr"] == 0.005
    # weights survive the round trip
    x = torch.randn(4, 10)
    assert torch.allclose(model(x), loaded.model(x))


def test_serialize_lazy_with_params():
    s = serialize_torch_obj_lazy(
        NetworkWithParameters,
        nn.MSELoss,
        torch.optim.SGD,
        optimizer_params={"lr": 0.01},
        model_parameters={"input_dim": 10, "hidden_dim": 30, "output_dim": 2},
    )
    env, shapes = load_base_torch(s)
    assert shapes == [[30, 10], [30], [2, 30], [2]]
    loaded = load_torch_model(s, from_json=True)
    assert loaded.model.fc1.out_features == 30
    assert type(loaded.criterion).__name__ == "MSELoss"
    assert type(loaded.optimizer).__name__ == "SGD"


def test_lazy_no_params():
    s = serialize_torch_obj_lazy(Net, nn.MSELoss, torch.optim.Adam, optimizer_params={"lr": 0.1})
    loaded = load_torch_model(s, from_json=True)
    assert isinstance(loaded.model, Net)


def test_fuzz_serialize_roundtrip_random_models():
    """Random small architectures through the full envelope: serialize ->
    load -> state_dict equality -> pipeline stopwords codec round-trip."""
    import random

    import torch
    import torch.nn as nn

    from sparktorch_amd.utils.codec import obj_to_stopwords, stopwords_to_obj
    from sparktorch_amd.utils.serialize import load_torch_model, serialize_torch_obj

    for seed in range(5):
        rng = random.Random(seed)
        torch.manual_seed(seed)
        dims = [rng.randint(2, 17) for _ in range(rng.randint(2, 4))]
        layers = []
        for a, b in zip(dims, dims[1:]):
            layers.append(nn.Linear(a, b))
            if rng.random() < 0.5:
                layers.append(nn.ReLU())
        net = nn.Sequential(*layers)
        crit = rng.choice([nn.MSELoss(), nn.CrossEntropyLoss()])
        obj = serialize_torch_obj(net, crit, torch.optim.Adam, lr=10 ** -rng.randint(2, 4))
        loaded = load_torch_model(obj, from_json=True)
        sd0, sd1 = net.state_dict(), loaded.model.state_dict()
        assert sd0.keys() == sd1.keys()
        for k in sd0:
            assert torch.equal(sd0[k], sd1[k]), (seed, k)
        # checkpoint carrier codec round-trip of the serialized object
        words = obj_to_stopwords(obj)
        assert stopwords_to_obj(words) == obj


def test_nested_pipeline_unwrap():
    """Carrier stages inside a nested pipeline are unwrapped recursively
    (reference pipeline_util.py:66-67)."""
    import torch
    import torch.nn as nn

    from sparktorch_amd import (
        LocalPipeline,
        PysparkPipelineWrapper,
        SparkTorchModel,
        create_spark_torch_model,
    )
    from sparktorch_amd.pipeline_util import _CarrierStage
    from sparktorch_amd.utils.codec import obj_to_stopwords

    net = nn.Linear(4, 2)
    m = create_spark_torch_model(net, inputCol="f", predictionCol="p")
    carrier = _CarrierStage(obj_to_stopwords(m))
    inner = LocalPipeline(stages=[carrier])
    outer = LocalPipeline(stages=[inner, _CarrierStage(obj_to_stopwords(m))])

    unwrapped = PysparkPipelineWrapper.unwrap(outer)
    assert isinstance(unwrapped.stages[0].stages[0], SparkTorchModel)
    assert isinstance(unwrapped.stages[1], SparkTorchModel)
    sd0 = net.state_dict()
    sd1 = unwrapped.stages[1].getPytorchModel().state_dict()
    assert all(torch.equal(sd0[k], sd1[k]) for k in sd0)
