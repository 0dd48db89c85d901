"""Lazy serialization: classes + ctor kwargs travel to workers, the model is
instantiated there (avoids driver memory for big nets).  Mirrors reference
examples/lazy_load_cnn.py."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import numpy as np
import torch
import torch.nn as nn

from sparktorch_amd import SparkTorch, serialize_torch_obj_lazy
from sparktorch_amd.compat.local import LocalDataFrame
from sparktorch_amd.models.mnist import MnistCNN


def main():
    rng = np.random.RandomState(0)
    rows = 1000
    df = LocalDataFrame.from_arrays(
        rng.rand(rows, 784).astype(np.float64),
        list(rng.randint(0, 10, rows).astype(np.float64)),
        num_partitions=2,
    )

    torch_obj = serialize_torch_obj_lazy(
        MnistCNN,
        nn.CrossEntropyLoss,
        torch.optim.Adam,
        optimizer_params={"lr": 0.001},
    )
    est = SparkTorch(
        inputCol="features",
        labelCol="label",
        predictionCol="predictions",
        torchObj=torch_obj,
        iters=5,
        miniBatch=128,
        mode="synchronous",
    )
    model = est.fit(df)
    print("trained + transformed:", model.transform(df).count(), "rows")


if __name__ == "__main__":
    main()
