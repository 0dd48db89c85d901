"""MNIST CNN, sync data-parallel (mirrors reference examples/simple_cnn.py +
cnn_network.py).  On GPU the convs run on the hand-written implicit-GEMM MFMA
path via the module converter."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import numpy as np
import torch
import torch.nn as nn

from sparktorch_amd import SparkTorch, serialize_torch_obj
from sparktorch_amd.compat.local import LocalDataFrame
from sparktorch_amd.models.mnist import MnistCNN


def main():
    rng = np.random.RandomState(0)
    rows = 2000
    feats = rng.rand(rows, 784).astype(np.float64)
    labels = rng.randint(0, 10, rows).astype(np.float64)
    df = LocalDataFrame.from_arrays(feats, list(labels), num_partitions=2)

    torch_obj = serialize_torch_obj(
        MnistCNN(), nn.CrossEntropyLoss(), torch.optim.Adam, lr=0.001
    )
    est = SparkTorch(
        inputCol="features",
        labelCol="label",
        predictionCol="predictions",
        torchObj=torch_obj,
        iters=10,
        miniBatch=128,
        verbose=1,
        device="cuda" if torch.cuda.is_available() else "cpu",
        mode="synchronous",
    )
    model = est.fit(df)
    out = model.transform(df)
    print("predicted", out.count(), "rows; sample:", out.take(3))


if __name__ == "__main__":
    main()
