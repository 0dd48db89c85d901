"""Asynchronous (HOGWILD!) parameter-server training: workers pull params and
push gradients over the binary-wire HTTP PS.  Mirrors the reference's
mode='hogwild' path (README.md hogwild example)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import numpy as np
import torch
import torch.nn as nn

from sparktorch_amd import SparkTorch, serialize_torch_obj
from sparktorch_amd.compat.local import LocalDataFrame, free_port
from sparktorch_amd.models.mnist import MnistMLP


def main():
    rng = np.random.RandomState(0)
    rows = 2000
    df = LocalDataFrame.from_arrays(
        rng.rand(rows, 784).astype(np.float64),
        list(rng.randint(0, 10, rows).astype(np.float64)),
        num_partitions=2,
    )
    torch_obj = serialize_torch_obj(
        MnistMLP(), nn.CrossEntropyLoss(), torch.optim.Adam, lr=0.001
    )
    est = SparkTorch(
        inputCol="features",
        labelCol="label",
        predictionCol="predictions",
        torchObj=torch_obj,
        iters=10,
        verbose=1,
        mode="hogwild",
        acquireLock=False,  # true lock-free hogwild
        port=free_port(),
        earlyStopPatience=20,
    )
    model = est.fit(df)
    print("hogwild trained;", model.transform(df).count(), "rows predicted")


if __name__ == "__main__":
    main()
