"""ResNet-18 sync-DP training on synthetic 3x224x224 Vector rows — the
BASELINE.json config-4 shape, run through the full SparkTorch estimator path
(serialize -> fit -> transform) on the local barrier engine.

On a GPU host every op runs on the hand-written CDNA4 kernels (NHWC
implicit-GEMM convs, fused BN+ReLU, fused residual joins, MFMA fc, fused
Adam); on CPU the same model runs through the eager fallback so the example
works anywhere.  ``python bench.py --model resnet18`` is the measured,
timed version of this workload.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import numpy as np
import torch
import torch.nn as nn

from sparktorch_amd import SparkTorch, serialize_torch_obj
from sparktorch_amd.compat.local import LocalDataFrame
from sparktorch_amd.models.resnet import ResNet18


def main():
    rows, classes = 64, 10
    rng = np.random.default_rng(0)
    feats = rng.standard_normal((rows, 3 * 224 * 224)).astype(np.float32) * 0.1
    labels = rng.integers(0, classes, rows).astype(np.float32)
    df = LocalDataFrame.from_arrays(feats, list(labels), num_partitions=2)

    net = ResNet18(num_classes=classes)
    torch_obj = serialize_torch_obj(net, nn.CrossEntropyLoss(), torch.optim.Adam, lr=1e-3)

    device = "cuda" if torch.cuda.is_available() else "cpu"
    spark_model = SparkTorch(
        inputCol="features",
        labelCol="label",
        predictionCol="predictions",
        torchObj=torch_obj,
        iters=4,
        verbose=1,
        miniBatch=16,
        device=device,
        mode="synchronous",
    ).fit(df)

    preds = spark_model.transform(df).collect()
    print("predicted classes (first 10):", [int(r["predictions"]) for r in preds[:10]])


if __name__ == "__main__":
    main()
