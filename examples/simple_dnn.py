"""MNIST MLP end-to-end: serialize -> fit (sync DP) -> transform -> pipeline
save/load.  Mirrors reference examples/simple_dnn.py:21-66, running on the
local no-JVM engine here (swap LocalDataFrame/LocalPipeline for a Spark
DataFrame/Pipeline on a cluster — the estimator code is identical).

Run: python examples/simple_dnn.py [--device cuda:0]
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import argparse
import tempfile

import numpy as np
import torch
import torch.nn as nn

from sparktorch_amd import (
    LocalPipeline,
    LocalPipelineModel,
    PysparkPipelineWrapper,
    SparkTorch,
    serialize_torch_obj,
)
from sparktorch_amd.compat.local import LocalDataFrame
from sparktorch_amd.models.mnist import MnistMLP


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--device", default="cuda:0" if torch.cuda.is_available() else "cpu")
    ap.add_argument("--rows", type=int, default=4000)
    args = ap.parse_args()

    rng = np.random.RandomState(0)
    feats = rng.rand(args.rows, 784).astype(np.float64)
    labels = rng.randint(0, 10, args.rows).astype(np.float64)
    df = LocalDataFrame.from_arrays(feats, list(labels), num_partitions=2)

    network = MnistMLP()
    torch_obj = serialize_torch_obj(
        network, nn.CrossEntropyLoss(), torch.optim.Adam, lr=0.001
    )

    spark_model = SparkTorch(
        inputCol="features",
        labelCol="label",
        predictionCol="predictions",
        torchObj=torch_obj,
        iters=20,
        miniBatch=256,
        earlyStopPatience=40,
        validationPct=0.2,
        verbose=1,
        device=args.device if args.device == "cpu" else "cuda",
        mode="synchronous",
    )

    pipeline = LocalPipeline(stages=[spark_model]).fit(df)

    with tempfile.TemporaryDirectory() as d:
        path = d + "/mnist_model"
        pipeline.write().overwrite().save(path)
        loaded = PysparkPipelineWrapper.unwrap(LocalPipelineModel.load(path))

    out = loaded.transform(df)
    preds = [r["predictions"] for r in out.collect()]
    acc = np.mean([p == l for p, l in zip(preds, labels)])
    print("rows=%d  example accuracy (random data): %.3f" % (len(preds), acc))


if __name__ == "__main__":
    main()
