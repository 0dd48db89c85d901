"""Inference helpers.  Parity with reference sparktorch/inference.py:10-61."""

from __future__ import annotations

import torch

from sparktorch_amd.torch_distributed import SparkTorchModel
from sparktorch_amd.utils.codec import obj_to_b64


def convert_to_serialized_torch(network: torch.nn.Module) -> str:
    """dill+base64 a trained network (reference inference.py:10-17)."""
    return obj_to_b64(network)


def create_spark_torch_model(
    network: torch.nn.Module,
    inputCol: str = "features",
    predictionCol: str = "predicted",
    useVectorOut: bool = False,
) -> SparkTorchModel:
    """Wrap a pretrained net as a transformer (reference inference.py:20-40)."""
    return SparkTorchModel(
        inputCol=inputCol,
        predictionCol=predictionCol,
        modStr=convert_to_serialized_torch(network),
        useVectorOut=useVectorOut,
    )


def attach_pytorch_model_to_pipeline(
    network: torch.nn.Module,
    pipeline_model,
    inputCol: str = "features",
    predictionCol: str = "predicted",
    useVectorOut: bool = False,
):
    """Append a pretrained net to an existing fitted pipeline
    (reference inference.py:43-61)."""
    stage = create_spark_torch_model(network, inputCol, predictionCol, useVectorOut)
    if hasattr(pipeline_model, "stages"):
        pipeline_model.stages.append(stage)
    else:  # pragma: no cover
        raise ValueError("pipeline_model has no stages attribute")
    return pipeline_model
