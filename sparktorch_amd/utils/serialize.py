"""Torch-object serialization layer.

Parity with reference sparktorch/util.py:
  * ``TorchObj`` / ``DataObj`` namedtuples       (util.py:31-35)
  * ``serialize_torch_obj``                      (util.py:183-202)
  * ``serialize_torch_obj_lazy``                 (util.py:149-180)
  * ``load_base_torch``                          (util.py:104-111)
  * ``load_torch_model``                         (util.py:114-146)
  * ``load_optimizer``                           (util.py:205-209)

The JSON envelope is identical: ``{"torch_obj": <b64(dill(TorchObj))>,
"shapes": [[...], ...]}``.  The ``shapes`` field is carried for format
compatibility (the reference stores but never consumes it).

MI355X-native additions: ``load_torch_model`` accepts a ``device`` and moves the
model there; fused-optimizer classes from :mod:`sparktorch_amd.ops.optim` are
accepted anywhere a torch optimizer class is.
"""

from __future__ import annotations

import json
from collections import namedtuple
from typing import Any, Dict, Optional, Type

import torch
import torch.nn as nn

from sparktorch_amd.utils.codec import b64_to_obj, obj_to_b64

# Field order matches reference util.py:31-35.
TorchObj = namedtuple(
    "TorchObj",
    ["model", "criterion", "optimizer", "optimizer_params", "is_lazy", "model_parameters"],
)

DataObj = namedtuple("DataObj", ["x_train", "y_train", "x_val", "y_val"])

# Hydrated (worker-side) object: instantiated model/criterion/optimizer.
LoadedTorch = namedtuple("LoadedTorch", ["model", "criterion", "optimizer"])


def _param_shapes(model: nn.Module) -> list:
    return [list(p.shape) for p in model.parameters()]


def serialize_torch_obj(
    model: nn.Module,
    criterion: Any,
    optimizer: Type[torch.optim.Optimizer],
    **kwargs,
) -> str:
    """Serialize an *instantiated* model + criterion + optimizer class.

    Reference: util.py:183-202.  ``kwargs`` are the optimizer constructor
    params (e.g. ``lr=0.001``).
    """
    obj = TorchObj(
        model=model,
        criterion=criterion,
        optimizer=optimizer,
        optimizer_params=kwargs,
        is_lazy=False,
        model_parameters=None,
    )
    return json.dumps({"torch_obj": obj_to_b64(obj), "shapes": _param_shapes(model)})


def serialize_torch_obj_lazy(
    model: Type[nn.Module],
    criterion: Type[Any],
    optimizer: Type[torch.optim.Optimizer],
    optimizer_params: Optional[Dict] = None,
    model_parameters: Optional[Dict] = None,
) -> str:
    """Serialize *classes* instead of instances; instantiated on the worker.

    Reference: util.py:149-180.  Avoids holding a large model on the driver; a
    temporary instance is created only to record parameter shapes.
    """
    tmp = model(**model_parameters) if model_parameters else model()
    shapes = _param_shapes(tmp)
    del tmp
    obj = TorchObj(
        model=model,
        criterion=criterion,
        optimizer=optimizer,
        optimizer_params=optimizer_params,
        is_lazy=True,
        model_parameters=model_parameters,
    )
    return json.dumps({"torch_obj": obj_to_b64(obj), "shapes": shapes})


def load_base_torch(serialized: str):
    """JSON envelope -> (b64 TorchObj string, shapes).  Reference util.py:104-111."""
    d = json.loads(serialized)
    return d["torch_obj"], d["shapes"]


def load_optimizer(
    optimizer_cls: Type[torch.optim.Optimizer],
    model: nn.Module,
    optimizer_params: Optional[Dict],
):
    """Bind an optimizer class to a model's parameters.  Reference util.py:205-209."""
    params = optimizer_params or {}
    return optimizer_cls(model.parameters(), **params)


def load_torch_model(
    serialized: str,
    from_json: bool = False,
    device: Optional[str] = None,
) -> LoadedTorch:
    """Deserialize a TorchObj and hydrate model/criterion/optimizer.

    Reference: util.py:114-146.  Lazy objects instantiate their classes here
    (util.py:126-135); eager objects arrive fully built.  The optimizer is
    always (re)bound to the hydrated model's parameters.
    """
    if from_json:
        serialized, _ = load_base_torch(serialized)
    obj: TorchObj = b64_to_obj(serialized)

    if obj.is_lazy:
        model = obj.model(**obj.model_parameters) if obj.model_parameters else obj.model()
        criterion = obj.criterion() if isinstance(obj.criterion, type) else obj.criterion
    else:
        model = obj.model
        criterion = obj.criterion

    if device is not None:
        model = model.to(device)

    optimizer = load_optimizer(obj.optimizer, model, obj.optimizer_params)
    return LoadedTorch(model=model, criterion=criterion, optimizer=optimizer)
