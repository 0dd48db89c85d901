"""sparktorch_amd — an MI355X-native distributed PyTorch training bridge with
the capabilities (and public API) of dmmiller612/sparktorch.

Public surface parity: reference sparktorch/__init__.py:1-4 plus the helper
classes the reference exports from submodules.
"""

from sparktorch_amd.inference import (
    attach_pytorch_model_to_pipeline,
    convert_to_serialized_torch,
    create_spark_torch_model,
)
from sparktorch_amd.pipeline_util import (
    LocalPipeline,
    LocalPipelineModel,
    PysparkPipelineWrapper,
    PysparkReaderWriter,
)
from sparktorch_amd.torch_distributed import SparkTorch, SparkTorchModel
from sparktorch_amd.utils.early_stopper import EarlyStopping
from sparktorch_amd.utils.rw_lock import RWLock
from sparktorch_amd.utils.serialize import (
    DataObj,
    TorchObj,
    serialize_torch_obj,
    serialize_torch_obj_lazy,
)

__version__ = "0.1.0"

__all__ = [
    "serialize_torch_obj",
    "serialize_torch_obj_lazy",
    "SparkTorch",
    "SparkTorchModel",
    "PysparkPipelineWrapper",
    "PysparkReaderWriter",
    "LocalPipeline",
    "LocalPipelineModel",
    "create_spark_torch_model",
    "convert_to_serialized_torch",
    "attach_pytorch_model_to_pipeline",
    "EarlyStopping",
    "RWLock",
    "TorchObj",
    "DataObj",
    "__version__",
]
