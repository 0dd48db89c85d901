from sparktorch_amd.utils.codec import b64_to_obj, obj_to_b64
from sparktorch_amd.utils.data import handle_data, handle_features
from sparktorch_amd.utils.early_stopper import EarlyStopping
from sparktorch_amd.utils.rw_lock import RWLock
from sparktorch_amd.utils.serialize import (
    DataObj,
    LoadedTorch,
    TorchObj,
    load_base_torch,
    load_optimizer,
    load_torch_model,
    serialize_torch_obj,
    serialize_torch_obj_lazy,
)
