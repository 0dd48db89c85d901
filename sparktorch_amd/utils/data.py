"""Data layer: row -> DataObj -> stacked device tensors.

Parity with reference:
  * ``handle_features``  (util.py:58-101) — list of per-row DataObj to stacked
    float32 train/validation tensors, random validation split, scalar labels
    wrapped, empty partition -> all-None.
  * ``handle_data``      (torch_distributed.py:44-56) — row mapper building
    per-row DataObj from a DataFrame row's feature vector + label.

MI355X-native difference: instead of ``np.stack`` over thousands of per-row
float64 arrays followed by a CPU cast (reference util.py:87-99), we
pre-allocate one contiguous float32 matrix and fill it row-wise; on a GPU
worker the engine then does a single pinned H2D copy.  The per-row DataObj
shape is preserved because it is the cross-partition wire format.
"""

from __future__ import annotations

from typing import Iterable, List, Optional

import numpy as np
import torch

from sparktorch_amd.utils.serialize import DataObj


def _to_numpy_features(x) -> np.ndarray:
    """Feature field of a row -> 1-D numpy array (Spark Vector, list, or ndarray)."""
    if hasattr(x, "toArray"):  # pyspark.ml.linalg Vector
        return np.asarray(x.toArray())
    return np.asarray(x)


def handle_data(inp_col: str, label_col: Optional[str]):
    """Return a partition mapper: rows -> per-row DataObj.

    Reference torch_distributed.py:44-56.  ``label_col=None`` means
    autoencoder mode downstream (y_train stays None here; the engine sets
    y=x, reference distributed.py:136).
    """

    def _map(partition) -> Iterable[DataObj]:
        for row in partition:
            x = _to_numpy_features(row[inp_col])
            y = row[label_col] if label_col is not None else None
            yield DataObj(x_train=x, y_train=y, x_val=None, y_val=None)

    return _map


def pack_partition(data: List[DataObj]):
    """Stack a partition of per-row DataObj into (X, y) float32 arrays.

    Single contiguous allocation + row-fill instead of np.stack of N object
    arrays; Spark DenseVector rows arrive float64, cast once here.
    """
    n = len(data)
    if n == 0:
        return None, None
    dim = int(np.asarray(data[0].x_train).size)
    x = np.empty((n, dim), dtype=np.float32)
    have_labels = data[0].y_train is not None
    y_rows: list = []
    for i, d in enumerate(data):
        x[i, :] = np.asarray(d.x_train, dtype=np.float32).reshape(-1)
        if have_labels:
            yi = d.y_train
            # scalar labels wrapped like reference util.py:72-74
            y_rows.append([yi] if np.isscalar(yi) else np.asarray(yi, dtype=np.float32).reshape(-1))
    y = np.asarray(y_rows, dtype=np.float32) if have_labels else None
    return x, y


def handle_features_device(
    data: List[DataObj],
    validation_pct: float = 0.0,
    device: str = "cuda:0",
    dtype=None,
) -> DataObj:
    """GPU ingest: partition rows -> device tensors with the cast ON DEVICE.

    Replaces the reference's CPU pack (np.stack of fp64 rows + ``.float()``
    cast + blocking ``.to(device)``, reference util.py:87-99) with the
    MI355X path promised in SURVEY.md §2.2:

      np.stack (C-level row gather, fp64 kept as-is, no CPU cast pass)
      -> pinned staging tensor -> hipMemcpyAsync H2D
      -> one ``cast_f64_bf16`` kernel (8 TB/s HBM vs a CPU cast pass)
      -> optional on-device validation split by index_select.

    Rows that already arrive fp32 skip the cast kernel.  Returns a DataObj of
    DEVICE tensors: x in ``dtype`` (default bf16), y fp32 (downstream losses
    cast labels as needed).
    """
    import torch as _torch

    dtype = dtype or _torch.bfloat16
    data = list(data)
    if len(data) == 0:
        return DataObj(None, None, None, None)

    from sparktorch_amd import ops as _ops

    ext = _ops.ext()

    first = np.asarray(data[0].x_train)
    if first.ndim == 1:
        # fast path: hand np.stack the row objects directly — a per-row
        # asarray().reshape() list comprehension costs ~0.5 us x row and
        # dominated the 131072-row ingest
        rows = [d.x_train for d in data]
    else:
        rows = [np.asarray(d.x_train).reshape(-1) for d in data]
    # stack straight INTO the pinned staging tensor: one C-level gather pass,
    # no separate pin_memory() copy of the whole matrix afterwards
    t_dtype = {np.dtype(np.float64): _torch.float64,
               np.dtype(np.float32): _torch.float32}.get(first.dtype)
    if t_dtype is None:
        rows = [np.asarray(r, dtype=np.float32).reshape(-1) for r in rows]
        t_dtype = _torch.float32
    x_pin = _torch.empty((len(rows), first.size), dtype=t_dtype, pin_memory=True)
    np.stack(rows, out=x_pin.numpy())
    x_dev = x_pin.to(device, non_blocking=True)  # hipMemcpyAsync from pinned
    if x_dev.dtype == _torch.float64:
        x = ext.cast_f64_bf16(x_dev) if dtype == _torch.bfloat16 else ext.cast_f64_f32(x_dev)
    elif x_dev.dtype != dtype:
        x = ext.cast_f32_bf16(x_dev) if dtype == _torch.bfloat16 else x_dev.to(dtype)
    else:
        x = x_dev

    y = None
    if data[0].y_train is not None:
        if np.isscalar(data[0].y_train):
            # C-loop over the scalar labels (reference wraps each as [y],
            # util.py:72-74 — same [n,1] shape, no per-row python list)
            y_host = np.fromiter(
                (d.y_train for d in data), dtype=np.float32, count=len(data)
            ).reshape(-1, 1)
        else:
            y_host = np.asarray(
                [np.asarray(d.y_train, dtype=np.float32).reshape(-1) for d in data],
                dtype=np.float32,
            )
        y_pin = _torch.from_numpy(y_host).pin_memory()
        y = y_pin.to(device, non_blocking=True)

    n = x.shape[0]
    x_val = y_val = None
    if validation_pct and validation_pct > 0.0 and n > 1:
        n_val = int(n * validation_pct)
        if n_val > 0:
            val_idx = np.random.choice(n, n_val, replace=False)
            mask = np.ones(n, dtype=bool)
            mask[val_idx] = False
            tr = _torch.from_numpy(np.nonzero(mask)[0]).to(device, non_blocking=True)
            va = _torch.from_numpy(np.nonzero(~mask)[0]).to(device, non_blocking=True)
            x_val = x.index_select(0, va)
            x = x.index_select(0, tr)
            if y is not None:
                y_val = y.index_select(0, va)
                y = y.index_select(0, tr)
    _torch.cuda.synchronize()  # pinned buffers die with this frame
    return DataObj(x_train=x, y_train=y, x_val=x_val, y_val=y_val)


def handle_features(data: List[DataObj], validation_pct: float = 0.0) -> DataObj:
    """Partition of row-DataObj -> one DataObj of stacked float32 torch tensors.

    Reference util.py:58-101: random validation split via choice + set
    difference; empty partition returns all-None fields.
    """
    data = list(data)
    if len(data) == 0:
        return DataObj(None, None, None, None)

    x, y = pack_partition(data)
    n = x.shape[0]

    x_val = y_val = None
    if validation_pct and validation_pct > 0.0 and n > 1:
        n_val = int(n * validation_pct)
        if n_val > 0:
            val_idx = np.random.choice(n, n_val, replace=False)
            mask = np.ones(n, dtype=bool)
            mask[val_idx] = False
            x_val = x[~mask]
            x = x[mask]
            if y is not None:
                y_val = y[~mask]
                y = y[mask]

    def _t(a):
        return torch.from_numpy(np.ascontiguousarray(a)) if a is not None else None

    return DataObj(x_train=_t(x), y_train=_t(y), x_val=_t(x_val), y_val=_t(y_val))
