"""Property-based round-trip tests (hypothesis) for the wire formats.

These guard the two codecs whose corruption would be silent: the PS binary
tensor wire (parallel/wire.py) and the pipeline checkpoint stopWords codec
(utils/codec.py).
"""

import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from sparktorch_amd.parallel.wire import (
    decode_state_dict,
    decode_tensors,
    encode_state_dict,
    encode_tensors,
)
from sparktorch_amd.utils.codec import obj_to_stopwords, stopwords_to_obj

DTYPES = [torch.float32, torch.float64, torch.int64, torch.int32,
          torch.float16, torch.bfloat16, torch.uint8, torch.bool, torch.int16]


@st.composite
def tensors(draw):
    dtype = draw(st.sampled_from(DTYPES))
    ndim = draw(st.integers(0, 4))
    shape = tuple(draw(st.integers(0, 5)) for _ in range(ndim))
    n = int(np.prod(shape)) if shape else 1
    if dtype == torch.bool:
        t = torch.randint(0, 2, (n,), dtype=torch.uint8).bool()
    elif dtype in (torch.int64, torch.int32, torch.int16, torch.uint8):
        t = torch.randint(0, 100, (n,), dtype=torch.int64).to(dtype)
    else:
        t = torch.randn(n).to(dtype)
    return t.reshape(shape)


@settings(max_examples=60, deadline=None)
@given(st.lists(tensors(), min_size=0, max_size=6))
def test_wire_tensor_roundtrip(ts):
    out = decode_tensors(encode_tensors(ts))
    assert len(out) == len(ts)
    for a, b in zip(ts, out):
        assert a.dtype == b.dtype and a.shape == b.shape
        assert torch.equal(a, b)
        b.add_(0) if b.is_floating_point() else b  # decoded tensors are writable


@settings(max_examples=30, deadline=None)
@given(st.dictionaries(st.text(min_size=1, max_size=12), tensors(), max_size=5))
def test_wire_state_dict_roundtrip(sd):
    out = decode_state_dict(encode_state_dict(sd))
    assert list(out.keys()) == list(sd.keys())
    for k in sd:
        assert torch.equal(out[k], sd[k])


@settings(max_examples=40, deadline=None)
@given(
    st.recursive(
        st.none() | st.booleans() | st.integers() | st.floats(allow_nan=False) | st.text(max_size=20),
        lambda c: st.lists(c, max_size=4) | st.dictionaries(st.text(max_size=8), c, max_size=4),
        max_leaves=12,
    )
)
def test_stopwords_codec_roundtrip(obj):
    sw = obj_to_stopwords(obj)
    assert isinstance(sw, list) and len(sw) == 2
    assert stopwords_to_obj(sw) == obj


@settings(max_examples=40, deadline=None)
@given(
    st.integers(1, 40),           # rows
    st.integers(1, 12),           # feature dim
    st.sampled_from(["f64", "f32", "list"]),
    st.booleans(),                # scalar labels or vector labels
)
def test_handle_features_pack_invariants(n, dim, kind, scalar_labels):
    from sparktorch_amd.utils.data import handle_features
    from sparktorch_amd.utils.serialize import DataObj

    rng = np.random.default_rng(n * 100 + dim)
    rows = []
    for i in range(n):
        x = rng.normal(size=dim)
        if kind == "f32":
            x = x.astype(np.float32)
        elif kind == "list":
            x = list(x)
        y = float(i) if scalar_labels else rng.normal(size=3).astype(np.float32)
        rows.append(DataObj(x, y, None, None))

    d = handle_features(rows, 0.0)
    assert d.x_train.shape == (n, dim) and d.x_train.dtype == torch.float32
    assert d.y_train.shape == (n, 1 if scalar_labels else 3)
    # row order and values preserved (fp32 rounding only)
    for i in (0, n - 1):
        assert np.allclose(d.x_train[i].numpy(),
                           np.asarray(rows[i].x_train, dtype=np.float32), atol=0)


@settings(max_examples=20, deadline=None)
@given(st.integers(4, 60), st.floats(0.05, 0.6))
def test_handle_features_validation_split_partitions(n, pct):
    from sparktorch_amd.utils.data import handle_features
    from sparktorch_amd.utils.serialize import DataObj

    rng = np.random.default_rng(7)
    rows = [DataObj(rng.normal(size=5), 1.0, None, None) for _ in range(n)]
    d = handle_features(rows, pct)
    n_val = int(n * pct)
    if n_val == 0:
        assert d.x_val is None
        assert d.x_train.shape[0] == n
    else:
        assert d.x_val.shape[0] == n_val
        assert d.x_train.shape[0] == n - n_val
