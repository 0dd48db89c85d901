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
