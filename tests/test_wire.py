import numpy as np
import pytest
import torch

from sparktorch_amd.parallel.wire import (
    decode_state_dict,
    decode_tensors,
    encode_state_dict,
    encode_tensors,
)


@pytest.mark.parametrize(
    "dtype",
    [torch.float32, torch.float64, torch.int64, torch.float16, torch.bfloat16, torch.uint8],
)
def test_tensor_roundtrip_dtypes(dtype):
    if dtype.is_floating_point:
        t = torch.randn(3, 5).to(dtype)
    else:
        t = torch.arange(15, dtype=dtype).reshape(3, 5)
    back = decode_tensors(encode_tensors([t]))[0]
    assert back.dtype == dtype
    assert back.shape == t.shape
    assert torch.equal(back, t)


def test_multiple_tensors_and_scalars():
    ts = [torch.randn(()), torch.randn(4), torch.randn(2, 3, 4)]
    back = decode_tensors(encode_tensors(ts))
    assert len(back) == 3
    for a, b in zip(ts, back):
        assert torch.equal(a, b)


def test_noncontiguous_input():
    t = torch.randn(4, 6).t()
    back = decode_tensors(encode_tensors([t]))[0]
    assert torch.equal(back, t)


def test_state_dict_roundtrip():
    from sparktorch_amd.models.simple_net import Net

    sd = Net().state_dict()
    back = decode_state_dict(encode_state_dict(sd))
    assert list(back.keys()) == list(sd.keys())
    for k in sd:
        assert torch.equal(back[k], sd[k])


def test_bad_magic_raises():
    with pytest.raises(ValueError):
        decode_tensors(b"\x00" * 16)
