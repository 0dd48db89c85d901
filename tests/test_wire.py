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


def test_server_error_budget_and_recovery():
    """The PS tolerates up to 10 malformed updates (HTTP 200 'tolerated'),
    surfaces 500 after, and keeps serving parameters throughout (reference
    server.py:141-144)."""
    import urllib.request

    import torch
    import torch.nn as nn

    from sparktorch_amd.compat.local import free_port
    from sparktorch_amd.models.simple_net import Net
    from sparktorch_amd.parallel.server import Server
    from sparktorch_amd.parallel.wire import decode_state_dict
    from sparktorch_amd.utils.serialize import serialize_torch_obj

    obj = serialize_torch_obj(Net(), nn.MSELoss(), torch.optim.Adam, lr=0.01)
    port = free_port()
    srv = Server(obj, port=port, acquire_lock=True, early_stop_patience=-1, window_len=2)
    srv.start_server()
    try:
        base = "http://127.0.0.1:%d" % port
        # wait for the PS subprocess to bind
        import time
        for _ in range(100):
            try:
                with urllib.request.urlopen(base + "/", timeout=2):
                    break
            except Exception:
                time.sleep(0.1)

        def post_garbage():
            req = urllib.request.Request(base + "/update", data=b"not-a-tensor-blob",
                                         method="POST")
            try:
                with urllib.request.urlopen(req, timeout=10) as r:
                    return r.status, r.read()
            except urllib.error.HTTPError as e:
                return e.code, e.read()

        for i in range(10):
            code, body = post_garbage()
            assert code == 200 and body == b"tolerated", (i, code, body)
        code, _ = post_garbage()
        assert code == 500

        # parameters still served after the budget blew
        with urllib.request.urlopen(base + "/parameters", timeout=10) as r:
            sd = decode_state_dict(r.read())
        assert "fc1.weight" in sd or len(sd) > 0
    finally:
        srv.stop_server()
