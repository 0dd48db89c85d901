"""Binary tensor wire format for the parameter server.

Replaces the reference's dill-serialized state_dict/gradient HTTP payloads
(hogwild.py:31-49, server.py:95-149) with a zero-pickle binary codec:
``[magic u32][count u32]`` then per tensor ``[dtype u8][ndim u8][pad u16]
[shape u64 x ndim][nbytes u64][raw bytes]``.  State dicts prepend a JSON key
table.  Raw bytes move via uint8 views, so bf16/fp16 round-trip without numpy
dtype support; decode makes exactly one copy per tensor out of the request
buffer (frombuffer + copy) so returned tensors own writable memory.
"""

from __future__ import annotations

import json
import struct
from typing import Dict, List, Sequence

import numpy as np
import torch

MAGIC = 0x53545731  # 'STW1'

_DTYPE_CODE = {
    torch.float32: 0,
    torch.float64: 1,
    torch.int64: 2,
    torch.int32: 3,
    torch.float16: 4,
    torch.bfloat16: 5,
    torch.uint8: 6,
    torch.bool: 7,
    torch.int16: 8,
    torch.int8: 9,
}
_CODE_DTYPE = {v: k for k, v in _DTYPE_CODE.items()}


def encode_tensors(tensors: Sequence[torch.Tensor]) -> bytes:
    parts: List[bytes] = [struct.pack("<II", MAGIC, len(tensors))]
    for t in tensors:
        t = t.detach()
        if t.is_cuda:
            t = t.cpu()
        t = t.contiguous()
        raw = t.reshape(-1).view(torch.uint8).numpy().tobytes() if t.numel() else b""
        parts.append(struct.pack("<BBH", _DTYPE_CODE[t.dtype], t.dim(), 0))
        parts.append(struct.pack("<%dQ" % t.dim(), *t.shape) if t.dim() else b"")
        parts.append(struct.pack("<Q", len(raw)))
        parts.append(raw)
    return b"".join(parts)


def decode_tensors(data: bytes) -> List[torch.Tensor]:
    magic, count = struct.unpack_from("<II", data, 0)
    if magic != MAGIC:
        raise ValueError("bad wire magic")
    off = 8
    out: List[torch.Tensor] = []
    for _ in range(count):
        code, ndim, _pad = struct.unpack_from("<BBH", data, off)
        off += 4
        shape = struct.unpack_from("<%dQ" % ndim, data, off) if ndim else ()
        off += 8 * ndim
        (nbytes,) = struct.unpack_from("<Q", data, off)
        off += 8
        dtype = _CODE_DTYPE[code]
        if nbytes:
            arr = np.frombuffer(data, dtype=np.uint8, count=nbytes, offset=off).copy()
            t = torch.from_numpy(arr).view(dtype).reshape(shape)
        else:
            t = torch.empty(shape, dtype=dtype)
        off += nbytes
        out.append(t)
    return out


def encode_state_dict(sd: Dict[str, torch.Tensor]) -> bytes:
    keys = list(sd.keys())
    header = json.dumps(keys).encode("utf-8")
    body = encode_tensors([sd[k] for k in keys])
    return struct.pack("<I", len(header)) + header + body


def decode_state_dict(data: bytes) -> Dict[str, torch.Tensor]:
    (hlen,) = struct.unpack_from("<I", data, 0)
    keys = json.loads(data[4 : 4 + hlen].decode("utf-8"))
    tensors = decode_tensors(data[4 + hlen :])
    return dict(zip(keys, tensors))
