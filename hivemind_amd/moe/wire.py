"""Fast tensor wire format for expert calls.

The generic RPC path wraps tensor buffers in msgpack (2 extra copies per
direction at multi-MB activation sizes). Expert forward/backward instead pack
tensors as ``[4B meta_len][msgpack meta][concat raw buffers]`` where the raw
section is assembled from zero-copy memoryviews -- exactly one copy on send
(the socket gather) and one on receive (frombuffer + torch copy).
"""

from __future__ import annotations

import struct
from typing import List, Sequence, Tuple

import numpy as np
import torch

from ..utils.serializer import MSGPackSerializer

_DTYPE_CODES = {
    torch.float32: "f4",
    torch.float16: "f2",
    torch.bfloat16: "bf",
    torch.int64: "i8",
    torch.int32: "i4",
    torch.int8: "i1",
    torch.uint8: "u1",
    torch.bool: "b1",
}
_CODES_DTYPE = {v: k for k, v in _DTYPE_CODES.items()}


def pack_tensors(uid: str, tensors: Sequence[torch.Tensor]) -> bytes:
    metas = []
    views: List[memoryview] = []
    for t in tensors:
        t = t.detach().contiguous()
        if t.device.type != "cpu":
            t = t.cpu()
        if t.dtype == torch.bfloat16:
            arr = t.view(torch.int16).numpy()
        else:
            arr = t.numpy()
        view = memoryview(arr).cast("B")
        metas.append([_DTYPE_CODES[t.dtype], list(t.shape), len(view)])
        views.append(view)
    meta_blob = MSGPackSerializer.dumps([uid, metas])
    return b"".join([struct.pack(">I", len(meta_blob)), meta_blob, *views])


def unpack_tensors(payload: bytes) -> Tuple[str, List[torch.Tensor]]:
    (meta_len,) = struct.unpack(">I", payload[:4])
    uid, metas = MSGPackSerializer.loads(payload[4 : 4 + meta_len])
    tensors = []
    offset = 4 + meta_len
    for code, shape, nbytes in metas:
        dtype = _CODES_DTYPE[code]
        chunk = np.frombuffer(payload, dtype=np.uint8, count=nbytes, offset=offset).copy()
        offset += nbytes
        if dtype == torch.bfloat16:
            tensor = torch.from_numpy(chunk).view(torch.int16).view(torch.bfloat16)
        else:
            tensor = torch.from_numpy(chunk).view(dtype)
        tensors.append(tensor.reshape(shape))
    return uid, tensors
