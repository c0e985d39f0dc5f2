"""Wire format for the RPC layer: msgpack messages with embedded tensors.

The reference speaks protobuf over gRPC (rpc/xla_service.proto); this
build keeps gRPC as the transport (generic method handlers — grpcio is in
the image, protoc codegen is not) and msgpack as the message encoding, with
the same method surface and message fields. Tensors cross
as {dtype, shape, raw little-endian bytes}; large variables are chunked by
the client to stay under message-size limits (the reference's 2 GB
protobuf workaround, client/client.cc:696)."""

from __future__ import annotations

from typing import Any, Dict

import msgpack
import numpy as np
import torch

_DT = {
    "f32": (torch.float32, np.float32),
    "f64": (torch.float64, np.float64),
    "bf16": (torch.bfloat16, np.uint16),
    "f16": (torch.float16, np.float16),
    "i64": (torch.int64, np.int64),
    "i32": (torch.int32, np.int32),
    "u8": (torch.uint8, np.uint8),
    "bool": (torch.bool, np.bool_),
}
_RDT = {v[0]: k for k, v in _DT.items()}


def encode_tensor(t: torch.Tensor) -> Dict[str, Any]:
    t = t.detach().contiguous().cpu()
    dt = _RDT[t.dtype]
    if t.dtype == torch.bfloat16:
        raw = t.view(torch.uint16).numpy().tobytes()
    else:
        raw = t.numpy().tobytes()
    return {"__tensor__": True, "dtype": dt, "shape": list(t.shape),
            "data": raw}


def decode_tensor(d: Dict[str, Any]) -> torch.Tensor:
    tdt, ndt = _DT[d["dtype"]]
    arr = np.frombuffer(d["data"], dtype=ndt).copy()
    t = torch.from_numpy(arr)
    if tdt == torch.bfloat16:
        t = t.view(torch.bfloat16)
    return t.reshape(d["shape"])


def _default(o):
    if isinstance(o, torch.Tensor):
        return encode_tensor(o)
    raise TypeError(type(o))


def pack(msg: Any) -> bytes:
    return msgpack.packb(msg, default=_default, use_bin_type=True)


def _hook(d):
    if d.get("__tensor__"):
        return decode_tensor(d)
    return d


def unpack(b: bytes) -> Any:
    return msgpack.unpackb(b, object_hook=_hook, raw=False,
                           strict_map_key=False)
