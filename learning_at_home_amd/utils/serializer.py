"""Tensor and object (de)serialization for the control plane.

Reference parity: `PytorchSerializer.dumps/loads` (SURVEY.md §2 "Tensor
serialization" — pickle/msgpack based, ~100 LoC). MI355X note: this is
control-plane only — the intra-node tensor plane is device-resident RCCL
buffers (learning_at_home_amd/parallel) and never touches this path.

Wire format for tensors: msgpack ExtType wrapping
  [dtype_str, shape, requires_grad, raw_little_endian_bytes]
Raw bytes come from a contiguous CPU view (zero copies beyond the one
unavoidable host staging for cross-host RPC).
"""
from __future__ import annotations

import io
from typing import Any

import msgpack
import numpy as np
import torch

_TENSOR_EXT = 0x54  # 'T'

# torch dtypes without a numpy equivalent are round-tripped through a raw
# uint8 view (bfloat16 being the one that matters).
_RAW_VIEW_DTYPES = {
    torch.bfloat16: "bfloat16",
    torch.float8_e4m3fn: "float8_e4m3fn",
    torch.float8_e5m2: "float8_e5m2",
}
_RAW_VIEW_NAMES = {v: k for k, v in _RAW_VIEW_DTYPES.items()}


def _pack_tensor(t: torch.Tensor) -> bytes:
    requires_grad = bool(t.requires_grad)
    t = t.detach()
    if t.device.type != "cpu":
        t = t.cpu()
    t = t.contiguous()
    if t.dtype in _RAW_VIEW_DTYPES:
        dtype_name = _RAW_VIEW_DTYPES[t.dtype]
        # reshape(-1) first: .view(uint8) rejects 0-dim tensors
        raw = t.reshape(-1).view(torch.uint8).numpy().tobytes()
    else:
        dtype_name = str(t.dtype).replace("torch.", "")
        raw = t.numpy().tobytes()
    return msgpack.packb([dtype_name, list(t.shape), requires_grad, raw], use_bin_type=True)


def _unpack_tensor(data: bytes) -> torch.Tensor:
    dtype_name, shape, requires_grad, raw = msgpack.unpackb(data, raw=False)
    if dtype_name in _RAW_VIEW_NAMES:
        torch_dtype = _RAW_VIEW_NAMES[dtype_name]
        if len(raw) == 0:  # frombuffer rejects empty buffers
            flat = torch.empty(0, dtype=torch_dtype)
        else:
            flat = torch.frombuffer(bytearray(raw), dtype=torch.uint8).view(torch_dtype)
    elif len(raw) == 0:
        flat = torch.empty(0, dtype=getattr(torch, dtype_name))
    else:
        np_arr = np.frombuffer(raw, dtype=np.dtype(dtype_name)).copy()
        flat = torch.from_numpy(np_arr)
    out = flat.reshape(shape)
    out.requires_grad_(requires_grad)
    return out


def _default(obj: Any):
    if isinstance(obj, torch.Tensor):
        return msgpack.ExtType(_TENSOR_EXT, _pack_tensor(obj))
    if isinstance(obj, np.ndarray):
        return msgpack.ExtType(_TENSOR_EXT, _pack_tensor(torch.from_numpy(obj)))
    if isinstance(obj, (np.integer,)):
        return int(obj)
    if isinstance(obj, (np.floating,)):
        return float(obj)
    raise TypeError(f"cannot serialize object of type {type(obj)}")


def _ext_hook(code: int, data: bytes):
    if code == _TENSOR_EXT:
        return _unpack_tensor(data)
    return msgpack.ExtType(code, data)


class MSGPackSerializer:
    """msgpack for plain control-plane values (DHT keys/values, metadata)."""

    @staticmethod
    def dumps(obj: Any) -> bytes:
        return msgpack.packb(obj, use_bin_type=True, default=_default)

    @staticmethod
    def loads(buf: bytes) -> Any:
        return msgpack.unpackb(buf, raw=False, strict_map_key=False, ext_hook=_ext_hook)


class PytorchSerializer:
    """Tensors and nested python structures of tensors ⇄ bytes.

    Uses msgpack + raw tensor payloads (not pickle): faster, no arbitrary
    code execution on loads, and dtype/shape/requires_grad preserved.
    Falls back to torch.save for objects msgpack can't express.
    """

    @staticmethod
    def dumps(obj: Any) -> bytes:
        try:
            return b"M" + MSGPackSerializer.dumps(obj)
        except TypeError:
            buf = io.BytesIO()
            torch.save(obj, buf)
            return b"P" + buf.getvalue()

    @staticmethod
    def loads(buf: bytes) -> Any:
        tag, body = buf[:1], buf[1:]
        if tag == b"M":
            return MSGPackSerializer.loads(body)
        elif tag == b"P":
            return torch.load(io.BytesIO(body), weights_only=False)
        raise ValueError(f"unknown serialization tag {tag!r}")
