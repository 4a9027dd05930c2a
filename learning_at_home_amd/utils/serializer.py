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

import builtins
import io
from typing import Any

import msgpack
import numpy as np
import torch

_TENSOR_EXT = 0x54  # 'T'
_EXC_EXT = 0x45  # 'E'


class RemoteException(Exception):
    """An exception raised on a remote peer whose type could not be safely
    reconstructed locally. Carries the original type name and message."""

    def __init__(self, type_name: str, message: str):
        super().__init__(f"{type_name}: {message}")
        self.type_name = type_name
        self.message = message

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


def _pack_exception(exc: BaseException) -> bytes:
    return msgpack.packb([type(exc).__name__, str(exc)], use_bin_type=True)


def _unpack_exception(data: bytes) -> BaseException:
    type_name, message = msgpack.unpackb(data, raw=False)
    # Reconstruct only builtin exception types (no arbitrary class lookup):
    # a constructor like ValueError(message) runs no user code.
    cand = getattr(builtins, type_name, None)
    if isinstance(cand, type) and issubclass(cand, BaseException):
        try:
            return cand(message)
        except Exception:  # exotic builtin signature (e.g. UnicodeDecodeError)
            pass
    return RemoteException(type_name, message)


def _default(obj: Any):
    if isinstance(obj, torch.Tensor):
        return msgpack.ExtType(_TENSOR_EXT, _pack_tensor(obj))
    if isinstance(obj, np.ndarray):
        return msgpack.ExtType(_TENSOR_EXT, _pack_tensor(torch.from_numpy(obj)))
    if isinstance(obj, BaseException):
        return msgpack.ExtType(_EXC_EXT, _pack_exception(obj))
    if isinstance(obj, (np.integer,)):
        return int(obj)
    if isinstance(obj, (np.floating,)):
        return float(obj)
    raise TypeError(f"cannot serialize object of type {type(obj)}")


def _ext_hook(code: int, data: bytes):
    if code == _TENSOR_EXT:
        return _unpack_tensor(data)
    if code == _EXC_EXT:
        return _unpack_exception(data)
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

    Uses msgpack + raw tensor payloads: no arbitrary code execution on
    loads, dtype/shape/requires_grad preserved, exceptions round-tripped
    as (type_name, message) pairs — builtin exception types are
    reconstructed, anything else becomes RemoteException.

    The pickle (torch.save) path exists ONLY for trusted same-host
    transports (multiprocessing pipes between our own processes) and is
    disabled by default: `loads` refuses a pickle payload and `dumps`
    refuses to produce one unless `allow_pickle=True` is passed
    explicitly. Network paths (expert RPC, averager sockets, DHT) must
    never pass it — a pickle payload from a peer is arbitrary code
    execution.
    """

    @staticmethod
    def dumps(obj: Any, *, allow_pickle: bool = False) -> bytes:
        try:
            return b"M" + MSGPackSerializer.dumps(obj)
        except TypeError:
            if not allow_pickle:
                raise
            buf = io.BytesIO()
            torch.save(obj, buf)
            return b"P" + buf.getvalue()

    @staticmethod
    def loads(buf: bytes, *, allow_pickle: bool = False) -> Any:
        tag, body = buf[:1], buf[1:]
        if tag == b"M":
            return MSGPackSerializer.loads(body)
        elif tag == b"P":
            if not allow_pickle:
                raise ValueError(
                    "refusing to unpickle a b'P' payload on an untrusted path; "
                    "pass allow_pickle=True only for trusted same-host pipes"
                )
            return torch.load(io.BytesIO(body), weights_only=False)
        raise ValueError(f"unknown serialization tag {tag!r}")
