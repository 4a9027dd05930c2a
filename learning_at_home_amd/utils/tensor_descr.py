"""Tensor schema descriptors for expert I/O (SURVEY.md §2 ExpertBackend
`get_info` — clients fetch these to build correctly-shaped RPC payloads)."""
from __future__ import annotations

from dataclasses import dataclass
from typing import Tuple

import torch

_DTYPES = {str(dt).replace("torch.", ""): dt for dt in (
    torch.float32, torch.float64, torch.float16, torch.bfloat16,
    torch.int64, torch.int32, torch.int16, torch.int8, torch.uint8, torch.bool,
)}


@dataclass(frozen=True)
class TensorDescriptor:
    shape: Tuple[int, ...]
    dtype: str = "float32"
    requires_grad: bool = False

    @classmethod
    def from_tensor(cls, t: torch.Tensor) -> "TensorDescriptor":
        return cls(tuple(t.shape), str(t.dtype).replace("torch.", ""), t.requires_grad)

    @property
    def torch_dtype(self) -> torch.dtype:
        return _DTYPES[self.dtype]

    def make_empty(self, **kwargs) -> torch.Tensor:
        return torch.empty(self.shape, dtype=self.torch_dtype, **kwargs).requires_grad_(self.requires_grad)

    def to_dict(self) -> dict:
        return {"shape": list(self.shape), "dtype": self.dtype, "requires_grad": self.requires_grad}

    @classmethod
    def from_dict(cls, d: dict) -> "TensorDescriptor":
        return cls(tuple(d["shape"]), d["dtype"], d["requires_grad"])


@dataclass(frozen=True)
class BatchTensorDescriptor(TensorDescriptor):
    """Descriptor whose leading dimension is the (variable) batch axis; shape
    stores the per-sample trailing dims with -1 in front."""

    @classmethod
    def from_tensor(cls, t: torch.Tensor) -> "BatchTensorDescriptor":
        return cls((-1, *t.shape[1:]), str(t.dtype).replace("torch.", ""), t.requires_grad)

    @classmethod
    def make(cls, *trailing_dims: int, dtype: str = "float32", requires_grad: bool = False):
        return cls((-1, *trailing_dims), dtype, requires_grad)

    def make_batch(self, batch_size: int, **kwargs) -> torch.Tensor:
        # zeros, not empty: this feeds schema-inference dummy forwards, and
        # uninitialized garbage can be huge/NaN and break arbitrary experts
        shape = (batch_size, *self.shape[1:])
        return torch.zeros(shape, dtype=self.torch_dtype, **kwargs).requires_grad_(self.requires_grad)
