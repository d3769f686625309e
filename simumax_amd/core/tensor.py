"""Symbolic tensors: shape + dtype only, no data.

Parity target: simumax/core/tensor.py:14-143 (TensorSize, Float8Tensor) —
a torch-like view over (shape, dtype) used for all activation/weight
accounting in the analytical model.
"""

from __future__ import annotations

import math
from typing import Iterable, List, Tuple

from .consts import DTYPE_BYTES


class TensorSize:
    """A data-free tensor: tracks shape and dtype, supports the handful of
    torch-style ops the op library needs (view/transpose/chunk/numel)."""

    __slots__ = ("_shape", "dtype")

    def __init__(self, shape: Iterable[int], dtype: str = "bf16"):
        self._shape = tuple(int(s) for s in shape)
        assert dtype in DTYPE_BYTES, f"unknown dtype {dtype}"
        self.dtype = dtype

    # ---- shape API -------------------------------------------------
    @property
    def shape(self) -> Tuple[int, ...]:
        return self._shape

    @property
    def ndim(self) -> int:
        return len(self._shape)

    def size(self, dim: int = None):
        if dim is None:
            return self._shape
        return self._shape[dim]

    def dim(self) -> int:
        return len(self._shape)

    def numel(self) -> int:
        return math.prod(self._shape) if self._shape else 0

    @property
    def bytes_per_element(self) -> int:
        return DTYPE_BYTES[self.dtype]

    def mem_bytes(self) -> int:
        return self.numel() * self.bytes_per_element

    # ---- torch-like transforms (all return new TensorSize) ---------
    def view(self, *shape) -> "TensorSize":
        if len(shape) == 1 and isinstance(shape[0], (list, tuple)):
            shape = tuple(shape[0])
        shape = list(shape)
        numel = self.numel()
        if -1 in shape:
            idx = shape.index(-1)
            known = math.prod(s for s in shape if s != -1)
            assert known > 0 and numel % known == 0, f"bad view {shape} of {self._shape}"
            shape[idx] = numel // known
        assert math.prod(shape) == numel, f"view {shape} != numel {numel}"
        return TensorSize(shape, self.dtype)

    reshape = view

    def transpose(self, d0: int, d1: int) -> "TensorSize":
        s = list(self._shape)
        s[d0], s[d1] = s[d1], s[d0]
        return TensorSize(s, self.dtype)

    def unsqueeze(self, dim: int) -> "TensorSize":
        s = list(self._shape)
        if dim < 0:
            dim += len(s) + 1
        s.insert(dim, 1)
        return TensorSize(s, self.dtype)

    def squeeze(self, dim: int) -> "TensorSize":
        s = list(self._shape)
        assert s[dim] == 1
        s.pop(dim)
        return TensorSize(s, self.dtype)

    def chunk(self, chunks: int, dim: int = 0) -> List["TensorSize"]:
        s = list(self._shape)
        assert s[dim] % chunks == 0
        s[dim] //= chunks
        return [TensorSize(s, self.dtype) for _ in range(chunks)]

    def to(self, dtype: str) -> "TensorSize":
        return TensorSize(self._shape, dtype)

    def clone(self) -> "TensorSize":
        return TensorSize(self._shape, self.dtype)

    def scale_dim(self, dim: int, num: int, den: int) -> "TensorSize":
        """Return a copy with shape[dim] scaled by num/den (exact)."""
        s = list(self._shape)
        assert (s[dim] * num) % den == 0, f"scale {s[dim]}*{num}/{den} not exact"
        s[dim] = s[dim] * num // den
        return TensorSize(s, self.dtype)

    def __eq__(self, other):
        return (
            isinstance(other, TensorSize)
            and self._shape == other._shape
            and self.dtype == other.dtype
        )

    def __hash__(self):
        return hash((self._shape, self.dtype))

    def __repr__(self):
        return f"TensorSize({list(self._shape)}, {self.dtype})"


class Float8Tensor(TensorSize):
    """An fp8 tensor that also carries the amax/scale metadata overhead the
    quantized-linear memory model accounts for (transpose cache etc.)."""

    def __init__(self, shape, dtype: str = "fp8", with_transpose_cache: bool = True):
        super().__init__(shape, dtype)
        self.with_transpose_cache = with_transpose_cache

    def mem_bytes(self) -> int:
        base = super().mem_bytes()
        if self.with_transpose_cache:
            base *= 2  # row-wise + column-wise copies kept by fp8 GEMM kernels
        return base
