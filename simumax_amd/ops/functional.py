"""Shape-manipulation modules: Concat / Split / Unsqueeze / AddFunction.

Parity target: simumax/core/transformer/function.py:10-110 and
simu_ops.py:5-44 — zero-flop ops that still move bytes and participate in
shape propagation.
"""

from __future__ import annotations

from ..core.module import MetaModule
from ..core.records import InputOutputInfo
from ..core.tensor import TensorSize
from .dense import Add  # noqa: F401  (re-export: Add lives with dense ops)


class Concat(MetaModule):
    """Concatenate the input tensors along `dim` (copies bytes)."""

    def __init__(self, strategy, system, dim=-1, name="concat"):
        super().__init__(strategy, system, name)
        self.dim = dim

    def create_output_info(self, input_info):
        ts = input_info.tensors
        dim = self.dim if self.dim >= 0 else ts[0].ndim + self.dim
        shape = list(ts[0].shape)
        shape[dim] = sum(t.shape[dim] for t in ts)
        return InputOutputInfo([TensorSize(shape, ts[0].dtype)])

    def _leaf_compute_info(self, info):
        b = self.input_info.total_bytes()
        info.fwd_accessed_mem = 2 * b          # read all + write out
        info.bwd_grad_act_accessed_mem = 2 * b  # split of the grad


class Split(MetaModule):
    """Split the input into `chunks` along `dim` (views: free fwd; the
    backward concat copies)."""

    def __init__(self, strategy, system, chunks, dim=-1, name="split"):
        super().__init__(strategy, system, name)
        self.chunks = chunks
        self.dim = dim

    def create_output_info(self, input_info):
        t = input_info.tensors[0]
        dim = self.dim if self.dim >= 0 else t.ndim + self.dim
        return InputOutputInfo(t.chunk(self.chunks, dim))

    def _leaf_compute_info(self, info):
        b = self.input_info.first.mem_bytes()
        info.bwd_grad_act_accessed_mem = 2 * b  # grad concat


class Unsqueeze(MetaModule):
    def __init__(self, strategy, system, dim, name="unsqueeze"):
        super().__init__(strategy, system, name)
        self.dim = dim

    def create_output_info(self, input_info):
        return InputOutputInfo([input_info.first.unsqueeze(self.dim)])


class AddFunction:
    """Function.apply-style helper (reference simu_ops parity): adds two
    shape-identical tensors through an Add module instance."""

    @staticmethod
    def apply(strategy, system, a: TensorSize, b: TensorSize,
              dbg=None) -> TensorSize:
        assert a.shape == b.shape, f"add shape mismatch {a.shape} vs {b.shape}"
        mod = Add(strategy, system)
        return mod(InputOutputInfo([a, b]), dbg).tensors[0]
