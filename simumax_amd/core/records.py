"""Additive per-module accounting records.

Parity target: simumax/core/model_struct.py:15-423 (RecomputeStatus,
InputOutputInfo, ModuleComputeInfo, ActivationInfo, ModuleMemoryInfo,
ModuleCostInfo, PathDebugContext, PointDebugInfo, Result).

All *Info records support `+` so a composite module's record is the sum of
its children's.
"""

from __future__ import annotations

import enum
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from .tensor import TensorSize


class RecomputeStatus(enum.Enum):
    NONE = 0          # activations cached normally
    RECOMPUTE = 1     # inside a recompute segment: nothing cached, re-forwarded in bwd
    FIRST = 2         # first module of a segment: caches segment input only
    LAST = 3          # last module of a segment (output feeds bwd directly)
    FIRST_LAST = 4    # single-module segment


@dataclass
class InputOutputInfo:
    tensors: List[TensorSize] = field(default_factory=list)

    @property
    def first(self) -> TensorSize:
        return self.tensors[0]

    def total_bytes(self) -> int:
        return sum(t.mem_bytes() for t in self.tensors)

    def clone(self) -> "InputOutputInfo":
        return InputOutputInfo([t.clone() for t in self.tensors])

    def __repr__(self):
        return f"IO({self.tensors})"


def _add(a, b):
    return a + b


@dataclass
class ModuleComputeInfo:
    """FLOPs and bytes-accessed per pass (per microbatch, per rank)."""

    fwd_flops: float = 0.0
    bwd_grad_act_flops: float = 0.0
    bwd_grad_w_flops: float = 0.0
    recompute_flops: float = 0.0
    fwd_accessed_mem: float = 0.0
    bwd_grad_act_accessed_mem: float = 0.0
    bwd_grad_w_accessed_mem: float = 0.0
    recompute_accessed_mem: float = 0.0
    # bytes moved by SEPARATE glue kernels around the op (layout copies,
    # autograd fan-in grad adds, main-grad cast+add hooks) — priced
    # additively at stream bandwidth, not folded into the roofline
    fwd_extra_mem: float = 0.0
    bwd_grad_act_extra_mem: float = 0.0
    bwd_grad_w_extra_mem: float = 0.0

    @property
    def bwd_flops(self):
        return self.bwd_grad_act_flops + self.bwd_grad_w_flops

    @property
    def all_flops(self):
        return self.fwd_flops + self.bwd_flops

    def __add__(self, other: "ModuleComputeInfo"):
        return ModuleComputeInfo(
            self.fwd_flops + other.fwd_flops,
            self.bwd_grad_act_flops + other.bwd_grad_act_flops,
            self.bwd_grad_w_flops + other.bwd_grad_w_flops,
            self.recompute_flops + other.recompute_flops,
            self.fwd_accessed_mem + other.fwd_accessed_mem,
            self.bwd_grad_act_accessed_mem + other.bwd_grad_act_accessed_mem,
            self.bwd_grad_w_accessed_mem + other.bwd_grad_w_accessed_mem,
            self.recompute_accessed_mem + other.recompute_accessed_mem,
            self.fwd_extra_mem + other.fwd_extra_mem,
            self.bwd_grad_act_extra_mem + other.bwd_grad_act_extra_mem,
            self.bwd_grad_w_extra_mem + other.bwd_grad_w_extra_mem,
        )


@dataclass
class ActivationInfo:
    """Activation accounting per microbatch.

    activation_mem_cache: bytes this module keeps alive from fwd to bwd.
    fwd/bwd_peak_mem_no_cache: transient peak while executing this module
    beyond the running cache (workspace, temporaries).
    """

    activation_mem_cache: float = 0.0
    fwd_peak_mem_no_cache: float = 0.0
    bwd_peak_mem_no_cache: float = 0.0
    # bytes freed when this module's bwd completes (defaults to cache)
    fwd_grad_mem: float = 0.0

    def __add__(self, other: "ActivationInfo"):
        return ActivationInfo(
            self.activation_mem_cache + other.activation_mem_cache,
            max(self.fwd_peak_mem_no_cache, other.fwd_peak_mem_no_cache),
            max(self.bwd_peak_mem_no_cache, other.bwd_peak_mem_no_cache),
            self.fwd_grad_mem + other.fwd_grad_mem,
        )


@dataclass
class ModuleMemoryInfo:
    """Static (per-iteration-lifetime) memory: weights, grads, optimizer state."""

    dense_weight_bytes: float = 0.0
    dense_grad_bytes: float = 0.0
    dense_state_bytes: float = 0.0
    moe_weight_bytes: float = 0.0
    moe_grad_bytes: float = 0.0
    moe_state_bytes: float = 0.0
    # analog of TE dummy-wgrad workspace shapes (kept for schema parity)
    dummy_wgrad_bytes: float = 0.0
    # persistent non-parameter caches (fp8 weight-quant copies): counted
    # in peak memory but NOT in the optimizer/DP param-count derivations
    # (dense/moe_weight_bytes / 2 = numel must stay true)
    cache_bytes: float = 0.0

    @property
    def weight_bytes(self):
        return self.dense_weight_bytes + self.moe_weight_bytes + self.cache_bytes

    @property
    def grad_bytes(self):
        return self.dense_grad_bytes + self.moe_grad_bytes

    @property
    def state_bytes(self):
        return self.dense_state_bytes + self.moe_state_bytes

    @property
    def all_bytes(self):
        return self.weight_bytes + self.grad_bytes + self.state_bytes

    def __add__(self, other: "ModuleMemoryInfo"):
        return ModuleMemoryInfo(
            self.dense_weight_bytes + other.dense_weight_bytes,
            self.dense_grad_bytes + other.dense_grad_bytes,
            self.dense_state_bytes + other.dense_state_bytes,
            self.moe_weight_bytes + other.moe_weight_bytes,
            self.moe_grad_bytes + other.moe_grad_bytes,
            self.moe_state_bytes + other.moe_state_bytes,
            max(self.dummy_wgrad_bytes, other.dummy_wgrad_bytes),
            self.cache_bytes + other.cache_bytes,
        )


@dataclass
class ModuleCostInfo:
    """Times in ms, per microbatch per rank."""

    fwd_compute_time: float = 0.0      # roofline(F compute, F bytes)
    bwd_grad_act_time: float = 0.0
    bwd_grad_w_time: float = 0.0
    recompute_compute_time: float = 0.0
    fwd_net_time: float = 0.0          # total comm in fwd
    bwd_net_time: float = 0.0
    recompute_net_time: float = 0.0
    fwd_net_exposed_time: float = 0.0  # non-overlapped part
    bwd_net_exposed_time: float = 0.0
    recompute_net_exposed_time: float = 0.0

    @property
    def bwd_compute_time(self):
        return self.bwd_grad_act_time + self.bwd_grad_w_time

    @property
    def fwd_time(self):
        return self.fwd_compute_time + self.fwd_net_exposed_time

    @property
    def bwd_time(self):
        return self.bwd_compute_time + self.bwd_net_exposed_time

    @property
    def recompute_time(self):
        return self.recompute_compute_time + self.recompute_net_exposed_time

    def __add__(self, other: "ModuleCostInfo"):
        return ModuleCostInfo(
            self.fwd_compute_time + other.fwd_compute_time,
            self.bwd_grad_act_time + other.bwd_grad_act_time,
            self.bwd_grad_w_time + other.bwd_grad_w_time,
            self.recompute_compute_time + other.recompute_compute_time,
            self.fwd_net_time + other.fwd_net_time,
            self.bwd_net_time + other.bwd_net_time,
            self.recompute_net_time + other.recompute_net_time,
            self.fwd_net_exposed_time + other.fwd_net_exposed_time,
            self.bwd_net_exposed_time + other.bwd_net_exposed_time,
            self.recompute_net_exposed_time + other.recompute_net_exposed_time,
        )


@dataclass
class PointDebugInfo:
    name: str = ""
    fwd_time: float = 0.0
    bwd_time: float = 0.0
    wgrad_time: float = 0.0
    extra: Dict = field(default_factory=dict)


@dataclass
class PathDebugContext:
    """Targets module paths to dump per-path F/B/W costs (debug_points)."""

    target_point: Optional[List[str]] = None
    points: Dict[str, PointDebugInfo] = field(default_factory=dict)
    capture_graph_only: bool = False
    graph_builder: object = None


class Result(dict):
    """Dict with attribute access, used for analysis outputs."""

    def __getattr__(self, k):
        try:
            return self[k]
        except KeyError as e:
            raise AttributeError(k) from e

    def __setattr__(self, k, v):
        self[k] = v
