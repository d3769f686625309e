"""MetaModule: the nn.Module analog whose __call__ propagates symbolic
shapes and computes model/act/compute/cost records in one pass.

Parity target: simumax/core/base_struct.py:233-1204 (MetaModule, LinearBase,
GroupLinearBase) — re-designed: children are registered in *call order*
during forward (no vars() scan), and leaf communication is declared as
CommEvent records that both the analytic coster and the discrete-event
simulator consume (one source of truth for the RCCL call-sites).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional

from .config import StrategyConfig, SystemConfig
from .consts import DTYPE_BYTES
from .records import (
    ActivationInfo,
    InputOutputInfo,
    ModuleComputeInfo,
    ModuleCostInfo,
    ModuleMemoryInfo,
    PathDebugContext,
)
from .tensor import TensorSize

_STAGES = ("fwd", "bwd_act", "bwd_w", "recompute")


@dataclass
class CommEvent:
    """One collective issued by a leaf op (priced by RCCL alpha-beta model)."""

    stage: str                 # fwd | bwd_act | bwd_w | recompute
    op_name: str               # all_reduce | all_gather | reduce_scatter | all2all | p2p
    size: int                  # payload bytes (per rank input size)
    comm_num: int              # participants
    comm_stage: str            # tp | dp | dp_cp | ep | etp | edp | cp | pp
    overlap: bool = False      # overlapped with compute (not exposed)
    time_ms: float = 0.0       # filled by the coster


class _CallStack:
    stack: List["MetaModule"] = []


class MetaModule:
    """Symbolic module. Subclasses either:
    * composites — implement forward(input_info) calling child modules, or
    * leaves — implement create_output_info + the _leaf_* hooks.
    """

    def __init__(self, strategy: StrategyConfig, system: SystemConfig, name: str = ""):
        self.strategy = strategy
        self.system = system
        self.name = name or type(self).__name__
        self.parent: Optional[MetaModule] = None
        self.children_ordered_module: List[MetaModule] = []
        self.enable_recompute = False
        self.is_variance_node = False

        self.input_info: Optional[InputOutputInfo] = None
        self.output_info: Optional[InputOutputInfo] = None
        self._info_ready = False
        self._model_info = ModuleMemoryInfo()
        self._act_info = ActivationInfo()
        self._compute_info = ModuleComputeInfo()
        self._cost_info = ModuleCostInfo()
        self.comm_ops: List[CommEvent] = []
        self.path_debug_context: Optional[PathDebugContext] = None

    # ---- identity ------------------------------------------------------
    @property
    def full_name(self) -> str:
        parts = []
        m = self
        while m is not None:
            parts.append(m.name)
            m = m.parent
        return ".".join(reversed(parts))

    def is_leaf(self) -> bool:
        return len(self.children_ordered_module) == 0

    def leaf_modules(self) -> List["MetaModule"]:
        """Ordered leaves in call order (the unit of the activation replay
        and of the event simulator's job queues)."""
        if self.is_leaf():
            return [self]
        out = []
        for c in self.children_ordered_module:
            out.extend(c.leaf_modules())
        return out

    # ---- bytes helpers ---------------------------------------------------
    @property
    def element_size(self) -> int:
        return DTYPE_BYTES[self.strategy.dtype]

    @property
    def grad_element_size(self) -> int:
        return 2 if self.strategy.grad_reduce_in_bf16 else 4

    # ---- leaf hooks (defaults: zero-cost shape pass-through) -------------
    def create_output_info(self, input_info: InputOutputInfo) -> InputOutputInfo:
        return input_info.clone()

    def _leaf_model_info(self, info: ModuleMemoryInfo):
        pass

    def _leaf_act_info(self, info: ActivationInfo):
        pass

    def _leaf_compute_info(self, info: ModuleComputeInfo):
        pass

    def _leaf_intra_net_info(self):
        """Append CommEvent records to self.comm_ops."""

    def get_input_shapes_desc(self, stage: str) -> str:
        return ""

    # op-table keys per stage (overridden by GEMM/SDP leaves)
    fwd_op = "default"
    bwd_act_op = "default"
    bwd_w_op = "default"

    # ---- comm helper -----------------------------------------------------
    def add_comm(self, stage, op_name, size, comm_num, comm_stage, overlap=False):
        if comm_num <= 1 or size <= 0:
            return
        self.comm_ops.append(
            CommEvent(stage=stage, op_name=op_name, size=int(size),
                      comm_num=int(comm_num), comm_stage=comm_stage, overlap=overlap)
        )

    def resolve_net(self, comm_stage: str) -> str:
        attr = {
            "tp": "tp_net", "cp": "cp_net", "pp": "pp_net", "dp": "dp_net",
            "dp_cp": "dp_net", "ep": "ep_net", "etp": "etp_net", "edp": "edp_net",
        }[comm_stage]
        net = getattr(self.strategy, attr)
        assert net and net != "auto", (
            f"network tier for {comm_stage} unresolved — call analysis_net first"
        )
        return net

    # ---- cost computation -----------------------------------------------
    def _price_comm(self):
        """Price every CommEvent and fold into cost_info net times."""
        totals = {s: 0.0 for s in _STAGES}
        exposed = {s: 0.0 for s in _STAGES}
        for ev in self.comm_ops:
            net = self.resolve_net(ev.comm_stage)
            ev.time_ms = self.system.compute_net_op_time(
                ev.op_name, ev.size, ev.comm_num, net=net,
                comm_stage=ev.comm_stage, strategy=self.strategy,
            )
            totals[ev.stage] += ev.time_ms
            if not ev.overlap:
                exposed[ev.stage] += ev.time_ms
        ci = self._cost_info
        ci.fwd_net_time = totals["fwd"]
        ci.bwd_net_time = totals["bwd_act"] + totals["bwd_w"]
        ci.fwd_net_exposed_time = exposed["fwd"]
        ci.bwd_net_exposed_time = exposed["bwd_act"] + exposed["bwd_w"]
        if self.enable_recompute:
            # recomputation re-runs fwd comm too
            ci.recompute_net_time = ci.fwd_net_time + totals["recompute"]
            ci.recompute_net_exposed_time = ci.fwd_net_exposed_time + exposed["recompute"]
        else:
            ci.recompute_net_time = totals["recompute"]
            ci.recompute_net_exposed_time = exposed["recompute"]

    def _comp_leaf_cost_info(self):
        sysc = self.system
        comp = self._compute_info

        def stage_time(op_name, stage, flops, mem, extra=0.0, mem_op=None,
                       extra_op=None):
            c = sysc.compute_op_accuracy_time(
                op_name, flops, shape_desc=self.get_input_shapes_desc(stage),
                reture_detail=True,
            )
            io = sysc.compute_mem_access_time(mem_op or op_name, mem,
                                              reture_detail=True)
            t = sysc.compute_end2end_time(c["compute_only_time"], io["io_time"])
            if extra > 0:
                # glue kernels (layout copies, grad fan-in adds, host-bound
                # small-launch chains) run as separate launches, additive
                t += sysc.compute_mem_access_time(
                    extra_op or "default", extra,
                    units=getattr(self, "extra_op_units", 0))
            return t

        ci = self._cost_info
        ci.fwd_compute_time = stage_time(self.fwd_op, "fwd", comp.fwd_flops,
                                         comp.fwd_accessed_mem,
                                         comp.fwd_extra_mem,
                                         getattr(self, "fwd_mem_op", None),
                                         getattr(self, "fwd_extra_op", None))
        ci.bwd_grad_act_time = stage_time(self.bwd_act_op, "bwd_grad_act",
                                          comp.bwd_grad_act_flops,
                                          comp.bwd_grad_act_accessed_mem,
                                          comp.bwd_grad_act_extra_mem,
                                          getattr(self, "bwd_act_mem_op", None),
                                          getattr(self, "bwd_act_extra_op", None))
        ci.bwd_grad_w_time = stage_time(self.bwd_w_op, "bwd_grad_w",
                                        comp.bwd_grad_w_flops,
                                        comp.bwd_grad_w_accessed_mem,
                                        comp.bwd_grad_w_extra_mem,
                                        getattr(self, "bwd_w_mem_op", None),
                                        getattr(self, "bwd_w_extra_op", None))
        self._price_comm()
        if self.enable_recompute and not self.is_variance_node:
            ci.recompute_compute_time = ci.fwd_compute_time
            comp.recompute_flops = comp.fwd_flops
            comp.recompute_accessed_mem = comp.fwd_accessed_mem
        else:
            ci.recompute_compute_time = 0.0
            ci.recompute_net_time = 0.0
            ci.recompute_net_exposed_time = 0.0

    # ---- orchestration ---------------------------------------------------
    def forward(self, input_info: InputOutputInfo) -> InputOutputInfo:
        """Composites override; leaves use create_output_info."""
        assert self.is_leaf() or type(self).forward is not MetaModule.forward
        return self.create_output_info(input_info)

    def __call__(self, input_info: InputOutputInfo,
                 path_debug_context: PathDebugContext = None) -> InputOutputInfo:
        assert not self._info_ready, f"{self.full_name} called twice"
        self.path_debug_context = path_debug_context
        if _CallStack.stack:
            parent = _CallStack.stack[-1]
            self.parent = parent
            parent.children_ordered_module.append(self)
        self.input_info = input_info

        _CallStack.stack.append(self)
        try:
            out = self.forward(input_info)
        finally:
            _CallStack.stack.pop()
        assert isinstance(out, InputOutputInfo), f"{self.full_name} returned {type(out)}"
        self.output_info = out

        if self.is_leaf():
            self._leaf_model_info(self._model_info)
            self._leaf_act_info(self._act_info)
            self._leaf_compute_info(self._compute_info)
            self._leaf_intra_net_info()
            self._comp_leaf_cost_info()
            if path_debug_context is not None and path_debug_context.graph_builder is not None:
                path_debug_context.graph_builder.add_node(self)
        else:
            for c in self.children_ordered_module:
                self._model_info = self._model_info + c._model_info
                self._act_info = self._act_info + c._act_info
                self._compute_info = self._compute_info + c._compute_info
                self._cost_info = self._cost_info + c._cost_info
        self._info_ready = True

        if (
            path_debug_context is not None
            and path_debug_context.target_point
            and self.full_name in path_debug_context.target_point
        ):
            from .records import PointDebugInfo

            path_debug_context.points[self.full_name] = PointDebugInfo(
                name=self.full_name,
                fwd_time=self._cost_info.fwd_compute_time,
                bwd_time=self._cost_info.bwd_grad_act_time,
                wgrad_time=self._cost_info.bwd_grad_w_time,
            )
            self._dump_debug_point()
        return out

    def _dump_debug_point(self):
        """Append this path's F/B/W costs to TMP_PATH/cost_log.json
        (reference parity: base_struct.py:771-800 debug_points)."""
        import json
        import os

        from .consts import TMP_PATH

        os.makedirs(TMP_PATH, exist_ok=True)
        path_file = os.path.join(TMP_PATH, "cost_log.json")
        data = {}
        if os.path.exists(path_file):
            try:
                with open(path_file, encoding="utf-8") as f:
                    data = json.load(f)
            except json.JSONDecodeError:
                data = {}
        ci = self._cost_info
        data[self.full_name] = {
            "cost_F": ci.fwd_compute_time,
            "cost_B": ci.bwd_grad_act_time,
            "cost_W": ci.bwd_grad_w_time,
            "recompute_F": ci.recompute_compute_time,
            "net_F": ci.fwd_net_time,
            "net_B": ci.bwd_net_time,
        }
        with open(path_file, "w", encoding="utf-8") as f:
            json.dump(data, f, indent=4, ensure_ascii=False)

    # ---- recompute DFS ---------------------------------------------------
    def set_recompute(self, flag: bool = True):
        """Mark this module (and its subtree) as recomputed in backward."""
        self.enable_recompute = flag
        for c in self.children_ordered_module:
            c.set_recompute(flag)

    # ---- accessors -------------------------------------------------------
    def _ready(self):
        assert self._info_ready, f"{self.full_name}: call the module first"

    def get_model_info(self) -> ModuleMemoryInfo:
        self._ready()
        return self._model_info

    def get_act_info(self) -> ActivationInfo:
        self._ready()
        return self._act_info

    def get_compute_info(self) -> ModuleComputeInfo:
        self._ready()
        return self._compute_info

    def get_cost_info(self) -> ModuleCostInfo:
        self._ready()
        return self._cost_info

    def all_comm_ops(self) -> List[CommEvent]:
        return [ev for leaf in self.leaf_modules() for ev in leaf.comm_ops]

    def __repr__(self):
        def lines(m, ind):
            pad = "  " * ind
            head = f"{pad}{m.name}({type(m).__name__})"
            if m.is_leaf() and m.output_info is not None:
                head += f" -> {m.output_info.tensors}"
            out = [head]
            for c in m.children_ordered_module:
                out.extend(lines(c, ind + 1))
            return out

        return "\n".join(lines(self, 0))


# --------------------------------------------------------------------------
# GEMM-shaped leaves: shape-key generators (the calibration lookup contract)
# --------------------------------------------------------------------------
class LinearBase(MetaModule):
    """Base for GEMM ops. The shape-key strings produced here are the exact
    lookup keys of the system-config efficiency tables; the HIP calibration
    harness enumerates shapes through these same methods (reference parity:
    base_struct.py:1136-1154)."""

    def __init__(self, input_size: int, output_size: int, strategy, system, name=""):
        super().__init__(strategy, system, name)
        self.input_size = int(input_size)
        self.output_size = int(output_size)

    @property
    def micro_input_tensor(self) -> TensorSize:
        return self.input_info.tensors[0]

    def get_gemm_bmnk(self, stage: str, format: bool = False):
        t = self.micro_input_tensor
        if t.ndim == 2:
            bs, seq_len = 1, t.shape[0]
        else:
            bs, seq_len = t.shape[0], t.shape[1]
        inp, out = self.input_size, self.output_size
        bs, seq_len, inp, out = int(bs), int(seq_len), int(inp), int(out)
        wgrad_dtype = "bf16" if self.strategy.grad_reduce_in_bf16 else "fp32"
        if stage == "fwd":
            if format:
                return [[bs, seq_len, inp], [inp, out], [bs, out]]
            return dict(B=bs, M=seq_len, K=inp, N=out, layout="TN",
                        accumulate=False, out_dtype="bf16")
        if stage == "bwd_grad_act":
            if format:
                return [[bs, seq_len, out], [out, inp], [bs, inp]]
            return dict(B=bs, M=seq_len, K=out, N=inp, layout="NN",
                        accumulate=False, out_dtype="bf16")
        if stage == "bwd_grad_w":
            if format:
                return [[1, out, bs * seq_len], [bs * seq_len, inp], [out, inp]]
            return dict(B=1, M=out, K=bs * seq_len, N=inp, layout="NT",
                        accumulate=True, out_dtype=wgrad_dtype)
        raise ValueError(stage)

    def get_input_shapes_desc(self, stage: str) -> str:
        if stage not in ("fwd", "bwd_grad_act", "bwd_grad_w"):
            return ""
        k = self.get_gemm_bmnk(stage)
        return (
            f"b={k['B']}, m={k['M']}, k={k['K']}, n={k['N']}, "
            f"layout={k['layout']}, accumulate={k['accumulate']}, "
            f"out_dtype={k['out_dtype']}"
        )


class GroupLinearBase(LinearBase):
    """Grouped GEMM over local experts (reference parity:
    base_struct.py:1188-1204 for the key format)."""

    def __init__(self, local_expert_num, input_size, output_size, strategy, system, name=""):
        super().__init__(input_size, output_size, strategy, system, name)
        self.local_expert_num = int(local_expert_num)

    def get_input_shapes_desc(self, stage: str) -> str:
        if stage not in ("fwd", "bwd_grad_act", "bwd_grad_w"):
            return ""
        tokens = self.input_info.tensors[0].size(0)
        assert tokens % self.local_expert_num == 0, (
            f"tokens {tokens} % experts {self.local_expert_num} != 0"
        )
        m = tokens // self.local_expert_num
        s = (
            f"ng={self.local_expert_num}, M={m}, N={self.output_size}, "
            f"K={self.input_size}"
        )
        s += (
            f", dtype={'fp8' if self.strategy.fp8 else 'bf16'}, out_dtype=bf16, "
            f"main_grad_dtype={'bf16' if self.strategy.grad_reduce_in_bf16 else 'fp32'}"
        )
        if stage == "fwd":
            s += ", stage=fwd, grad=False, accumulate=False, use_split_accumulator=False, single_output=True"
        elif stage == "bwd_grad_act":
            s += ", stage=bwd_grad_act, grad=True, accumulate=False, use_split_accumulator=True, single_output=False"
        else:
            s += ", stage=bwd_grad_w, grad=True, accumulate=True, use_split_accumulator=True, single_output=False"
        return s
