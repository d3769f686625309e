"""MoE operator library (L3): Router, Permutation, UnPermutation,
GroupLinearCol/Row, ExpertMLP.

Parity target: simumax/core/transformer/moe_module.py (Router:20-213,
Permutation:214-530, UnPermutation:531-834, GroupLinear*:835-1369,
ExpertMLP:1370-1565) — EP dispatch/combine are RCCL all-to-alls over the
xGMI mesh; grouped GEMMs are priced by the `group_matmul` per-shape table
measured by the CDNA4 grouped-GEMM HIP harness (key format in
core.module.GroupLinearBase).
"""

from __future__ import annotations

from ..core.module import GroupLinearBase, MetaModule
from ..core.records import InputOutputInfo
from ..core.tensor import TensorSize
from .dense import Add, FP32, LinearBase, MLP, ParamMixin, Swiglu


class Router(LinearBase, ParamMixin):
    """Gating GEMM [tokens,H]x[H,E] + topk/softmax bookkeeping.
    Reference: moe_module.py:20-213."""

    fwd_op = "matmul"
    bwd_act_op = "matmul"
    bwd_w_op = "matmul"

    def __init__(self, model_cfg, strategy, system, name="router"):
        super().__init__(model_cfg.hidden_size, model_cfg.expert_num, strategy,
                         system, name)
        self.topk = model_cfg.topk
        self.m = model_cfg

    def create_output_info(self, input_info):
        t = input_info.tensors[0]
        b, s = t.shape[0], t.shape[1]
        # probs [B,S,topk] fp32 + logits
        return InputOutputInfo([TensorSize([b, s, self.topk], "fp32")])

    def _leaf_model_info(self, info):
        self.add_param(info, self.input_size * self.output_size)

    def _leaf_act_info(self, info):
        t = self.input_info.first
        b, s = t.shape[0], t.shape[1]
        logits = b * s * self.output_size * FP32
        info.activation_mem_cache = t.mem_bytes() + logits + b * s * self.topk * (FP32 + 4)

    fwd_extra_op = "moe_routing"
    bwd_act_extra_op = "moe_routing_bwd"

    @property
    def extra_op_units(self):
        # the routing chain's launch count scales with LOCAL experts (the
        # per-expert grouped-GEMM loop + per-expert index bookkeeping);
        # measured 2026-09 on MI355X: idle/layer-mb 0.285 ms at E=8 vs
        # 3.54 ms at E=162 (scripts/moe_idle_probe.py)
        return self.m.expert_num // self.strategy.ep_size

    def _leaf_compute_info(self, info):
        k = self.get_gemm_bmnk("fwd")
        flops = 2 * k["B"] * k["M"] * k["K"] * k["N"]
        info.fwd_flops = flops
        info.bwd_grad_act_flops = flops
        info.bwd_grad_w_flops = flops
        t = self.input_info.first
        logits = t.numel() // t.shape[-1] * self.output_size * FP32
        info.fwd_accessed_mem = t.mem_bytes() + 3 * logits
        info.bwd_grad_act_accessed_mem = t.mem_bytes() + 3 * logits
        info.bwd_grad_w_accessed_mem = t.mem_bytes() + logits
        # the routing chain (softmax/topk/argsort/bincount/cumsum + index
        # bookkeeping) is a host-launch-bound run of ~50 tiny kernels; its
        # calibrated per-layer latency lives in bandwidth["moe_routing"]
        # (measured by scripts/insitu_calib.py). Priced once fwd, once bwd.
        info.fwd_extra_mem = logits
        info.bwd_grad_act_extra_mem = logits


class Permutation(MetaModule):
    """Dispatch: permute1 (bandwidth keys permute_fwd/permute_bwd) ->
    EP all2all -> optional ETP all_gather -> permute2 (capacity padding).
    Reference: moe_module.py:214-530."""

    def __init__(self, model_cfg, strategy, system, name="permutation"):
        super().__init__(strategy, system, name)
        self.m = model_cfg
        self.topk = model_cfg.topk
        self.ep = strategy.ep_size
        self.etp = strategy.etp_size

    def _per_expert_capacity(self, tokens):
        """Tokens one source rank contributes per expert (capacity-padded)."""
        import math

        cap = self.m.capacity if self.m.moe_pad_expert_input_to_capacity else 1
        return int(math.ceil(tokens * self.topk / self.m.expert_num * cap))

    def _expanded_tokens(self, tokens):
        return self._per_expert_capacity(tokens) * self.m.expert_num

    def create_output_info(self, input_info):
        t = input_info.tensors[0]
        b, s, h = t.shape[0], t.shape[1], t.shape[-1]
        # after dispatch: E/ep local experts, each holding cap tokens from
        # every one of the ep source ranks, gathered over etp
        cap = self._per_expert_capacity(b * s)
        local_experts = self.m.expert_num // self.ep
        n = local_experts * cap * self.ep * self.etp
        return InputOutputInfo([TensorSize([n, h], t.dtype)])

    def _leaf_act_info(self, info):
        t = self.input_info.first
        tokens = t.shape[0] * t.shape[1]
        # index maps (int32) for unpermute in bwd
        info.activation_mem_cache = self._expanded_tokens(tokens) * 4 * 2
        info.fwd_peak_mem_no_cache = self.output_info.first.mem_bytes()

    def _leaf_compute_info(self, info):
        in_b = self.input_info.first.mem_bytes()
        out_b = self.output_info.first.mem_bytes()
        # trainer dispatch sequence (train/moe.py): zero-fill the padded
        # buffer, index_select the kept tokens (r+w), index_copy into the
        # capacity slots (r+w) -> ~5 passes over the expanded buffer
        info.fwd_accessed_mem = 2 * out_b + 3 * max(in_b * self.topk, out_b)
        # bwd: index_select the slot grads (r+w) + index_add into dX (r+r+w)
        info.bwd_grad_act_accessed_mem = 2 * out_b + 3 * in_b

    def _comp_leaf_cost_info(self):
        sysc = self.system
        comp = self._compute_info
        ci = self._cost_info
        ci.fwd_compute_time = sysc.compute_mem_access_time("permute_fwd",
                                                           comp.fwd_accessed_mem)
        ci.bwd_grad_act_time = sysc.compute_mem_access_time("permute_bwd",
                                                            comp.bwd_grad_act_accessed_mem)
        ci.bwd_grad_w_time = 0.0
        self._price_comm()
        ci.recompute_compute_time = ci.fwd_compute_time if self.enable_recompute else 0.0

    def _leaf_intra_net_info(self):
        t = self.input_info.first
        tokens = t.shape[0] * t.shape[1]
        h = t.shape[-1]
        payload = self._expanded_tokens(tokens) * h * self.element_size
        if self.ep > 1:
            self.add_comm("fwd", "all2all", payload, self.ep, "ep")
            self.add_comm("bwd_act", "all2all", payload, self.ep, "ep")
            if self.strategy.dispatch_probs:
                probs = self._expanded_tokens(tokens) * FP32
                self.add_comm("fwd", "all2all", probs, self.ep, "ep")
        if self.etp > 1:
            self.add_comm("fwd", "all_gather", payload * self.etp, self.etp, "etp")
            self.add_comm("bwd_act", "reduce_scatter", payload * self.etp, self.etp, "etp")


class UnPermutation(MetaModule):
    """Combine: optional ETP reduce_scatter -> EP all2all (reverse) ->
    unpermute + probs-weighted sum. Reference: moe_module.py:531-834."""

    def __init__(self, model_cfg, strategy, system, name="unpermutation"):
        super().__init__(strategy, system, name)
        self.m = model_cfg
        self.topk = model_cfg.topk
        self.ep = strategy.ep_size
        self.etp = strategy.etp_size

    def create_output_info(self, input_info):
        # combine collapses the expanded/padded tokens back to [n_local, H];
        # the ExpertMLP composite restores the [B, S, H] view
        t = input_info.tensors[0]
        n = t.shape[0] // (self.topk * self.etp)
        return InputOutputInfo([TensorSize([n, t.shape[-1]], t.dtype)])

    def _leaf_act_info(self, info):
        t = self.input_info.first
        n_exp = t.shape[0] // self.etp
        # probs (fp32) + expert outputs kept for dprobs unless fused
        info.activation_mem_cache = n_exp * FP32
        if not self.strategy.dispatch_probs:
            info.activation_mem_cache += t.mem_bytes() // self.etp

    def _leaf_compute_info(self, info):
        in_b = self.input_info.first.mem_bytes() // self.etp
        out_b = self.output_info.first.mem_bytes()
        # trainer combine (train/moe.py): index_select expert outputs (r+w),
        # probs-weighted multiply (r+w), zero-fill + index_add into the
        # token buffer (r+r+w)
        info.fwd_accessed_mem = 4 * in_b + 3 * out_b
        # bwd: index_select dout, multiply by probs, dprobs reduction
        # (reads y and dout), index_add back to slot grads
        info.bwd_grad_act_accessed_mem = 6 * in_b + 2 * out_b
        n = self.input_info.first.numel() // self.etp
        info.fwd_flops = 2 * n
        info.bwd_grad_act_flops = 2 * n

    def _comp_leaf_cost_info(self):
        sysc = self.system
        comp = self._compute_info
        ci = self._cost_info
        ci.fwd_compute_time = sysc.compute_mem_access_time("permute_fwd",
                                                           comp.fwd_accessed_mem)
        ci.bwd_grad_act_time = sysc.compute_mem_access_time("permute_bwd",
                                                            comp.bwd_grad_act_accessed_mem)
        ci.bwd_grad_w_time = 0.0
        self._price_comm()
        ci.recompute_compute_time = ci.fwd_compute_time if self.enable_recompute else 0.0

    def _leaf_intra_net_info(self):
        payload = self.input_info.first.mem_bytes() // self.etp
        if self.etp > 1:
            self.add_comm("fwd", "reduce_scatter", payload * self.etp, self.etp, "etp")
            self.add_comm("bwd_act", "all_gather", payload * self.etp, self.etp, "etp")
        if self.ep > 1:
            self.add_comm("fwd", "all2all", payload, self.ep, "ep")
            self.add_comm("bwd_act", "all2all", payload, self.ep, "ep")


class GroupLinearCol(GroupLinearBase, ParamMixin):
    """Grouped GEMM over local experts, column-parallel across etp.
    Op key group_matmul / fp8_group_matmul. Reference: moe_module.py:835-1369."""

    def __init__(self, local_expert_num, input_size, output_size, strategy,
                 system, name="group_linear_col"):
        super().__init__(local_expert_num, input_size, output_size, strategy,
                         system, name)
        self.fwd_op = self.bwd_act_op = self.bwd_w_op = (
            "fp8_group_matmul" if strategy.fp8 else "group_matmul"
        )

    def create_output_info(self, input_info):
        t = input_info.tensors[0]
        return InputOutputInfo([TensorSize([t.shape[0], self.output_size], t.dtype)])

    def _leaf_model_info(self, info):
        self.add_param(info,
                       self.local_expert_num * self.input_size * self.output_size,
                       is_expert=True)

    def _leaf_act_info(self, info):
        cache = self.input_info.first.mem_bytes()
        if self.strategy.fp8 and self.strategy.cache_groupgemm_col_fp8_inputs:
            cache //= self.element_size  # keep fp8 copy instead of bf16
        if self.strategy.offload_groupgemm_col_inputs:
            cache = 0
        info.activation_mem_cache = cache

    def _leaf_compute_info(self, info):
        tokens = self.input_info.first.shape[0]
        flops = 2 * tokens * self.input_size * self.output_size
        info.fwd_flops = flops
        info.bwd_grad_act_flops = flops
        info.bwd_grad_w_flops = flops
        e = self.element_size
        in_b = tokens * self.input_size * e
        w_b = self.local_expert_num * self.input_size * self.output_size * e
        out_b = tokens * self.output_size * e
        info.fwd_accessed_mem = in_b + w_b + out_b
        info.bwd_grad_act_accessed_mem = out_b + w_b + in_b
        info.bwd_grad_w_accessed_mem = out_b + in_b + w_b * self.grad_element_size // e


class GroupLinearRow(GroupLinearCol):
    """Row-parallel grouped GEMM (weights sharded over etp on K)."""


class ExpertMLP(MetaModule):
    """Full MoE layer: shared-expert MLP in parallel with
    router -> permutation -> gl1 -> swiglu -> gl2 -> unpermutation (+ add).
    Reference: moe_module.py:1370-1565."""

    def __init__(self, model_cfg, strategy, system, name="expert_mlp"):
        super().__init__(strategy, system, name)
        m = model_cfg
        self.m = m
        ep, etp = strategy.ep_size, strategy.etp_size
        assert m.expert_num % ep == 0, f"experts {m.expert_num} % ep {ep}"
        local_experts = m.expert_num // ep
        ffn_local = m.moe_ffn_hidden_size
        assert ffn_local % etp == 0
        ffn_local //= etp
        self.router = Router(m, strategy, system)
        self.permutation = Permutation(m, strategy, system)
        fc1_out = 2 * ffn_local if m.use_swiglu else ffn_local
        self.gl1 = GroupLinearCol(local_experts, m.hidden_size, fc1_out,
                                  strategy, system, "gl1")
        self.act = Swiglu(strategy, system, weighted=strategy.dispatch_probs)
        self.gl2 = GroupLinearRow(local_experts, ffn_local, m.hidden_size,
                                  strategy, system, "gl2")
        self.unpermutation = UnPermutation(m, strategy, system)
        if m.moe_shared_expert_intermediate_size:
            self.shared_mlp = MLP(m.hidden_size,
                                  m.moe_shared_expert_intermediate_size,
                                  strategy, system, "shared_mlp",
                                  use_swiglu=m.use_swiglu)
            self.shared_add = Add(strategy, system, "shared_add")

    def forward(self, input_info):
        dbg = self.path_debug_context
        self.router(input_info, dbg)
        disp = self.permutation(input_info, dbg)
        h = self.gl1(disp, dbg)
        a = self.act(h, dbg)
        y = self.gl2(a, dbg)
        out = self.unpermutation(y, dbg)
        # reshape back to [B, S, H]
        t = input_info.tensors[0]
        out = InputOutputInfo([TensorSize(list(t.shape), t.dtype)])
        if self.m.moe_shared_expert_intermediate_size:
            s = self.shared_mlp(input_info, dbg)
            out = self.shared_add(InputOutputInfo([out.tensors[0], s.tensors[0]]), dbg)
        return out

    def apply_recompute(self):
        cfg = self.strategy.parse_mlp_recompute()
        mods = self.strategy.megatron_recompute_module_set
        if cfg.recompute_fc1:
            self.gl1.set_recompute()
        if cfg.recompute_act or "moe_act" in mods:
            self.act.set_recompute()
        if cfg.recompute_fc2:
            self.gl2.set_recompute()
        if "moe" in mods:
            for mod in (self.router, self.permutation, self.gl1, self.act,
                        self.gl2, self.unpermutation):
                mod.set_recompute()
        if hasattr(self, "shared_mlp"):
            self.shared_mlp.apply_recompute()
