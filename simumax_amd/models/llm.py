"""Model assembly (L4): LLMBlock, LLMModel per PP-stage chunk, and the
activation-lifetime replay producing PeakPoint.

Parity target: simumax/core/transformer/language_model.py:12-607
(PeakPoint, LLMBlock, LLMModel.compute_activations,
get_all_gemm_cost_info / analysis_op_info).

The replay walks the chunk's ordered leaves: a forward pass accumulating
each leaf's activation cache (recompute segments keep only their input),
then a reverse pass that re-forwards recompute segments before freeing —
the peak of that timeline plus in-flight-microbatch caches is the 288 GB
budget the strategy search sizes against.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

from ..core.module import MetaModule
from ..core.records import InputOutputInfo
from ..ops.dense import (
    Add,
    Attention,
    Embedding,
    LayerNorm,
    LinearCol,
    MLAAttention,
    MLP,
    ParallelCE,
)


@dataclass
class PeakPoint:
    """Activation-memory timeline summary for one microbatch through one
    chunk (reference parity: language_model.py:12-97)."""

    cache_mem: float = 0.0        # bytes held from fwd end until bwd starts
    fwd_peak_mem: float = 0.0
    bwd_peak_mem: float = 0.0
    fwd_peak_point: str = ""
    bwd_peak_point: str = ""

    @property
    def peak_mem(self):
        return max(self.fwd_peak_mem, self.bwd_peak_mem)

    @property
    def peak_point(self):
        return (
            self.fwd_peak_point
            if self.fwd_peak_mem >= self.bwd_peak_mem
            else self.bwd_peak_point
        )

    def to_dict(self):
        return dict(
            cache_mem=self.cache_mem,
            fwd_peak_mem=self.fwd_peak_mem,
            bwd_peak_mem=self.bwd_peak_mem,
            peak_mem=self.peak_mem,
            fwd_peak_point=self.fwd_peak_point,
            bwd_peak_point=self.bwd_peak_point,
            peak_point=self.peak_point,
        )


class LLMBlock(MetaModule):
    """input_norm -> attention(+residual) -> pre_mlp_norm -> mlp(+residual).
    Reference: language_model.py:98-208."""

    def __init__(self, model_cfg, strategy, system, layer_idx=0, use_moe=False,
                 name=None):
        super().__init__(strategy, system, name or f"layer{layer_idx}")
        self.layer_idx = layer_idx
        m = model_cfg
        self.input_norm = LayerNorm(m.hidden_size, strategy, system, "input_norm")
        if m.attention_type == "mla":
            self.attention = MLAAttention(m, strategy, system)
        else:
            self.attention = Attention(m, strategy, system)
        self.attn_residual = Add(strategy, system, "attn_residual")
        self.pre_mlp_norm = LayerNorm(m.hidden_size, strategy, system, "pre_mlp_norm")
        if use_moe:
            from ..ops.moe import ExpertMLP

            self.mlp = ExpertMLP(m, strategy, system)
        else:
            self.mlp = MLP(m.hidden_size, m.intermediate_size, strategy, system,
                           use_swiglu=m.use_swiglu)
        self.mlp_residual = Add(strategy, system, "mlp_residual")

    def forward(self, input_info):
        dbg = self.path_debug_context
        x = self.input_norm(input_info, dbg)
        a = self.attention(x, dbg)
        r = self.attn_residual(
            InputOutputInfo([a.tensors[0], input_info.tensors[0]]), dbg
        )
        y = self.pre_mlp_norm(r, dbg)
        h = self.mlp(y, dbg)
        return self.mlp_residual(InputOutputInfo([h.tensors[0], r.tensors[0]]), dbg)

    def apply_recompute(self, full_block=False):
        if full_block:
            self.set_recompute(True)
            return
        if self.strategy.recompute_granularity == "full_block":
            # layers beyond recompute_layer_num are NOT recomputed at all
            # (full_block is a per-layer, not per-submodule, choice)
            return
        cfg_attn = self.strategy.parse_attention_recompute()
        cfg_mlp = self.strategy.parse_mlp_recompute()
        if cfg_attn.recompute_norm:
            self.input_norm.set_recompute()
        self.attention.apply_recompute()
        if cfg_mlp.recompute_norm:
            self.pre_mlp_norm.set_recompute()
        self.mlp.apply_recompute()


class LLMModel(MetaModule):
    """One PP-stage chunk: optional Embedding, N blocks (dense layers first
    for MoE models), optional final norm + vocab LinearCol + ParallelCE.
    Reference: language_model.py:210-467."""

    def __init__(self, model_cfg, strategy, system, layer_num, with_embedding,
                 with_loss, first_layer_idx=0, name="llm_model"):
        super().__init__(strategy, system, name)
        m = model_cfg
        self.model_cfg = m
        self.with_embedding = with_embedding
        self.with_loss = with_loss
        self.layer_num = layer_num
        if with_embedding:
            self.embedding = Embedding(m.vocab_size, m.hidden_size, strategy, system)
        self.blocks: List[LLMBlock] = []
        for i in range(layer_num):
            gidx = first_layer_idx + i
            use_moe = m.model_type == "moe" and gidx >= m.dense_layers
            blk = LLMBlock(m, strategy, system, layer_idx=gidx, use_moe=use_moe)
            self.blocks.append(blk)
            setattr(self, f"block{i}", blk)
        if with_loss:
            self.final_norm = LayerNorm(m.hidden_size, strategy, system, "final_norm")
            tp = strategy.tp_size
            assert m.vocab_size % tp == 0
            self.lm_head = LinearCol(m.hidden_size, m.vocab_size // tp, strategy,
                                     system, "lm_head")
            self.ce = ParallelCE(strategy, system)

    def forward(self, input_info):
        dbg = self.path_debug_context
        x = input_info
        if self.with_embedding:
            x = self.embedding(x, dbg)
        for blk in self.blocks:
            x = blk(x, dbg)
        if self.with_loss:
            x = self.final_norm(x, dbg)
            logits = self.lm_head(x, dbg)
            x = self.ce(logits, dbg)
        return x

    # ---- recompute orchestration ----------------------------------------
    def apply_recompute(self):
        s = self.strategy
        if not s.enable_recompute:
            return
        full = s.recompute_granularity == "full_block"
        n_rc = s.recompute_layer_num if (full and s.recompute_layer_num) else (
            self.layer_num if full else 0
        )
        for i, blk in enumerate(self.blocks):
            blk.apply_recompute(full_block=full and i < n_rc)

    # ---- activation-lifetime replay --------------------------------------
    def compute_activations(self) -> PeakPoint:
        """Replay fwd then bwd over ordered leaves; recompute segments hold
        only their input tensor between fwd and bwd and re-materialize
        during bwd. Asserts the cache returns to zero (reference parity:
        language_model.py:463-465)."""
        leaves = self.leaf_modules()
        segments = self._segments(leaves)
        pp = PeakPoint()
        cache = 0.0

        # -------- forward
        for seg in segments:
            if seg["recompute"]:
                seg_input = seg["leaves"][0].input_info.total_bytes()
                seg["held"] = seg_input
                # the checkpointed forward runs in NO-GRAD mode
                # (torch.utils.checkpoint, use_reentrant=False):
                # intermediates are freed as soon as consumed, so the
                # transient is each leaf's LIVE working set (input +
                # output + workspace), not the segment's summed caches
                for leaf in seg["leaves"]:
                    ai = leaf.get_act_info()
                    out_b = (leaf.output_info.total_bytes()
                             if leaf.output_info is not None else 0.0)
                    live = (leaf.input_info.total_bytes() + out_b
                            + ai.fwd_peak_mem_no_cache)
                    peak_here = cache + seg_input + live
                    if peak_here > pp.fwd_peak_mem:
                        pp.fwd_peak_mem = peak_here
                        pp.fwd_peak_point = leaf.full_name
                cache += seg_input
            else:
                for leaf in seg["leaves"]:
                    ai = leaf.get_act_info()
                    peak_here = cache + ai.activation_mem_cache + ai.fwd_peak_mem_no_cache
                    if peak_here > pp.fwd_peak_mem:
                        pp.fwd_peak_mem = peak_here
                        pp.fwd_peak_point = leaf.full_name
                    cache += ai.activation_mem_cache
                seg["held"] = sum(
                    l.get_act_info().activation_mem_cache for l in seg["leaves"]
                )
        pp.cache_mem = cache

        # -------- backward (reverse segment order)
        for seg in reversed(segments):
            if seg["recompute"]:
                # re-forward: segment caches rematerialize on top of cache
                remat = 0.0
                for leaf in seg["leaves"]:
                    ai = leaf.get_act_info()
                    remat += ai.activation_mem_cache
                    peak_here = cache + remat + ai.fwd_peak_mem_no_cache
                    if peak_here > pp.bwd_peak_mem:
                        pp.bwd_peak_mem = peak_here
                        pp.bwd_peak_point = leaf.full_name + "(recompute)"
                # now run segment bwd, freeing as we go
                live = remat
                for leaf in reversed(seg["leaves"]):
                    ai = leaf.get_act_info()
                    peak_here = cache + live + ai.bwd_peak_mem_no_cache
                    if peak_here > pp.bwd_peak_mem:
                        pp.bwd_peak_mem = peak_here
                        pp.bwd_peak_point = leaf.full_name + "(bwd)"
                    live -= ai.activation_mem_cache
                cache -= seg["held"]
            else:
                live = seg["held"]
                for leaf in reversed(seg["leaves"]):
                    ai = leaf.get_act_info()
                    peak_here = cache - seg["held"] + live + ai.bwd_peak_mem_no_cache
                    if peak_here > pp.bwd_peak_mem:
                        pp.bwd_peak_mem = peak_here
                        pp.bwd_peak_point = leaf.full_name + "(bwd)"
                    live -= ai.activation_mem_cache
                cache -= seg["held"]
        assert cache > -1.0, f"activation cache went negative: {cache}"
        assert abs(cache) < 1.0, f"activation cache did not return to zero: {cache}"
        self.peak_point = pp
        return pp

    @staticmethod
    def _segments(leaves):
        """Group consecutive leaves by recompute status, splitting at
        layer boundaries: each checkpointed LAYER is its own segment
        (torch.utils.checkpoint wraps one block at a time, so only one
        block's activations rematerialize during its backward — merging
        consecutive recomputed layers would claim every block's caches
        live at once)."""
        import re

        segs = []
        cur: Optional[dict] = None
        cur_key = None
        for leaf in leaves:
            rc = bool(leaf.enable_recompute)
            m = re.search(r"\.layer(\d+)\.", leaf.full_name)
            key = (rc, m.group(1) if (rc and m) else None)
            if cur is None or key != cur_key:
                cur = {"recompute": rc, "leaves": [], "held": 0.0}
                segs.append(cur)
                cur_key = key
            cur["leaves"].append(leaf)
        return segs

    # ---- calibration-shape enumeration -----------------------------------
    def analysis_op_info(self):
        """Dump every (op key, shape_desc, flops) this chunk prices — the
        HIP calibration harness sweeps exactly these keys (reference parity:
        language_model.py:469-595 / the self-referential trick of
        test_gemm_efficiency.py:258-347)."""
        out = {}
        for leaf in self.leaf_modules():
            ci = leaf.get_compute_info()
            for stage, opname, flops in (
                ("fwd", leaf.fwd_op, ci.fwd_flops),
                ("bwd_grad_act", leaf.bwd_act_op, ci.bwd_grad_act_flops),
                ("bwd_grad_w", leaf.bwd_w_op, ci.bwd_grad_w_flops),
            ):
                desc = leaf.get_input_shapes_desc(stage)
                if not desc or flops == 0:
                    continue
                out.setdefault(opname, {})[desc] = {
                    "flops": flops,
                    "stage": stage,
                    "module": leaf.full_name,
                }
        return out

    def get_all_gemm_cost_info(self):
        from ..core.module import LinearBase

        rows = []
        for leaf in self.leaf_modules():
            if not isinstance(leaf, LinearBase):
                continue
            ci = leaf.get_cost_info()
            rows.append(
                dict(
                    module=leaf.full_name,
                    fwd_shape=leaf.get_input_shapes_desc("fwd"),
                    fwd_time=ci.fwd_compute_time,
                    bwd_act_time=ci.bwd_grad_act_time,
                    bwd_w_time=ci.bwd_grad_w_time,
                )
            )
        return rows
