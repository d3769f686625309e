"""PerfLLM: the perf-analysis orchestrator (L5).

Parity target: simumax/core/perf_llm.py:293-3696 (PerfBase.configure /
run_estimate / analysis_net, PerfLLM.build / analysis_mem / analysis_cost /
calculate_1f1b_bubble / analysis / simulate / search APIs) plus the
straggler model (perf_llm.py:255-291) and the ZeRO-1 DP / optimizer
time model (perf_llm.py:1470-1597).

MI355X-first notes:
* network tiers resolve to the xGMI FC8 intra-node tier for any group that
  fits inside the 8-GPU node; inter_node is only used beyond one node.
* the 1F1B schedule is computed by an exact dependency recurrence (not a
  closed form): at pp*mbc scale this is trivial and it doubles as the
  schedule-record source for trace export.
"""

from __future__ import annotations

import json
import math
import os
from copy import deepcopy
from dataclasses import dataclass
from typing import Dict, List, Optional

from ..core.config import ModelConfig, StrategyConfig, SystemConfig
from ..core.records import PathDebugContext, Result
from ..core.tensor import TensorSize
from ..core.records import InputOutputInfo
from ..core.utils import (
    HumanReadableSize,
    get_pp_p2p_comm_size,
    human_readable_result,
    stage_layers,
)
from ..models.llm import LLMModel

GiB = 1024**3


# --------------------------------------------------------------------------
# straggler model (reference parity: perf_llm.py:255-291)
# --------------------------------------------------------------------------
def get_effective_straggler_sample_count(strategy: StrategyConfig, num_per_node: int) -> int:
    nodes = max(1, math.ceil(strategy.world_size / num_per_node))
    return min(nodes, strategy.dp_size, max(1, strategy.edp_size))


def estimate_straggler_increase_ratio(n: int) -> float:
    """1 + log2(n)/(log2(n)+1) * 0.09 * sqrt(log2(n)) — log-damped
    machine-level slowdown (reference perf_llm.py:255-291)."""
    if n <= 1:
        return 1.0
    ns = math.log2(n)
    return 1.0 + ns / (ns + 1) * 0.09 * math.sqrt(ns)


# --------------------------------------------------------------------------
# 1F1B schedule recurrence
# --------------------------------------------------------------------------
@dataclass
class ScheduleRecord:
    stage: int
    mb: int
    kind: str   # 'F' | 'B'
    start: float
    end: float


def schedule_1f1b(pp: int, mbc: int, fwd: List[float], bwd: List[float],
                  p2p: float) -> (float, List[ScheduleRecord]):
    """Exact non-interleaved 1F1B: per-stage streams of warmup-F /
    steady (1F,1B) / cooldown-B with matched send/recv rendezvous."""
    # build per-stage op streams in Megatron order
    streams = []
    for i in range(pp):
        warm = min(pp - i - 1, mbc)
        ops = [("F", m) for m in range(warm)]
        nf, nb = warm, 0
        while nb < mbc:
            if nf < mbc:
                ops.append(("F", nf)); nf += 1
            ops.append(("B", nb)); nb += 1
        streams.append(ops)

    f_end = [[None] * mbc for _ in range(pp)]
    b_end = [[None] * mbc for _ in range(pp)]
    ptr = [0] * pp
    t = [0.0] * pp
    records: List[ScheduleRecord] = []
    remaining = sum(len(s) for s in streams)
    while remaining:
        progressed = False
        for i in range(pp):
            while ptr[i] < len(streams[i]):
                kind, m = streams[i][ptr[i]]
                if kind == "F":
                    dep = 0.0 if i == 0 else (
                        f_end[i - 1][m] + p2p if f_end[i - 1][m] is not None else None
                    )
                    dur = fwd[i]
                else:
                    if f_end[i][m] is None:
                        break
                    if i == pp - 1:
                        dep = f_end[i][m]
                    else:
                        dep = (b_end[i + 1][m] + p2p
                               if b_end[i + 1][m] is not None else None)
                    dur = bwd[i]
                if dep is None:
                    break
                start = max(t[i], dep)
                end = start + dur
                records.append(ScheduleRecord(i, m, kind, start, end))
                if kind == "F":
                    f_end[i][m] = end
                else:
                    b_end[i][m] = end
                t[i] = end
                ptr[i] += 1
                remaining -= 1
                progressed = True
        if not progressed:
            raise RuntimeError("1F1B schedule deadlock (dependency cycle)")
    return max(t), records


# --------------------------------------------------------------------------
# PerfLLM
# --------------------------------------------------------------------------
class PerfBase:
    def __init__(self):
        self.strategy: Optional[StrategyConfig] = None
        self.model_config: Optional[ModelConfig] = None
        self.system: Optional[SystemConfig] = None
        self._configured = False

    def configure(self, strategy_config: StrategyConfig, model_config: ModelConfig,
                  system_config: SystemConfig):
        self.strategy = deepcopy(strategy_config)
        self.model_config = deepcopy(model_config)
        self.system = deepcopy(system_config)
        self.strategy.sanity_check()
        self.model_config.sanity_check()
        self.system.sanity_check()
        self._cross_sanity_check()
        self._configured = True

    def _cross_sanity_check(self):
        s, m = self.strategy, self.model_config
        assert m.head_num % s.tp_size == 0, "head_num % tp != 0"
        if s.cp_size > 1 and s.cp_comm_type == "a2a":
            assert (m.head_num // s.tp_size) % s.cp_size == 0, (
                "CP a2a requires local head_num divisible by cp"
            )
        if m.attention_type == "mla":
            assert s.tp_size == 1, "MLA requires tp_size == 1"
        if m.model_type == "moe":
            assert m.expert_num % s.ep_size == 0

    # ---- net tier resolution (reference: analysis_net perf_llm.py:369-474)
    def analysis_net(self, re_analysis: bool = True):
        s = self.strategy
        n = self.system.num_per_node
        tp, cp, pp, ep, etp = s.tp_size, s.cp_size, s.pp_size, s.ep_size, s.etp_size
        dp, edp = s.dp_size, s.edp_size

        def span(group_size, stride_ranks):
            return group_size * stride_ranks

        choice = {
            # order tp-cp-dp-pp: strides in ranks
            "tp_net": span(tp, 1),
            "cp_net": span(cp, tp),
            "dp_net": span(dp, tp * cp),
            "pp_net": span(pp, tp * cp * dp),
            # expert order etp-ep-edp-pp
            "etp_net": span(etp, 1),
            "ep_net": span(ep, etp),
            "edp_net": span(edp, etp * ep),
        }
        has_low = "low_intra_node" in self.system.networks
        for attr, sp in choice.items():
            cur = getattr(s, attr)
            if cur != "auto" and not re_analysis:
                continue
            if attr == "pp_net":
                # adjacent-stage p2p: consecutive stages are co-located in a
                # node when the stage stride (world/pp ranks) is smaller than
                # the node, even if the full pp group spans nodes
                # (ref analysis_high_link_net: world_size//pp_size < num_per_node)
                stride = s.world_size // pp if pp > 1 else 1
                tier = "high_intra_node" if stride < n else (
                    "high_intra_node" if sp <= n else "inter_node")
            elif sp <= n:
                tier = "high_intra_node"
            else:
                tier = "inter_node"
            if tier == "high_intra_node" and has_low and attr in ("dp_net", "edp_net"):
                # dense-DP/edp traffic rides the lower intra-node tier when
                # the system config defines one (ref: intra_with_pcie tiers);
                # single-fabric xGMI configs simply omit low_intra_node
                tier = "low_intra_node"
            setattr(s, attr, tier)
        return {k: getattr(s, k) for k in choice}


class PerfLLM(PerfBase):
    """Analytic estimator with the reference's public API."""

    def __init__(self):
        super().__init__()
        self.chunks: List[LLMModel] = []
        self.stage_layer_counts: List[int] = []
        self.debug_points: Optional[List[str]] = None
        self.debug_ctx: Optional[PathDebugContext] = None
        self._estimated = False

    # ---- build -----------------------------------------------------------
    def build(self):
        s, m = self.strategy, self.model_config
        self.stage_layer_counts = stage_layers(s, m)
        self.chunks = []
        self.vchunks = None
        first_idx = 0
        for stage in range(s.pp_size):
            layer_num = self.stage_layer_counts[stage]
            chunk = LLMModel(
                m, s, self.system,
                layer_num=layer_num,
                with_embedding=(stage == 0),
                with_loss=(stage == s.pp_size - 1),
                first_layer_idx=first_idx,
                name=f"stage{stage}",
            )
            first_idx += layer_num
            self.chunks.append(chunk)
        vp = max(1, s.interleaving_size)
        if vp > 1:
            # virtual-chunk models: layers split over pp*vp virtual stages
            # in chunk-major order (Megatron interleaving; reference parity
            # perf_llm.py:786-835)
            assert m.layer_num % (s.pp_size * vp) == 0, (
                "interleaving requires layer_num % (pp*vp) == 0")
            per_v = m.layer_num // (s.pp_size * vp)
            self.vchunks = []
            for stage in range(s.pp_size):
                row = []
                for c in range(vp):
                    v = c * s.pp_size + stage
                    row.append(LLMModel(
                        m, s, self.system, layer_num=per_v,
                        with_embedding=(v == 0),
                        with_loss=(v == s.pp_size * vp - 1),
                        first_layer_idx=v * per_v,
                        name=f"stage{stage}.chunk{c}",
                    ))
                self.vchunks.append(row)

    def _input_info_for_stage(self, stage: int) -> InputOutputInfo:
        s, m = self.strategy, self.model_config
        b = s.micro_batch_size
        seq = s.seq_len // s.cp_size
        if stage == 0:
            return InputOutputInfo([TensorSize([b, seq], "int64")])
        if s.enable_sequence_parallel:
            seq //= s.tp_size
        return InputOutputInfo([TensorSize([b, seq, m.hidden_size], s.dtype)])

    def run_estimate(self):
        assert self._configured, "call configure() first"
        self.model_config.maybe_pad_vocab_size(self.strategy.tp_size)
        self.analysis_net(re_analysis=True)
        self.build()
        self.debug_ctx = PathDebugContext(target_point=self.debug_points)
        for stage, chunk in enumerate(self.chunks):
            chunk(self._input_info_for_stage(stage), self.debug_ctx)
            chunk.apply_recompute()
            # recompute flags were applied after the call; refresh cost-deps
            self._refresh_recompute_costs(chunk)
            chunk.compute_activations()
        if self.vchunks is not None:
            for stage, row in enumerate(self.vchunks):
                for c, chunk in enumerate(row):
                    v = c * self.strategy.pp_size + stage
                    chunk(self._input_info_for_stage(0 if v == 0 else stage + 1),
                          self.debug_ctx)
                    chunk.apply_recompute()
                    self._refresh_recompute_costs(chunk)
                    chunk.compute_activations()
        self._estimated = True

    def _refresh_recompute_costs(self, chunk: LLMModel):
        rf = getattr(self.system.accelerator, "recompute_factor", 1.0) or 1.0
        for leaf in chunk.leaf_modules():
            if leaf.enable_recompute:
                ci = leaf._cost_info
                if not leaf.is_variance_node:
                    ci.recompute_compute_time = ci.fwd_compute_time * rf
                    ci.recompute_net_time = ci.fwd_net_time
                    ci.recompute_net_exposed_time = ci.fwd_net_exposed_time
                    leaf._compute_info.recompute_flops = leaf._compute_info.fwd_flops
                    leaf._compute_info.recompute_accessed_mem = (
                        leaf._compute_info.fwd_accessed_mem
                    )
        # re-aggregate composite records bottom-up
        def agg(mod):
            if mod.is_leaf():
                return
            from ..core.records import (ActivationInfo, ModuleComputeInfo,
                                        ModuleCostInfo, ModuleMemoryInfo)
            mod._model_info = ModuleMemoryInfo()
            mod._act_info = ActivationInfo()
            mod._compute_info = ModuleComputeInfo()
            mod._cost_info = ModuleCostInfo()
            for c in mod.children_ordered_module:
                agg(c)
                mod._model_info = mod._model_info + c._model_info
                mod._act_info = mod._act_info + c._act_info
                mod._compute_info = mod._compute_info + c._compute_info
                mod._cost_info = mod._cost_info + c._cost_info
        agg(chunk)

    # ---- memory ----------------------------------------------------------
    def _inflight_microbatches(self, stage: int) -> int:
        s = self.strategy
        if s.pp_size == 1:
            return 1
        vp = max(1, s.interleaving_size)
        if vp == 1:
            return min(s.micro_batch_num, s.pp_size - stage)
        # Megatron interleaved estimate: warmup microbatches per stage
        inflight = (s.pp_size - stage - 1) * 2 + (vp - 1) * s.pp_size + 1
        inflight = math.ceil(inflight / vp)
        return min(s.micro_batch_num, max(1, inflight))

    def analysis_mem(self) -> Result:
        assert self._estimated, "call run_estimate() first"
        s = self.strategy
        out = Result()
        stages = []
        vp = max(1, s.interleaving_size)
        for stage, chunk in enumerate(self.chunks):
            model_info = chunk.get_model_info()
            pp_point = chunk.peak_point
            inflight = self._inflight_microbatches(stage)
            cache_per_mb = pp_point.cache_mem
            if vp > 1 and self.vchunks is not None:
                # stage holds its vp virtual chunks; in-flight fwd acts are
                # per-VIRTUAL-chunk caches (reference: perf_llm.py:1801-1828)
                from .vpp import interleaved_inflight_microbatches

                model_info = self.vchunks[stage][0].get_model_info()
                for c in range(1, vp):
                    model_info = model_info + self.vchunks[stage][c].get_model_info()
                caches = [self.vchunks[stage][c].peak_point.cache_mem
                          for c in range(vp)]
                inflight = interleaved_inflight_microbatches(
                    s.pp_size, vp, s.micro_batch_num, stage)
                cache_per_mb = max(caches)
                pp_point = max((self.vchunks[stage][c].peak_point
                                for c in range(vp)),
                               key=lambda x: x.peak_mem)
            # fused-wgrad placeholder grads are freshly allocated per
            # backward and STOLEN by AccumulateGrad (kernels/ops.py), so
            # they are transient (modeled per-linear in bwd_peak_mem) —
            # no persistent shared-dummy term remains.
            dummy_wgrad = 0.0
            peak = (
                model_info.all_bytes
                + dummy_wgrad
                + (inflight - 1) * cache_per_mb
                + pp_point.peak_mem
            )
            peak /= s.mem_factor
            stages.append(
                dict(
                    stage=stage,
                    dummy_wgrad_mem=dummy_wgrad,
                    weight_mem=model_info.weight_bytes,
                    grad_mem=model_info.grad_bytes,
                    state_mem=model_info.state_bytes,
                    model_mem=model_info.all_bytes,
                    act_cache_per_mb_mem=cache_per_mb,
                    inflight_microbatches=inflight,
                    fwd_peak_act_mem=pp_point.fwd_peak_mem,
                    bwd_peak_act_mem=pp_point.bwd_peak_mem,
                    peak_point=pp_point.peak_point,
                    peak_mem=peak,
                )
            )
        out["stages_raw"] = stages
        out["max_peak_mem"] = max(st["peak_mem"] for st in stages)
        out["max_peak_stage"] = max(stages, key=lambda st: st["peak_mem"])["stage"]
        out["mem_gbs_budget"] = self.system.accelerator.mem_gbs
        out["oom"] = out["max_peak_mem"] > self.system.accelerator.mem_gbs * GiB
        out["stages"] = [human_readable_result(st) for st in stages]
        out["max_peak_mem_str"] = HumanReadableSize.format_bytes(out["max_peak_mem"])
        return out

    # ---- DP / optimizer time (reference: perf_llm.py:1470-1597) ----------
    def _compute_dp_time(self, stage: int) -> float:
        s = self.strategy
        chunk = self.chunks[stage]
        mi = chunk.get_model_info()
        if s.dp_size <= 1 and s.edp_size <= 1:
            return 0.0
        total = 0.0
        bucket = max(40 * 1024**2, 1024**2 * s.dp_size) * 4

        def priced(bytes_total, group, stage_name):
            if group <= 1 or bytes_total <= 0:
                return 0.0
            net = getattr(s, "dp_net" if stage_name in ("dp", "dp_cp") else "edp_net")
            n_buckets = max(1, math.ceil(bytes_total / bucket))
            per = bytes_total / n_buckets
            t = 0.0
            for _ in range(n_buckets):
                if s.zero_state >= 1:
                    t += self.system.compute_net_op_time(
                        "reduce_scatter", per, group, net=net,
                        comm_stage=stage_name, strategy=s)
                    # gathered params are bf16 (2 B); the bucket bytes were
                    # built with grad_e B/elem, so convert with the SAME
                    # grad_e regardless of use_fp32_accum_grad
                    t += self.system.compute_net_op_time(
                        "all_gather", per / grad_e * 2,
                        group, net=net, comm_stage=stage_name, strategy=s)
                else:
                    t += self.system.compute_net_op_time(
                        "all_reduce", per, group, net=net,
                        comm_stage=stage_name, strategy=s)
            return t

        grad_e = 2 if s.grad_reduce_in_bf16 else 4
        dense_params = mi.dense_weight_bytes / 2  # numel (bf16 weights)
        moe_params = mi.moe_weight_bytes / 2
        total += priced(dense_params * grad_e, s.dp_size * s.cp_size, "dp_cp")
        total += priced(moe_params * grad_e, s.edp_size, "edp")
        return total

    def _compute_optim_time(self, stage: int) -> float:
        """Megatron mixed-precision Adam as HBM traffic: zero_grad + l2-norm
        + clip + adam update + param cast/copy over the local shard."""
        s = self.strategy
        mi = self.chunks[stage].get_model_info()
        numel = (mi.dense_weight_bytes / 2) / max(1, _zero_div_dense(s)) + (
            mi.moe_weight_bytes / 2
        ) / max(1, _zero_div_moe(s))
        from ..core.consts import OPTIMIZER_TRAFFIC_BYTES_PER_PARAM

        bytes_traffic = numel * OPTIMIZER_TRAFFIC_BYTES_PER_PARAM
        return self.system.compute_mem_access_time("optimizer", bytes_traffic)

    # ---- cost ------------------------------------------------------------
    def analysis_cost(self) -> Result:
        assert self._estimated
        s, m = self.strategy, self.model_config
        pp, mbc = s.pp_size, s.micro_batch_num

        fwd, bwd = [], []
        per_stage = []
        for chunk in self.chunks:
            ci = chunk.get_cost_info()
            f = ci.fwd_compute_time + ci.fwd_net_exposed_time
            b = (
                ci.bwd_compute_time
                + ci.bwd_net_exposed_time
                + ci.recompute_compute_time
                + ci.recompute_net_exposed_time
            )
            fwd.append(f)
            bwd.append(b)
            per_stage.append(dict(
                fwd_time=f,
                bwd_time=b,
                fwd_compute_time=ci.fwd_compute_time,
                bwd_compute_time=ci.bwd_compute_time,
                recompute_time=ci.recompute_compute_time,
                fwd_net_exposed_time=ci.fwd_net_exposed_time,
                bwd_net_exposed_time=ci.bwd_net_exposed_time,
            ))

        p2p_time = 0.0
        if pp > 1:
            p2p_size = get_pp_p2p_comm_size(s, m)
            p2p_time = self.system.compute_net_op_time(
                "p2p", p2p_size, 2, net=s.pp_net, comm_stage="pp", strategy=s)

        vp = max(1, s.interleaving_size)
        if pp == 1:
            pipeline_time = mbc * (fwd[0] + bwd[0])
            self.schedule_records = None
            bubble_time = 0.0
        elif vp == 1:
            pipeline_time, records = schedule_1f1b(pp, mbc, fwd, bwd, p2p_time)
            self.schedule_records = records
            ideal = max(mbc * (f + b) for f, b in zip(fwd, bwd))
            bubble_time = pipeline_time - ideal
        else:
            # exact interleaved sync-VPP schedule (Megatron schedule table)
            from .vpp import schedule_interleaved

            vf = [[0.0] * vp for _ in range(pp)]
            vb = [[0.0] * vp for _ in range(pp)]
            for stage in range(pp):
                for c in range(vp):
                    ci = self.vchunks[stage][c].get_cost_info()
                    vf[stage][c] = ci.fwd_compute_time + ci.fwd_net_exposed_time
                    vb[stage][c] = (ci.bwd_compute_time + ci.bwd_net_exposed_time
                                    + ci.recompute_compute_time
                                    + ci.recompute_net_exposed_time)
            pipeline_time, records = schedule_interleaved(
                pp, vp, mbc, vf, vb, p2p_time)
            self.schedule_records = records
            ideal = max(mbc * (sum(vf[s_]) + sum(vb[s_])) for s_ in range(pp))
            bubble_time = pipeline_time - ideal

        # straggler
        n = get_effective_straggler_sample_count(s, self.system.num_per_node)
        straggler = estimate_straggler_increase_ratio(n) if s.enable_straggler_model else 1.0
        pipeline_time *= straggler

        dp_time_raw = max(self._compute_dp_time(i) for i in range(pp))
        if s.overlap_grad_reduce and dp_time_raw > 0:
            # bucketed grad reduce rides the LAST microbatch's backward
            # (Megatron no_sync semantics); only the tail is exposed
            bwd_last = max(bwd) if bwd else 0.0
            dp_time = max(0.0, dp_time_raw - bwd_last)
        else:
            dp_time = dp_time_raw
        optim_time = max(self._compute_optim_time(i) for i in range(pp))
        iter_time = pipeline_time + dp_time + optim_time

        tokens = s.global_batch_size * s.seq_len
        flops_token = m.flops_per_token(s.seq_len)
        total_flops = flops_token * tokens
        peak_tflops = self.system.accelerator.op["matmul"].tflops
        mfu = total_flops / (iter_time / 1e3) / (s.world_size * peak_tflops * 1e12)
        tgs = tokens / (iter_time / 1e3) / s.world_size
        tflops_per_gpu = total_flops / (iter_time / 1e3) / s.world_size / 1e12

        out = Result()
        out.update(
            dict(
                iter_time=iter_time,
                pipeline_time=pipeline_time,
                bubble_time=bubble_time,
                p2p_time_per_hop=p2p_time,
                dp_time=dp_time,
                dp_time_raw=dp_time_raw,
                optim_time=optim_time,
                straggler_ratio=straggler,
                mfu=mfu,
                mfu_6nd_with_attn=mfu,
                tgs=tgs,
                tflops_per_gpu=tflops_per_gpu,
                tokens_per_iter=tokens,
                flops_per_token=flops_token,
                per_stage=per_stage,
                chunk_fwd_times=fwd,
                chunk_bwd_times=bwd,
            )
        )
        return out

    # ---- summary ---------------------------------------------------------
    def analysis(self, save_path: Optional[str] = None) -> Result:
        mem = self.analysis_mem()
        cost = self.analysis_cost()
        res = Result()
        res["mem_result"] = mem
        res["compute_result"] = cost
        res["base_info"] = {
            "model": self.model_config.model_name,
            "system": self.system.sys_name,
            "parallelism": self.strategy.parallelism,
            "global_batch_size": self.strategy.global_batch_size,
            "seq_len": self.strategy.seq_len,
        }
        summary = (
            f"=== {self.model_config.model_name} on {self.system.sys_name} ===\n"
            f"parallelism: {self.strategy.parallelism}\n"
            f"iter_time: {cost['iter_time']:.2f} ms  MFU: {cost['mfu']*100:.2f}%  "
            f"TGS: {cost['tgs']:.1f} tokens/s/gpu  "
            f"TFLOPS/GPU: {cost['tflops_per_gpu']:.1f}\n"
            f"peak_mem: {mem['max_peak_mem_str']} (stage {mem['max_peak_stage']})"
        )
        print(summary)
        if save_path:
            os.makedirs(save_path, exist_ok=True)
            with open(os.path.join(save_path, "compute_result.json"), "w") as f:
                json.dump(human_readable_result(dict(cost)), f, indent=2, default=str)
            with open(os.path.join(save_path, "mem_result.json"), "w") as f:
                json.dump(dict(mem), f, indent=2, default=str)
            with open(os.path.join(save_path, "base_info.json"), "w") as f:
                json.dump(res["base_info"], f, indent=2, default=str)
            with open(os.path.join(save_path, "net_info.json"), "w") as f:
                json.dump(self.system.real_comm_bw, f, indent=2, default=str)
            with open(os.path.join(save_path, "efficiency_coverage.json"), "w") as f:
                json.dump({"miss_efficiency": self.system.miss_efficiency,
                           "hit_efficiency": self.system.hit_efficiency},
                          f, indent=2, default=str)
            with open(os.path.join(save_path, "model_arch"), "w") as f:
                f.write(repr(self.chunks[0]))
        return res

    # ---- op-info dump (calibration enumeration) --------------------------
    def analysis_op_info(self) -> Dict:
        assert self._estimated
        merged: Dict[str, Dict] = {}
        for chunk in self.chunks:
            for op, shapes in chunk.analysis_op_info().items():
                merged.setdefault(op, {}).update(shapes)
        return merged

    # ---- simulate (L6 event-driven replay) -------------------------------
    def simulate(self, save_path: str, merge_lanes: bool = True):
        from ..sim.runner import run_simulation

        return run_simulation(self, save_path, merge_lanes=merge_lanes)

    # ---- search APIs ------------------------------------------------------
    def search_max_micro_batch_size(self, max_mbs: int = 32) -> Optional[int]:
        """Largest mbs that fits memory, fixed micro_batch_num."""
        best = None
        base = deepcopy(self.strategy)
        for mbs in range(1, max_mbs + 1):
            st = deepcopy(base)
            st.micro_batch_size = mbs
            try:
                self.configure(st, self.model_config, self.system)
                self.run_estimate()
                mem = self.analysis_mem()
            except AssertionError:
                continue
            if not mem["oom"]:
                best = mbs
            else:
                break
        if best is not None:
            st = deepcopy(base)
            st.micro_batch_size = best
            self.configure(st, self.model_config, self.system)
            self.run_estimate()
        return best

    def search_max_micro_batch_size_fixed_gbs(self, global_batch_size: int,
                                              max_mbs: int = 32) -> Optional[int]:
        best = None
        base = deepcopy(self.strategy)
        for mbs in range(1, max_mbs + 1):
            per_dp = global_batch_size // base.dp_size
            if per_dp % mbs != 0:
                continue
            st = deepcopy(base)
            st.micro_batch_size = mbs
            st.micro_batch_num = per_dp // mbs
            try:
                self.configure(st, self.model_config, self.system)
                self.run_estimate()
                mem = self.analysis_mem()
            except AssertionError:
                continue
            if not mem["oom"]:
                best = mbs
        if best is not None:
            per_dp = global_batch_size // base.dp_size
            st = deepcopy(base)
            st.micro_batch_size = best
            st.micro_batch_num = per_dp // best
            self.configure(st, self.model_config, self.system)
            self.run_estimate()
        return best

    # ---- recompute-aware search family (ref perf_llm.py:3213-3355) -------
    def _stage_peak_gbytes(self, mem: Result) -> float:
        return max(st["peak_mem"] for st in mem["stages_raw"]) / 1024**3

    def _search_record(self, cost: Result, mem: Result) -> Result:
        s = self.strategy
        return Result(
            parallelism=s.parallelism,
            tp=s.tp_size, pp=s.pp_size, ep=s.ep_size, dp=s.dp_size,
            micro_batch_size=s.micro_batch_size,
            micro_batch_num=s.micro_batch_num,
            recompute_granularity=s.recompute_granularity,
            recompute_layer_num=s.recompute_layer_num,
            attn_recompute=s.attn_recompute,
            mla_rms_recompute=s.mla_rms_recompute,
            mlp_recompute=s.mlp_recompute,
            mlp_rms_recompute=s.mlp_rms_recompute,
            mfu=cost["mfu"], iter_time=cost["iter_time"],
            peak_mem=mem["max_peak_mem"],
        )

    def search_best_strategy_no_recompute(self, gmi_error: float = 6.0,
                                          best_mfu: float = 0.0,
                                          all_search_result: Optional[list] = None,
                                          save_path: Optional[str] = None) -> Result:
        """Evaluate the current parallel strategy without recompute; returns
        the record if it fits in memory minus the gmi_error margin (GiB
        reserved for RCCL buffers / allocator overhead the model doesn't
        price — ref perf_llm.py:3330-3355)."""
        s = self.strategy
        s.enable_recompute = False
        s.recompute_granularity = None
        s.recompute_layer_num = 0
        budget = self.system.accelerator.mem_gbs - gmi_error
        self.run_estimate()
        mem = self.analysis_mem()
        cost = self.analysis_cost()
        if self._stage_peak_gbytes(mem) > budget:
            return Result()
        rec = self._search_record(cost, mem)
        if all_search_result is not None:
            all_search_result.append(rec)
        if cost["mfu"] > best_mfu and save_path is not None:
            self.analysis(save_path)
        return rec if cost["mfu"] > best_mfu else Result()

    def search_best_recompute_layer_num(self, layer_num: Optional[int] = None,
                                        use_reserved_memory: bool = True,
                                        gmi_error: float = 6.0,
                                        best_mfu: float = 0.0,
                                        all_search_result: Optional[list] = None,
                                        save_path: Optional[str] = None) -> Result:
        """Binary-search the smallest per-stage full-recompute layer count
        that fits memory (fewer recomputed layers = higher MFU, so the
        smallest feasible count is MFU-optimal); tracks the best MFU seen
        (ref perf_llm.py:3270-3330)."""
        del use_reserved_memory  # mem_factor already covers reserved margin
        s = self.strategy
        layer_num = layer_num or self.model_config.layer_num
        budget = self.system.accelerator.mem_gbs - gmi_error
        ori = (s.enable_recompute, s.recompute_granularity, s.recompute_layer_num)
        s.enable_recompute = True
        s.recompute_granularity = "full_block"
        best = Result()
        left, right = 0, math.ceil(layer_num / s.pp_size)
        while left <= right:
            n = (left + right) // 2
            # n == 0 probes the no-recompute point (the model treats
            # recompute_layer_num 0 under full_block as "all layers", the
            # config-parity default)
            s.enable_recompute = n > 0
            s.recompute_layer_num = n
            self.run_estimate()
            mem = self.analysis_mem()
            cost = self.analysis_cost()
            if self._stage_peak_gbytes(mem) > budget:
                left = n + 1
                continue
            right = n - 1
            rec = self._search_record(cost, mem)
            if all_search_result is not None:
                all_search_result.append(rec)
            if cost["mfu"] >= best_mfu:
                best_mfu = cost["mfu"]
                best = rec
                if save_path is not None:
                    self.analysis(save_path)
        (s.enable_recompute, s.recompute_granularity,
         s.recompute_layer_num) = ori
        return best

    # the three selective combinations the reference's search walks
    # (perf_llm.py:3227-3246): everything, attention-side only, mlp-side only
    _SELECTIVE_COMBOS = (
        dict(attn_recompute=True, mla_rms_recompute=True,
             mlp_recompute=True, mlp_rms_recompute=True),
        dict(attn_recompute=True, mla_rms_recompute=True,
             mlp_recompute=False, mlp_rms_recompute=False),
        dict(attn_recompute=False, mla_rms_recompute=False,
             mlp_recompute=True, mlp_rms_recompute=True),
    )

    def search_best_selective_recompute(self, use_reserved_memory: bool = True,
                                        gmi_error: float = 6.0,
                                        best_mfu: float = 0.0,
                                        all_search_result: Optional[list] = None,
                                        save_path: Optional[str] = None) -> Result:
        """Walk the curated selective-recompute combinations, keep the best
        feasible MFU (ref perf_llm.py:3213-3267)."""
        del use_reserved_memory
        if self.strategy.megatron_recompute:
            raise NotImplementedError(
                "search does not support megatron_recompute; evaluate those "
                "strategies explicitly")
        s = self.strategy
        budget = self.system.accelerator.mem_gbs - gmi_error
        ori = (s.enable_recompute, s.recompute_granularity, s.attn_recompute,
               s.mla_rms_recompute, s.mlp_recompute, s.mlp_rms_recompute)
        s.enable_recompute = True
        s.recompute_granularity = "selective_recompute"
        s.recompute_layer_num = 0
        best = Result()
        for combo in self._SELECTIVE_COMBOS:
            for k, v in combo.items():
                setattr(s, k, v)
            self.run_estimate()
            mem = self.analysis_mem()
            cost = self.analysis_cost()
            if self._stage_peak_gbytes(mem) > budget:
                continue
            rec = self._search_record(cost, mem)
            if all_search_result is not None:
                all_search_result.append(rec)
            if cost["mfu"] > best_mfu:
                best_mfu = cost["mfu"]
                best = rec
                if save_path is not None:
                    self.analysis(save_path)
        (s.enable_recompute, s.recompute_granularity, s.attn_recompute,
         s.mla_rms_recompute, s.mlp_recompute, s.mlp_rms_recompute) = ori
        return best

    def search_best_parallel_strategy(self, world_size: int, global_batch_size: int,
                                      tp_candidates=(1, 2, 4, 8),
                                      pp_candidates=(1, 2, 4, 8),
                                      ep_candidates=(1,),
                                      recompute_search_type=("no_recompute",),
                                      gmi_error: float = 6.0,
                                      probe_mbs: bool = False,
                                      max_mbs: int = 32,
                                      all_search_result: Optional[list] = None,
                                      verbose=False) -> Optional[Result]:
        """Grid-search tp/pp/ep; per candidate optionally probe the largest
        memory-feasible micro-batch size at fixed global batch, then search
        the requested recompute families (ref perf_llm.py:3355-3578)."""
        if isinstance(recompute_search_type, str):
            recompute_search_type = (recompute_search_type,)
        base_strategy = deepcopy(self.strategy)
        model_cfg = deepcopy(self.model_config)
        system_cfg = deepcopy(self.system)
        best = None
        for tp in tp_candidates:
            for pp in pp_candidates:
                for ep in ep_candidates:
                    if tp * pp > world_size:
                        continue
                    st = deepcopy(base_strategy)
                    st.world_size = world_size
                    st.tp_size, st.pp_size, st.ep_size = tp, pp, ep
                    try:
                        st.sanity_check()
                        dp = st.dp_size
                        if global_batch_size % dp != 0:
                            continue
                        self.configure(st, model_cfg, system_cfg)
                        self.strategy.reset_global_batch_size(global_batch_size)
                        if probe_mbs:
                            # memory-guarded probe: the largest mbs whose
                            # analysis_mem fits, at fixed global batch
                            if self.search_max_micro_batch_size_fixed_gbs(
                                    global_batch_size, max_mbs=max_mbs) is None:
                                continue
                        self.run_estimate()
                    except (AssertionError, ZeroDivisionError):
                        continue
                    best_mfu = best["mfu"] if best else 0.0
                    for kind in recompute_search_type:
                        try:
                            if kind == "no_recompute":
                                rec = self.search_best_strategy_no_recompute(
                                    gmi_error=gmi_error, best_mfu=best_mfu,
                                    all_search_result=all_search_result)
                            elif kind == "full_block":
                                rec = self.search_best_recompute_layer_num(
                                    gmi_error=gmi_error, best_mfu=best_mfu,
                                    all_search_result=all_search_result)
                            elif kind == "selective_recompute":
                                rec = self.search_best_selective_recompute(
                                    gmi_error=gmi_error, best_mfu=best_mfu,
                                    all_search_result=all_search_result)
                            else:
                                raise ValueError(f"unknown recompute search "
                                                 f"type: {kind}")
                        except (AssertionError, ZeroDivisionError):
                            continue
                        if rec and (best is None or rec["mfu"] > best["mfu"]):
                            best = rec
                            best_mfu = rec["mfu"]
                        if verbose and rec:
                            print(f"tp{tp} pp{pp} ep{ep} {kind}: "
                                  f"MFU {rec['mfu']*100:.2f}% "
                                  f"iter {rec['iter_time']:.1f} ms")
        return best


def _zero_div_dense(s: StrategyConfig):
    return s.dp_size * s.cp_size if s.zero_state >= 1 else 1


def _zero_div_moe(s: StrategyConfig):
    return s.edp_size if s.zero_state >= 1 else 1
