"""Layer-profile cache for fast strategy search.

Parity target: the reference's search-profile caches
(simumax/core/perf_llm.py:69-252, 837-1379 — CachedChunkProfile /
CachedUnitRuntimeProfile keyed by a strategy projection with
assembly-only fields stripped): per-layer cost/memory profiles are
pp-independent, so a search over pp values reuses one profiled estimate.

Usage: FastEstimator.rank(strategy) gives an (iter_time, mfu, peak_mem)
estimate assembled from cached per-layer profiles; exact PerfLLM runs are
reserved for the final candidates.
"""

from __future__ import annotations

import json
from copy import deepcopy
from typing import Dict

from ..core.config import ModelConfig, StrategyConfig, SystemConfig
from ..core.utils import get_pp_p2p_comm_size, stage_layers
from ..perf.perf_llm import (PerfLLM, estimate_straggler_increase_ratio,
                             get_effective_straggler_sample_count,
                             schedule_1f1b)

# strategy fields that do NOT affect per-layer profiles (assembly-only)
_ASSEMBLY_FIELDS = {
    "pp_size", "interleaving_size", "micro_batch_num",
    "num_layers_in_first_pipeline_stage", "num_layers_in_last_pipeline_stage",
    "account_for_embedding_in_pipeline_split",
    "account_for_loss_in_pipeline_split", "pp_net", "pp_comm_async",
    "microbatch_group_size_per_vp_stage", "recompute_layer_num",
}


def projection_key(strategy: StrategyConfig, model_cfg: ModelConfig,
                   system: SystemConfig) -> str:
    from dataclasses import fields

    d = {f.name: getattr(strategy, f.name) for f in fields(strategy)
         if f.name not in _ASSEMBLY_FIELDS
         and not f.name.startswith("_")}
    # world_size enters only through dp/edp sharding of state; keep it
    d["__model"] = model_cfg.model_name, model_cfg.layer_num
    d["__system"] = system.sys_name
    return json.dumps(d, sort_keys=True, default=str)


class LayerProfileCache:
    def __init__(self):
        self._cache: Dict[str, dict] = {}
        self.hits = 0
        self.misses = 0

    def profile(self, strategy: StrategyConfig, model_cfg: ModelConfig,
                system: SystemConfig) -> dict:
        key = projection_key(strategy, model_cfg, system)
        if key in self._cache:
            self.hits += 1
            return self._cache[key]
        self.misses += 1
        # run a pp=1 estimate on a 2-layer clone; block[-1] is an interior
        # layer (MoE models: a MOE layer), ends = chunk minus layers
        st = deepcopy(strategy)
        st.pp_size = 1
        st.interleaving_size = 1
        st.micro_batch_num = 1
        st.num_layers_in_first_pipeline_stage = None
        st.num_layers_in_last_pipeline_stage = None
        mc = deepcopy(model_cfg)
        mc.layer_num = max(2, min(2 + mc.dense_layers, model_cfg.layer_num))
        p = PerfLLM()
        p.configure(st, mc, system)
        p.run_estimate()
        chunk = p.chunks[0]
        layer = chunk.blocks[-1]
        lci, lai, lmi = (layer.get_cost_info(), layer.get_act_info(),
                         layer.get_model_info())
        tci, tmi = chunk.get_cost_info(), chunk.get_model_info()
        n_built = len(chunk.blocks)
        prof = dict(
            layer_fwd=lci.fwd_time,
            layer_bwd=lci.bwd_time + lci.recompute_time,
            layer_cache=lai.activation_mem_cache,
            layer_peak_extra=max(lai.fwd_peak_mem_no_cache,
                                 lai.bwd_peak_mem_no_cache),
            layer_model=lmi.all_bytes,
            layer_params=lmi.weight_bytes / 2,
            ends_fwd=tci.fwd_time - n_built * lci.fwd_time,
            ends_bwd=(tci.bwd_time + tci.recompute_time)
                     - n_built * (lci.bwd_time + lci.recompute_time),
            ends_cache=chunk.peak_point.cache_mem
                       - n_built * lai.activation_mem_cache,
            ends_model=tmi.all_bytes - n_built * lmi.all_bytes,
            ends_params=(tmi.weight_bytes - n_built * lmi.weight_bytes) / 2,
            perf=p,  # keeps the configured system for net pricing
        )
        self._cache[key] = prof
        return prof


class FastEstimator:
    """Assemble (iter_time, mfu, peak_mem) for any pp/mbc from cached
    per-layer profiles. Accuracy: within a few % of the exact estimate —
    use for RANKING, confirm winners with PerfLLM."""

    def __init__(self, model_cfg: ModelConfig, system: SystemConfig):
        self.model_cfg = model_cfg
        self.system = system
        self.cache = LayerProfileCache()

    def estimate(self, strategy: StrategyConfig) -> dict:
        prof = self.cache.profile(strategy, self.model_cfg, self.system)
        s = strategy
        pp, mbc = s.pp_size, s.micro_batch_num
        mc = deepcopy(self.model_cfg)
        mc.maybe_pad_vocab_size(s.tp_size)
        layers = stage_layers(s, mc)
        fwd = [n * prof["layer_fwd"] + (prof["ends_fwd"] if i in (0, pp - 1) and pp == 1
               else (prof["ends_fwd"] * 0.5 if i in (0, pp - 1) else 0.0))
               for i, n in enumerate(layers)]
        bwd = [n * prof["layer_bwd"] + (prof["ends_bwd"] if i in (0, pp - 1) and pp == 1
               else (prof["ends_bwd"] * 0.5 if i in (0, pp - 1) else 0.0))
               for i, n in enumerate(layers)]
        p2p = 0.0
        perf = prof["perf"]
        if pp > 1:
            p2p = perf.system.compute_net_op_time(
                "p2p", get_pp_p2p_comm_size(s, mc), 2,
                net="high_intra_node", comm_stage="pp", strategy=perf.strategy)
        if pp == 1:
            pipeline = mbc * (fwd[0] + bwd[0])
        else:
            pipeline, _ = schedule_1f1b(pp, mbc, fwd, bwd, p2p)
        n = get_effective_straggler_sample_count(s, self.system.num_per_node)
        if s.enable_straggler_model:
            pipeline *= estimate_straggler_increase_ratio(n)
        # dp + optimizer from params
        total_params = [n_ * prof["layer_params"] +
                        (prof["ends_params"] if i in (0, pp - 1) else 0)
                        for i, n_ in enumerate(layers)]
        from ..core.consts import OPTIMIZER_TRAFFIC_BYTES_PER_PARAM

        optim = max(
            self.system.compute_mem_access_time(
                "optimizer",
                pr / (s.dp_size * s.cp_size if s.zero_state else 1)
                * OPTIMIZER_TRAFFIC_BYTES_PER_PARAM)
            for pr in total_params)
        dp_time = 0.0
        if s.dp_size > 1:
            grad_e = 2 if s.grad_reduce_in_bf16 else 4
            for i, pr in enumerate(total_params):
                op = "reduce_scatter" if s.zero_state else "all_reduce"
                t = perf.system.compute_net_op_time(
                    op, pr * grad_e, s.dp_size * s.cp_size,
                    net="high_intra_node", comm_stage="dp_cp",
                    strategy=perf.strategy)
                if s.zero_state:
                    t += perf.system.compute_net_op_time(
                        "all_gather", pr * 2, s.dp_size * s.cp_size,
                        net="high_intra_node", comm_stage="dp_cp",
                        strategy=perf.strategy)
                dp_time = max(dp_time, t)
            if s.overlap_grad_reduce:
                dp_time = max(0.0, dp_time - max(bwd))
        iter_time = pipeline + dp_time + optim
        # memory
        peaks = []
        for i, n_ in enumerate(layers):
            model = n_ * prof["layer_model"] + (
                prof["ends_model"] if i in (0, pp - 1) else 0)
            cache = n_ * prof["layer_cache"] + (
                prof["ends_cache"] if i in (0, pp - 1) else 0)
            inflight = min(mbc, pp - i) if pp > 1 else 1
            peak = (model + (inflight - 1) * cache + cache
                    + prof["layer_peak_extra"]) / s.mem_factor
            peaks.append(peak)
        tokens = s.global_batch_size * s.seq_len
        flops = mc.flops_per_token(s.seq_len) * tokens
        mfu = flops / (iter_time / 1e3) / (
            s.world_size * self.system.accelerator.op["matmul"].tflops * 1e12)
        return dict(iter_time=iter_time, mfu=mfu, peak_mem=max(peaks),
                    pipeline_time=pipeline, dp_time=dp_time, optim_time=optim)
