"""StrategySearcher: grid search over parallelism + recompute with
network auto-selection, max-mbs probing, and MFU ranking.

Parity target: simumax/tuning/strategy_searcher.py:33-216.
"""

from __future__ import annotations

from copy import deepcopy
from dataclasses import dataclass, field
from typing import List, Optional, Sequence

from ..core.config import ModelConfig, StrategyConfig, SystemConfig
from ..core.records import Result
from ..perf.perf_llm import PerfLLM


@dataclass
class SearchSpace:
    tp: Sequence[int] = (1, 2, 4, 8)
    pp: Sequence[int] = (1, 2, 4, 8)
    ep: Sequence[int] = (1,)
    cp: Sequence[int] = (1,)
    recompute: Sequence[Optional[str]] = (None, "selective_recompute",
                                          "full_block")
    max_mbs: int = 8


@dataclass
class SearchResult:
    rows: List[Result] = field(default_factory=list)

    @property
    def best(self) -> Optional[Result]:
        return max(self.rows, key=lambda r: r["mfu"]) if self.rows else None

    def top(self, n=10):
        return sorted(self.rows, key=lambda r: -r["mfu"])[:n]


class StrategySearcher:
    def __init__(self, model_config: ModelConfig, system_config: SystemConfig,
                 base_strategy: StrategyConfig):
        self.model_config = model_config
        self.system_config = system_config
        self.base = base_strategy

    def search(self, world_size: int, global_batch_size: int,
               space: SearchSpace = None, verbose: bool = False,
               fast_prefilter: bool = True, exact_top_k: int = 8) -> SearchResult:
        """Grid search. With fast_prefilter, candidates are first RANKED by
        the layer-profile cache (tuning.profile_cache, the reference's
        CachedChunkProfile analog) and only the top exact_top_k get the
        exact PerfLLM evaluation."""
        space = space or SearchSpace()
        if fast_prefilter:
            ranked = self._fast_rank(world_size, global_batch_size, space,
                                     verbose)
            out = SearchResult()
            for cand in ranked[:exact_top_k]:
                row = self._evaluate(world_size, global_batch_size, *cand,
                                     space)
                if row is not None:
                    out.rows.append(row)
                    if verbose:
                        print(f"[exact] tp{cand[0]} pp{cand[1]} ep{cand[2]} "
                              f"cp{cand[3]} rc={cand[4]}: "
                              f"MFU {row['mfu']*100:.2f}%")
            return out
        out = SearchResult()
        for tp in space.tp:
            for pp in space.pp:
                for ep in space.ep:
                    for cp in space.cp:
                        for rc in space.recompute:
                            row = self._evaluate(world_size, global_batch_size,
                                                 tp, pp, ep, cp, rc, space)
                            if row is not None:
                                out.rows.append(row)
                                if verbose:
                                    print(f"tp{tp} pp{pp} ep{ep} cp{cp} "
                                          f"rc={rc}: MFU {row['mfu']*100:.2f}% "
                                          f"mbs{row['mbs']} "
                                          f"peak {row['peak_mem']/2**30:.1f} GiB")
        return out

    def _fast_rank(self, world, gbs, space, verbose):
        from copy import deepcopy as _dc

        from .profile_cache import FastEstimator

        fe = FastEstimator(self.model_config, self.system_config)
        scored = []
        budget = self.system_config.accelerator.mem_gbs * 0.94 * 1024**3
        for tp in space.tp:
            for pp in space.pp:
                for ep in space.ep:
                    for cp in space.cp:
                        for rc in space.recompute:
                            st = _dc(self.base)
                            st.world_size = world
                            st.tp_size, st.pp_size = tp, pp
                            st.ep_size, st.cp_size = ep, cp
                            st.enable_recompute = rc is not None
                            st.recompute_granularity = rc
                            for a in ("tp_net", "cp_net", "pp_net", "dp_net",
                                      "ep_net", "etp_net", "edp_net"):
                                setattr(st, a, "high_intra_node")
                            try:
                                st.sanity_check()
                                if gbs % st.dp_size != 0:
                                    continue
                                st.micro_batch_size = 1
                                st.micro_batch_num = gbs // st.dp_size
                                if st.pp_size > 1 and st.micro_batch_num < st.pp_size:
                                    continue
                                if self.model_config.layer_num % st.pp_size:
                                    continue
                                est = fe.estimate(st)
                            except (AssertionError, ZeroDivisionError,
                                    RuntimeError):
                                continue
                            if est["peak_mem"] > budget:
                                continue
                            scored.append((est["mfu"], (tp, pp, ep, cp, rc)))
                            if verbose:
                                print(f"[fast]  tp{tp} pp{pp} ep{ep} cp{cp} "
                                      f"rc={rc}: ~MFU {est['mfu']*100:.2f}%")
        scored.sort(key=lambda x: -x[0])
        if verbose:
            print(f"[fast] profiles: {fe.cache.misses} built, "
                  f"{fe.cache.hits} cache hits")
        return [c for _, c in scored]

    def _evaluate(self, world, gbs, tp, pp, ep, cp, recompute, space):
        st = deepcopy(self.base)
        st.world_size = world
        st.tp_size, st.pp_size, st.ep_size, st.cp_size = tp, pp, ep, cp
        st.enable_recompute = recompute is not None
        st.recompute_granularity = recompute
        if recompute == "full_block":
            st.recompute_layer_num = 0
        # reset auto nets so analysis_net re-picks tiers for this shape
        for a in ("tp_net", "cp_net", "pp_net", "dp_net", "ep_net",
                  "etp_net", "edp_net"):
            setattr(st, a, "auto")
        try:
            st.sanity_check()
            if gbs % st.dp_size != 0:
                return None
            perf = PerfLLM()
            perf.configure(st, self.model_config, self.system_config)
            mbs = perf.search_max_micro_batch_size_fixed_gbs(
                gbs, max_mbs=space.max_mbs)
            if mbs is None:
                return None
            mem = perf.analysis_mem()
            cost = perf.analysis_cost()
        except (AssertionError, ZeroDivisionError):
            return None
        return Result(
            tp=tp, pp=pp, ep=ep, cp=cp, dp=perf.strategy.dp_size, mbs=mbs,
            mbc=perf.strategy.micro_batch_num, recompute=recompute,
            mfu=cost["mfu"], iter_time=cost["iter_time"],
            tgs=cost["tgs"], peak_mem=mem["max_peak_mem"],
            parallelism=perf.strategy.parallelism,
        )
