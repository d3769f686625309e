"""Strategy search with recompute families on Llama-3 70B (the reference's
search workflow: examples/search/llm_search.py + perf_llm.py:3213-3578).

Grid-searches tp/pp with a memory-guarded micro-batch probe, then per
candidate walks no-recompute, binary-searched full-block layer counts and
the curated selective-recompute combinations, ranking by MFU under the
288 GB HBM3E budget minus a 6 GiB runtime margin."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from simumax_amd import (ModelConfig, PerfLLM, StrategyConfig, SystemConfig,
                         get_simu_model_config, get_simu_strategy_config,
                         get_simu_system_config)


def main():
    p = PerfLLM()
    p.configure(
        StrategyConfig.init_from_config_file(
            get_simu_strategy_config("tp2_pp2_dp2_mbs1_selective")),
        ModelConfig.init_from_config_file(get_simu_model_config("llama3-70b")),
        SystemConfig.init_from_config_file(get_simu_system_config("mi355x")),
    )
    all_results = []
    best = p.search_best_parallel_strategy(
        world_size=8, global_batch_size=32,
        tp_candidates=(1, 2, 4), pp_candidates=(1, 2, 4),
        recompute_search_type=("no_recompute", "full_block",
                               "selective_recompute"),
        probe_mbs=True, gmi_error=6.0,
        all_search_result=all_results, verbose=True)
    print(f"\nevaluated {len(all_results)} feasible candidates")
    if best:
        print(f"best: {best['parallelism']}")
        print(f"  recompute: {best['recompute_granularity']} "
              f"(layers={best['recompute_layer_num']})")
        print(f"  MFU {best['mfu']*100:.2f}%  iter {best['iter_time']:.1f} ms  "
              f"peak {best['peak_mem']/2**30:.1f} GiB")


if __name__ == "__main__":
    main()
