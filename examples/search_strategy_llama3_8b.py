"""Grid-search the best parallel strategy for Llama-3 8B on one MI355X node.

Parity target: /root/reference/examples/search_strategy_llama3_8b.py.
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from simumax_amd import (ModelConfig, StrategyConfig, SystemConfig,
                         get_simu_model_config, get_simu_strategy_config,
                         get_simu_system_config)
from simumax_amd.tuning.strategy_searcher import SearchSpace, StrategySearcher


def main():
    searcher = StrategySearcher(
        ModelConfig.init_from_config_file(get_simu_model_config("llama3-8b")),
        SystemConfig.init_from_config_file(get_simu_system_config("mi355x")),
        StrategyConfig.init_from_config_file(
            get_simu_strategy_config("tp1_pp1_dp8_mbs1")),
    )
    res = searcher.search(world_size=8, global_batch_size=32,
                          space=SearchSpace(tp=(1, 2, 4, 8), pp=(1, 2, 4),
                                            recompute=(None, "selective_recompute")),
                          verbose=True)
    best = res.best
    print(f"\nBEST: {best['parallelism']} rc={best['recompute']} "
          f"MFU {best['mfu']*100:.2f}% iter {best['iter_time']:.1f} ms")


if __name__ == "__main__":
    main()
