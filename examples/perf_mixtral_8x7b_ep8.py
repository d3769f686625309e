"""Mixtral-8x7B, EP8 expert parallelism on the MI355X system config.

Parity target: the reference's examples/ perf-script family.
"""
import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from simumax_amd import (ModelConfig, PerfLLM, StrategyConfig, SystemConfig,
                         get_simu_model_config, get_simu_strategy_config,
                         get_simu_system_config)


def main():
    perf_model = PerfLLM()
    perf_model.configure(
        strategy_config=StrategyConfig.init_from_config_file(
            get_simu_strategy_config("ep8_pp1_dp8_mbs1")),
        model_config=ModelConfig.init_from_config_file(
            get_simu_model_config("mixtral-8x7b")),
        system_config=SystemConfig.init_from_config_file(
            get_simu_system_config("mi355x")),
    )
    perf_model.run_estimate()
    name = f"{perf_model.model_config.model_name}_{perf_model.system.sys_name}"
    perf_model.analysis(name)


if __name__ == "__main__":
    main()
