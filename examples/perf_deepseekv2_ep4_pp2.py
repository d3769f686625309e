"""DeepSeek-V2 (4-layer), EP4/PP2 on the MI355X system config.

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
            get_simu_strategy_config("ep4_pp2_dp4_mbs1")),
        model_config=ModelConfig.init_from_config_file(
            get_simu_model_config("deepseekv2-l4")),
        system_config=SystemConfig.init_from_config_file(
            get_simu_system_config("mi355x")),
    )
    perf_model.run_estimate()
    name = f"{perf_model.model_config.model_name}_{perf_model.system.sys_name}"
    perf_model.analysis(name)


if __name__ == "__main__":
    main()
