"""DeepSeek-V2 (4-layer) MoE+MLA, EP8 on MI355X."""
import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from simumax_amd import (ModelConfig, PerfLLM, StrategyConfig, SystemConfig,
                         get_simu_model_config, get_simu_strategy_config,
                         get_simu_system_config)

perf = PerfLLM()
perf.configure(
    StrategyConfig.init_from_config_file(get_simu_strategy_config("ep8_pp1_dp8_mbs1")),
    ModelConfig.init_from_config_file(get_simu_model_config("deepseekv2-l4")),
    SystemConfig.init_from_config_file(get_simu_system_config("mi355x")),
)
perf.run_estimate()
perf.analysis(f"{perf.model_config.model_name}_{perf.system.sys_name}")
