"""simumax_amd: MI355X-native analytical LLM-training simulator.

A from-scratch rebuild of the SimuMax capability set (PerfLLM API,
system/strategy/model config schema, trace & memory-snapshot artifacts)
with the cost model derived for CDNA4 (MFMA, 288 GB HBM3E, RCCL over the
8-GPU xGMI mesh) and the calibration harness built on hand-written
HIP/gfx950 kernels.
"""

__version__ = "0.1.0"

from .core.config import ModelConfig, StrategyConfig, SystemConfig  # noqa: F401
from .perf.perf_llm import PerfLLM  # noqa: F401
from .registry import (  # noqa: F401
    get_simu_model_config,
    get_simu_strategy_config,
    get_simu_system_config,
    show_simu_model_configs,
    show_simu_strategy_configs,
    show_simu_system_configs,
)


def config_from_names(model: str, strategy: str, system: str):
    """Resolve registry names to the (strategy, model, system) config
    triple in `PerfLLM.configure` argument order:

        perf = PerfLLM()
        perf.configure(*config_from_names("llama3-8b",
                                          "tp1_pp2_dp4_mbs1", "mi355x"))
    """
    return (
        StrategyConfig.init_from_config_file(get_simu_strategy_config(strategy)),
        ModelConfig.init_from_config_file(get_simu_model_config(model)),
        SystemConfig.init_from_config_file(get_simu_system_config(system)),
    )
