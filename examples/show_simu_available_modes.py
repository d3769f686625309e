"""List shipped configs + supported modes (parity:
examples/show_simu_avaliable_modes.py)."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from simumax_amd import (show_simu_model_configs, show_simu_strategy_configs,
                         show_simu_system_configs)
from simumax_amd.core.config import (VALID_CP_A2A_MODES,
                                     VALID_MEGATRON_RECOMPUTE_MODULES,
                                     VALID_RECOMPUTE_GRANULARITY)

print("models:   ", ", ".join(show_simu_model_configs()))
print("strategies:", ", ".join(show_simu_strategy_configs()))
print("systems:  ", ", ".join(show_simu_system_configs()))
print("recompute granularities:", VALID_RECOMPUTE_GRANULARITY)
print("megatron recompute modules:", VALID_MEGATRON_RECOMPUTE_MODULES)
print("cp a2a modes:", VALID_CP_A2A_MODES)
