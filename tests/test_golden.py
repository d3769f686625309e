"""Golden-result regression: analysis outputs pinned against committed
goldens (the reference's ResultCheck pattern; regenerate with
`python tests/test_golden.py --regen` after INTENDED cost-model changes)."""

import json
import os
import sys

import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from simumax_amd import (ModelConfig, PerfLLM, StrategyConfig, SystemConfig,
                         get_simu_model_config, get_simu_strategy_config,
                         get_simu_system_config)
from simumax_amd.testing.base_test_tool import ResultCheck

GOLDEN_DIR = os.path.join(os.path.dirname(__file__), "goldens")

CASES = [
    ("llama3-8b", "tp1_pp2_dp4_mbs1"),
    ("llama3-8b", "tp8_pp1_dp1_mbs1"),
    ("llama3-8b", "tp1_pp4_vp2_sync_mbs1_mbc8"),
    ("llama3-70b-l12", "tp2_pp2_dp2_mbs1_selective"),
    ("llama3-405b", "tp8_pp1_dp1_mbs1"),
    ("qwen3-32b-l12", "tp2_pp1_dp4_mbs1"),
    ("mixtral-8x7b-l8", "ep8_pp1_dp8_mbs1"),
    ("deepseekv2-l4", "ep8_pp1_dp8_mbs1"),
    ("deepseekv2-l4", "ep4_pp2_dp4_mbs1"),
]


def run_case(model, strategy):
    p = PerfLLM()
    p.configure(
        StrategyConfig.init_from_config_file(get_simu_strategy_config(strategy)),
        ModelConfig.init_from_config_file(get_simu_model_config(model)),
        SystemConfig.init_from_config_file(get_simu_system_config("mi355x")),
    )
    p.run_estimate()
    cost = p.analysis_cost()
    mem = p.analysis_mem()
    return {
        "iter_time": cost["iter_time"],
        "mfu": cost["mfu"],
        "pipeline_time": cost["pipeline_time"],
        "dp_time": cost["dp_time"],
        "optim_time": cost["optim_time"],
        "bubble_time": cost["bubble_time"],
        "max_peak_mem": mem["max_peak_mem"],
        "stage_peaks": [st["peak_mem"] for st in mem["stages_raw"]],
        "stage_weights": [st["weight_mem"] for st in mem["stages_raw"]],
    }


@pytest.mark.parametrize("model,strategy", CASES)
def test_golden(model, strategy):
    path = os.path.join(GOLDEN_DIR, f"{model}__{strategy}.json")
    assert os.path.exists(path), (
        f"golden missing: regenerate with `python {__file__} --regen`")
    with open(path) as f:
        golden = json.load(f)
    got = run_case(model, strategy)
    rc = ResultCheck(rel_tol=1e-6)
    assert rc.check(got, golden), (
        f"cost-model drift vs golden {path}:\n{rc.report()}\n"
        "If the change is intended, regenerate the goldens.")


if __name__ == "__main__":
    if "--regen" in sys.argv:
        os.makedirs(GOLDEN_DIR, exist_ok=True)
        for model, strategy in CASES:
            path = os.path.join(GOLDEN_DIR, f"{model}__{strategy}.json")
            with open(path, "w") as f:
                json.dump(run_case(model, strategy), f, indent=1)
            print("wrote", path)
