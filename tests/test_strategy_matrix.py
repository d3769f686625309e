"""Strategy-space smoke matrix: estimate() must survive a broad grid of
parallelism combinations with sane outputs (positive times, memory
under the device ceiling treated as data, not crash)."""

import copy

import pytest

from simumax_amd import (ModelConfig, PerfLLM, StrategyConfig, SystemConfig,
                         get_simu_model_config, get_simu_system_config)

DENSE = ModelConfig.init_from_config_file(get_simu_model_config("llama2-tiny"))
MOE = ModelConfig.init_from_config_file(
    get_simu_model_config("mixtral-8x7b-l8"))

GRID = [
    # (model, world, tp, pp, ep, cp, extra)
    ("dense", 8, 1, 1, 1, 1, {}),
    ("dense", 8, 2, 1, 1, 1, {}),
    ("dense", 8, 2, 1, 1, 1, {"enable_sequence_parallel": True}),
    ("dense", 8, 4, 2, 1, 1, {}),
    ("dense", 8, 1, 4, 1, 1, {}),
    ("dense", 8, 1, 2, 1, 1, {"interleaving_size": 2,
                              "micro_batch_num": 8}),
    ("dense", 8, 1, 1, 1, 2, {}),
    ("dense", 8, 1, 1, 1, 4, {"cp_comm_type": "all_gather"}),
    ("dense", 8, 1, 1, 1, 4, {"cp_comm_type": "ring"}),
    ("dense", 8, 1, 1, 1, 4, {"cp_comm_type": "ring",
                              "cp_sharding": "zigzag"}),
    ("dense", 8, 2, 1, 1, 2, {"enable_sequence_parallel": True}),
    ("dense", 8, 1, 1, 1, 1, {"zero_state": 1}),
    ("dense", 8, 2, 2, 1, 2, {}),
    ("dense", 16, 2, 2, 1, 1, {}),
    ("dense", 32, 2, 4, 1, 2, {}),
    ("dense", 8, 1, 1, 1, 1, {"fp8": True}),
    ("dense", 8, 1, 2, 1, 1, {"enable_recompute": True,
                              "recompute_granularity": "full_block",
                              "recompute_layer_num": 2}),
    ("moe", 8, 1, 1, 2, 1, {}),
    ("moe", 8, 1, 1, 8, 1, {}),
    ("moe", 8, 1, 2, 4, 1, {}),
    ("moe", 8, 2, 1, 2, 1, {"enable_sequence_parallel": True}),
    ("moe", 16, 1, 2, 8, 1, {}),
    ("moe", 8, 1, 1, 4, 2, {}),
    ("moe", 8, 1, 1, 2, 1, {"zero_state": 1}),
]


@pytest.mark.parametrize("model,world,tp,pp,ep,cp,extra", GRID)
def test_strategy_matrix(model, world, tp, pp, ep, cp, extra):
    mc = copy.deepcopy(DENSE if model == "dense" else MOE)
    kw = dict(seq_len=4096, micro_batch_size=1, micro_batch_num=max(2, pp),
              world_size=world, tp_size=tp, pp_size=pp, ep_size=ep,
              cp_size=cp, enable_sequence_parallel=False, zero_state=0,
              use_fp32_accum_grad=True, cross_entropy_loss_fusion=True,
              attention_sparse_ratio=0.5, mem_factor=1.0)
    kw.update(extra)
    st = StrategyConfig(**kw)
    st.sanity_check()
    p = PerfLLM()
    p.configure(st, mc, SystemConfig.init_from_config_file(
        get_simu_system_config("mi355x")))
    p.run_estimate()
    c = p.analysis_cost()
    m = p.analysis_mem()
    assert c["iter_time"] > 0
    assert c["mfu"] > 0
    assert m["max_peak_mem"] > 0
