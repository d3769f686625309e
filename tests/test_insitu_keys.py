"""Contract tests: the in-situ timing registry must emit byte-identical
shape-key strings to the ones the simulator looks up, or the calibration
overlay silently misses. Guards kernels/insitu.py against drift in
core/module.py's key generators."""

import pytest

from simumax_amd.core.config import ModelConfig, StrategyConfig, SystemConfig
from simumax_amd.kernels import insitu
from simumax_amd import (PerfLLM, get_simu_model_config,
                         get_simu_system_config)


@pytest.fixture(scope="module")
def perf():
    mc = ModelConfig.init_from_config_file(get_simu_model_config("llama3-8b"))
    mc.layer_num = 2
    st = StrategyConfig(
        seq_len=4096, micro_batch_size=1, micro_batch_num=1,
        world_size=1, tp_size=1, pp_size=1,
        enable_sequence_parallel=False, zero_state=0,
        use_fp32_accum_grad=True, enable_recompute=False,
        cross_entropy_loss_fusion=True, attention_sparse_ratio=0.5,
        mem_factor=1.0)
    p = PerfLLM()
    p.configure(st, mc, SystemConfig.init_from_config_file(
        get_simu_system_config("mi355x")))
    p.run_estimate()
    return p


def _leaf(perf, cls_name, name_part=""):
    for leaf in perf.chunks[0].leaf_modules():
        if type(leaf).__name__ == cls_name and name_part in leaf.full_name:
            return leaf
    raise AssertionError(f"{cls_name} {name_part} not found")


def test_gemm_fwd_key_matches(perf):
    qkv = _leaf(perf, "LinearCol", "qkv_proj")
    sim_key = qkv.get_input_shapes_desc("fwd")
    # the trainer calls FusedLinear with x [B,S,K] and weight [N,K]
    ins_key = insitu.gemm_key(1, 4096, 4096, 6144, "TN", False, "bf16")
    assert ins_key == sim_key


def test_gemm_dgrad_key_matches(perf):
    qkv = _leaf(perf, "LinearCol", "qkv_proj")
    sim_key = qkv.get_input_shapes_desc("bwd_grad_act")
    ins_key = insitu.gemm_key(1, 4096, 6144, 4096, "NN", False, "bf16")
    assert ins_key == sim_key


def test_gemm_wgrad_key_matches(perf):
    qkv = _leaf(perf, "LinearCol", "qkv_proj")
    sim_key = qkv.get_input_shapes_desc("bwd_grad_w")
    ins_key = insitu.gemm_key(1, 6144, 4096, 4096, "NT", True, "fp32")
    assert ins_key == sim_key


def test_sdp_key_matches(perf):
    core = _leaf(perf, "CoreAttention")
    sim_key = core.get_input_shapes_desc("fwd")
    # the trainer calls flash_attention(q [B,S,Hq,D], k/v [B,S,Hkv,D])
    ins_key = insitu.sdp_key(1, 4096, 32, 8, 128, 128)
    assert ins_key == sim_key


def test_grouped_key_matches():
    from simumax_amd.train.moe import _grouped_key

    mc = ModelConfig.init_from_config_file(
        get_simu_model_config("mixtral-8x7b-l8"))
    st = StrategyConfig(
        seq_len=4096, micro_batch_size=1, micro_batch_num=1,
        world_size=1, tp_size=1, pp_size=1,
        enable_sequence_parallel=False, zero_state=0,
        use_fp32_accum_grad=True, enable_recompute=False,
        cross_entropy_loss_fusion=True, attention_sparse_ratio=0.5,
        mem_factor=1.0)
    p = PerfLLM()
    p.configure(st, mc, SystemConfig.init_from_config_file(
        get_simu_system_config("mi355x")))
    p.run_estimate()
    gl = _leaf(p, "GroupLinearCol")
    sim_key = gl.get_input_shapes_desc("fwd")
    # trainer: x [E, cap, H] @ w1 [E, H, 2I]; cap = N*topk/E
    ins_key = _grouped_key(8, 1024, 4096, 2 * 14336, "fwd")
    assert ins_key == sim_key


def test_efficiency_definitions_roundtrip():
    """insitu eff -> merge -> SystemConfig pricing must reproduce the
    measured time for a GEMM shape."""
    key = insitu.gemm_key(1, 4096, 4096, 6144, "TN", False, "bf16")
    flops = insitu._gemm_flops(key)
    t_ms = 0.25
    eff = flops / (t_ms / 1e3) / (insitu.PEAK_BF16_TFLOPS * 1e12)
    sysc = SystemConfig.init_from_config_file(get_simu_system_config("mi355x"))
    sysc.accelerator.op["matmul"].accurate_efficient_factor[key] = eff
    t = sysc.compute_op_accuracy_time("matmul", flops, shape_desc=key)
    assert abs(t - t_ms) / t_ms < 1e-9


def test_bandwidth_summarize_units():
    """bw_<op> records aggregate to a stream efficiency vs the 8 TB/s
    peak (minus per-launch latency)."""
    import simumax_amd.kernels.insitu as I

    class FakeEvt:
        def __init__(self, t):
            self.t = t

        def elapsed_time(self, other):
            return other.t - self.t

    I._RECORDS.clear()
    # 4 GiB moved in 1 ms -> 4000 GiB/s -> eff 0.5 of 8000 GiB/s
    I._RECORDS[("bw_rmsnorm_fwd", str(4 * 1024**3))] = [
        (FakeEvt(0.0), FakeEvt(1.0 + 0.004), False)]
    import torch

    sync = torch.cuda.synchronize
    torch.cuda.synchronize = lambda: None
    try:
        out = I.summarize()
    finally:
        torch.cuda.synchronize = sync
        I._RECORDS.clear()
    eff = out["bandwidth"]["rmsnorm_fwd_eff"]
    assert abs(eff - 0.5) < 1e-3, eff


def test_mla_sdp_key_matches():
    """The flash wrapper's qkv_contiguous heuristic (Dqk != Dv -> False)
    must match MLACoreAttention's key."""
    from simumax_amd.core.config import ModelConfig
    from simumax_amd.kernels import insitu
    from simumax_amd import (PerfLLM, StrategyConfig, SystemConfig,
                             get_simu_model_config, get_simu_system_config)

    mc = ModelConfig.init_from_config_file(
        get_simu_model_config("deepseekv2-l4"))
    st = StrategyConfig(
        seq_len=4096, micro_batch_size=1, micro_batch_num=1,
        world_size=1, tp_size=1, pp_size=1, ep_size=1,
        enable_sequence_parallel=False, zero_state=0,
        use_fp32_accum_grad=True, enable_recompute=False,
        cross_entropy_loss_fusion=True, attention_sparse_ratio=0.5,
        mem_factor=1.0)
    p = PerfLLM()
    p.configure(st, mc, SystemConfig.init_from_config_file(
        get_simu_system_config("mi355x")))
    p.run_estimate()
    core = next(l for l in p.chunks[0].leaf_modules()
                if type(l).__name__ == "MLACoreAttention")
    sim_key = core.get_input_shapes_desc("fwd")
    ins_key = insitu.sdp_key(1, 4096, 128, 128, 192, 128,
                             contiguous=192 == 128)
    assert ins_key == sim_key


def test_insitu_overlay_applies():
    """apply_insitu_overlay writes per-shape efficiencies and bandwidth
    factors onto a live SystemConfig (bench/validation self-calibration)."""
    from simumax_amd import SystemConfig, get_simu_system_config
    from simumax_amd.calib.insitu_overlay import apply_insitu_overlay

    sysc = SystemConfig.init_from_config_file(get_simu_system_config("mi355x"))
    key = next(iter(sysc.accelerator.op["matmul"].accurate_efficient_factor))
    summary = {
        "matmul": {key: {"eff": 0.1234, "t_ms": 1.0, "n": 1}},
        "bandwidth": {"optimizer_eff": 0.4321, "rmsnorm_fwd_eff": 0.9,
                      "moe_routing_ms": 0.5},
    }
    n = apply_insitu_overlay(sysc, summary)
    assert n >= 4
    assert sysc.accelerator.op["matmul"].accurate_efficient_factor[key] == 0.1234
    assert sysc.accelerator.bandwidth["optimizer"].efficient_factor == 0.4321
    assert sysc.accelerator.bandwidth["rmsnorm_fwd"].efficient_factor == 0.9
    assert sysc.accelerator.bandwidth["moe_routing"].latency_us == 500.0


def _fake_records(I):
    class FakeEvt:
        def __init__(self, t):
            self.t = t

        def elapsed_time(self, other):
            return other.t - self.t

    return FakeEvt


def _summarize_nosync(I):
    import torch

    sync = torch.cuda.synchronize
    torch.cuda.synchronize = lambda: None
    try:
        return I.summarize()
    finally:
        torch.cuda.synchronize = sync
        I._RECORDS.clear()


def test_recompute_rerun_filtering_and_factor():
    """A compute key seen both outside and inside backward is a
    checkpoint rerun: the fwd efficiency median must come from the
    true-fwd instances only, and the rerun/fwd time ratio becomes the
    measured recompute_factor in the meta table."""
    import simumax_amd.kernels.insitu as I

    FakeEvt = _fake_records(I)
    I._RECORDS.clear()
    key = I.gemm_key(1, 4096, 4096, 4096, "TN", False, "bf16")
    # 2 fwd instances at 1.0 ms, 2 rerun instances (in backward) at 0.8
    I._RECORDS[("matmul", key)] = [
        (FakeEvt(0.0), FakeEvt(1.0), False),
        (FakeEvt(0.0), FakeEvt(1.0), False),
        (FakeEvt(0.0), FakeEvt(0.8), True),
        (FakeEvt(0.0), FakeEvt(0.8), True),
    ]
    out = _summarize_nosync(I)
    row = out["matmul"][key]
    assert abs(row["t_ms"] - 1.0) < 1e-9      # fwd median, reruns excluded
    assert row["n"] == 2
    assert abs(out["meta"]["recompute_factor"] - 0.8) < 1e-3


def test_no_meta_factor_from_bw_only_mix():
    """bw_ ops legitimately reuse one key across fwd and bwd (e.g. rope);
    without a compute-table mixed key no recompute_factor is emitted and
    the bandwidth efficiency uses the fwd instances only."""
    import simumax_amd.kernels.insitu as I

    FakeEvt = _fake_records(I)
    I._RECORDS.clear()
    byt = str(4 * 1024**3)
    I._RECORDS[("bw_rope", byt)] = [
        (FakeEvt(0.0), FakeEvt(1.0 + 0.004), False),   # fwd: eff 0.5
        (FakeEvt(0.0), FakeEvt(2.0 + 0.004), True),    # bwd: slower
    ]
    out = _summarize_nosync(I)
    assert "meta" not in out
    assert abs(out["bandwidth"]["rope_eff"] - 0.5) < 1e-3


def test_fp8_cache_bytes_not_in_param_count():
    """fp8 weight-quant caches count toward peak memory but must not
    inflate the optimizer/DP param-count derivations (which read
    dense/moe_weight_bytes / 2 as numel)."""
    import copy

    from simumax_amd import (ModelConfig, PerfLLM, StrategyConfig,
                             SystemConfig, get_simu_model_config,
                             get_simu_system_config)

    mc = ModelConfig.init_from_config_file(get_simu_model_config("llama3-8b"))
    res = {}
    for fp8 in (False, True):
        st = StrategyConfig(
            seq_len=2048, micro_batch_size=1, micro_batch_num=1,
            world_size=1, tp_size=1, pp_size=1, fp8=fp8,
            enable_recompute=False, enable_sequence_parallel=False,
            zero_state=0, use_fp32_accum_grad=True,
            cross_entropy_loss_fusion=True, attention_sparse_ratio=0.5,
            mem_factor=1.0)
        sysc = SystemConfig.init_from_config_file(
            get_simu_system_config("mi355x"))
        p = PerfLLM()
        p.configure(st, copy.deepcopy(mc), sysc)
        p.run_estimate()
        mi = p.chunks[0].get_model_info()
        res[fp8] = (p.analysis_cost()["optim_time"], mi.dense_weight_bytes,
                    mi.cache_bytes, mi.weight_bytes)
    # optimizer traffic identical; caches only in cache_bytes/weight_bytes
    assert abs(res[True][0] - res[False][0]) < 1e-6
    assert res[True][1] == res[False][1]
    assert res[True][2] > 0 and res[False][2] == 0
    assert res[True][3] > res[False][3]


def test_network_overlay_rescales_tier_efficiency():
    """A measured RCCL all-reduce wall time rescales the xGMI tier's
    per-op efficiency so the model reproduces the measurement."""
    from simumax_amd import SystemConfig, get_simu_system_config
    from simumax_amd.calib.insitu_overlay import apply_insitu_overlay

    sysc = SystemConfig.init_from_config_file(
        get_simu_system_config("mi355x"))
    B, n = 160.0 * 1024**2, 8
    t_pred = sysc.compute_net_op_time("all_reduce", B, n,
                                      net="high_intra_node")
    # pretend the real collective is 1.5x slower than the spec guess
    meas = t_pred * 1.5
    applied = apply_insitu_overlay(sysc, {"network": {"all_reduce": {
        "bytes": B, "comm_num": n, "ms": meas,
        "net": "high_intra_node"}}})
    assert applied == 1
    t_new = sysc.compute_net_op_time("all_reduce", B, n,
                                     net="high_intra_node")
    # the affine solve reproduces the measurement exactly (to rounding)
    assert abs(t_new - meas) / meas < 0.01, (t_new, meas)
