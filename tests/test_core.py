"""Unit tests for L0/L1: TensorSize, configs, cost primitives."""

import math

import pytest

from simumax_amd.core.config import ModelConfig, StrategyConfig
from simumax_amd.core.tensor import TensorSize
from simumax_amd.core.utils import HumanReadableSize, stage_layers


def test_tensor_size_basic():
    t = TensorSize([2, 4096, 4096], "bf16")
    assert t.numel() == 2 * 4096 * 4096
    assert t.mem_bytes() == t.numel() * 2
    assert t.view(2, -1).shape == (2, 4096 * 4096)
    assert t.transpose(1, 2).shape == (2, 4096, 4096)
    assert t.to("fp32").mem_bytes() == t.numel() * 4
    a, b = t.chunk(2, dim=1)
    assert a.shape == (2, 2048, 4096)


def test_strategy_derived():
    s = StrategyConfig(seq_len=4096, micro_batch_size=1, micro_batch_num=8,
                       world_size=8, tp_size=2, pp_size=2)
    assert s.dp_size == 2
    assert s.global_batch_size == 16
    assert s.edp_size == 4
    s.sanity_check()


def test_strategy_format_string():
    s = StrategyConfig.init_from_format_strings("seq4096.mbs1.mbc4.gbs32 tp2.pp2 world_size:8")
    assert s.tp_size == 2 and s.pp_size == 2
    assert s.global_batch_size == 32


def test_model_param_count(llama3_8b):
    # Llama-3 8B has 8.03B params
    assert abs(llama3_8b.param_numel / 1e9 - 8.03) < 0.01


def test_vocab_padding(llama3_8b):
    llama3_8b.maybe_pad_vocab_size(tp_size=8)
    assert llama3_8b.vocab_size % (128 * 8) == 0
    assert llama3_8b.vocab_size >= llama3_8b.orig_vocab_size


def test_flops_per_token(llama3_8b):
    # 6ND lower bound: 6 * 8.03e9 = 48.2 GFLOP/token; attention adds more
    f = llama3_8b.flops_per_token(4096)
    assert 6 * 8.0e9 < f < 6 * 8.0e9 * 1.6


def test_compute_op_time_table(mi355x_system):
    sysc = mi355x_system
    flops = 2 * 4096 * 4096 * 4096
    t_default = sysc.compute_op_accuracy_time("matmul", flops, "nonexistent-shape")
    # inject an accurate factor and confirm exact-key lookup wins
    sysc.accelerator.op["matmul"].accurate_efficient_factor = {"k1": 1.0}
    t_hit = sysc.compute_op_accuracy_time("matmul", flops, "k1")
    assert t_hit < t_default
    assert t_hit == pytest.approx(flops / (2500e12 * 1.0) * 1e3)


def test_fc8_network_scaling(mi355x_system):
    """xGMI FC8: an n=2 collective uses 1 of 7 links, n=8 all 7."""
    sysc = mi355x_system
    size = 1 << 30
    t2 = sysc.compute_net_op_time("all_gather", size, 2, net="high_intra_node",
                                  comm_stage="tp")
    t8 = sysc.compute_net_op_time("all_gather", size, 8, net="high_intra_node",
                                  comm_stage="tp")
    # bandwidth ratio 7x, payload ratio (1-1/2)/(1-1/8) = 4/7 → t2/t8 ≈ 4x
    assert 2.5 < t2 / t8 < 5.0


def test_net_zero_for_single_rank(mi355x_system):
    assert mi355x_system.compute_net_op_time(
        "all_reduce", 1 << 20, 1, net="high_intra_node", comm_stage="tp") == 0.0


def test_mem_access_time(mi355x_system):
    one_gib = 1 << 30
    cfg = mi355x_system.accelerator.bandwidth["default"]
    t = mi355x_system.compute_mem_access_time("default", one_gib)
    assert t == pytest.approx(
        one_gib / (cfg.gbps * 1024**3 * cfg.efficient_factor) * 1e3
        + cfg.latency_us / 1e3)


def test_stage_layers_uneven():
    s = StrategyConfig(seq_len=4096, micro_batch_size=1, micro_batch_num=4,
                       world_size=8, pp_size=4,
                       num_layers_in_first_pipeline_stage=6,
                       num_layers_in_last_pipeline_stage=6)
    m = ModelConfig(hidden_size=512, head_num=8, kv_head_num=8, layer_num=32,
                    vocab_size=32000, intermediate_size=1376, use_swiglu=True)
    assert stage_layers(s, m) == [6, 10, 10, 6]


def test_human_readable_roundtrip():
    s = HumanReadableSize.format_bytes(66.44 * 1024**3)
    assert s == "66.44 GB"
    assert HumanReadableSize.from_string(s) == pytest.approx(66.44 * 1024**3)


def test_rccl_fit_bw_latency():
    """alpha-beta fit recovers synthetic bw/latency (CPU-only)."""
    from simumax_amd.calib.rccl_sweep import fit_bw_latency

    bw_true = 900 * 1024**3          # bytes/s
    lat_true_ms = 0.02
    n, scale, offset = 8, 2, -1
    rows = []
    for size in [2**i for i in range(24, 33)]:
        actual = size * scale + size * scale / n * offset
        t_ms = actual / bw_true * 1e3 + lat_true_ms
        rows.append((size, t_ms))
    bw, lat = fit_bw_latency(rows, scale, offset, n)
    assert bw == pytest.approx(900, rel=0.01)
    assert lat == pytest.approx(lat_true_ms, rel=0.05)


def test_sanity_check_rejects_bad_configs():
    """Invalid parallel layouts must fail loudly at config time, not as
    a shape error deep inside estimate()."""
    import pytest

    # world not divisible by tp*pp
    s = StrategyConfig(seq_len=4096, micro_batch_size=1, micro_batch_num=1,
                       world_size=8, tp_size=3, pp_size=1)
    with pytest.raises((AssertionError, ValueError)):
        s.sanity_check()
    # unknown dtype
    s = StrategyConfig(seq_len=4096, micro_batch_size=1, micro_batch_num=1,
                       world_size=8, tp_size=1, pp_size=1, dtype="int7")
    with pytest.raises((AssertionError, ValueError)):
        s.sanity_check()
    # SP sequence shard must divide seq_len
    s = StrategyConfig(seq_len=4098, micro_batch_size=1, micro_batch_num=1,
                       world_size=8, tp_size=4, pp_size=1,
                       enable_sequence_parallel=True)
    with pytest.raises((AssertionError, ValueError)):
        s.sanity_check()
    # ep cannot exceed world/(tp*pp)
    s = StrategyConfig(seq_len=4096, micro_batch_size=1, micro_batch_num=1,
                       world_size=4, tp_size=2, pp_size=2, ep_size=4)
    with pytest.raises((AssertionError, ValueError, ZeroDivisionError)):
        s.sanity_check()
