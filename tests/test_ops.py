"""Per-leaf-op formula tests: flops/bytes/comm vs hand-computed values."""

import pytest

from simumax_amd.core.config import StrategyConfig
from simumax_amd.core.records import InputOutputInfo
from simumax_amd.core.tensor import TensorSize
from simumax_amd.ops.dense import (Attention, CoreAttention, LayerNorm,
                                   LinearCol, LinearRow, ParallelCE, Swiglu)


def make_strategy(**kw):
    base = dict(seq_len=4096, micro_batch_size=1, micro_batch_num=1,
                world_size=8, tp_size=1, pp_size=1,
                enable_sequence_parallel=False, enable_recompute=False)
    base.update(kw)
    s = StrategyConfig(**base)
    # resolve nets directly for unit tests
    for a in ("tp_net", "cp_net", "pp_net", "dp_net", "ep_net", "etp_net", "edp_net"):
        setattr(s, a, "high_intra_node")
    return s


def io(*shape, dtype="bf16"):
    return InputOutputInfo([TensorSize(shape, dtype)])


def test_linear_col_flops_bytes(mi355x_system):
    s = make_strategy()
    lin = LinearCol(4096, 8192, s, mi355x_system)
    lin(io(1, 4096, 4096))
    ci = lin.get_compute_info()
    assert ci.fwd_flops == 2 * 4096 * 4096 * 8192
    assert ci.bwd_grad_act_flops == ci.fwd_flops
    assert ci.bwd_grad_w_flops == ci.fwd_flops
    e = 2
    assert ci.fwd_accessed_mem == (4096 * 4096 + 4096 * 8192 + 4096 * 8192) * e
    # activation cache = input
    assert lin.get_act_info().activation_mem_cache == 4096 * 4096 * 2
    # shape keys
    assert lin.get_input_shapes_desc("fwd") == (
        "b=1, m=4096, k=4096, n=8192, layout=TN, accumulate=False, out_dtype=bf16"
    )
    assert "layout=NT, accumulate=True, out_dtype=fp32" in lin.get_input_shapes_desc("bwd_grad_w")


def test_linear_col_sp_comm(mi355x_system):
    s = make_strategy(tp_size=2, enable_sequence_parallel=True)
    lin = LinearCol(4096, 4096, s, mi355x_system)  # local N = 8192/2
    lin(io(1, 2048, 4096))  # sequence-sharded input
    out = lin.output_info.first
    assert out.shape == (1, 4096, 4096)  # seq re-gathered
    kinds = [(ev.stage, ev.op_name) for ev in lin.comm_ops]
    assert ("fwd", "all_gather") in kinds
    assert ("bwd_act", "reduce_scatter") in kinds
    assert ("bwd_w", "all_gather") in kinds
    ag = next(ev for ev in lin.comm_ops if ev.stage == "fwd")
    assert ag.size == 1 * 4096 * 4096 * 2  # full gathered bytes
    # GEMM M uses the gathered seq
    assert "m=4096" in lin.get_input_shapes_desc("fwd")


def test_linear_row_tp_allreduce(mi355x_system):
    s = make_strategy(tp_size=4, enable_sequence_parallel=False)
    lin = LinearRow(2048, 4096, s, mi355x_system)  # local K = 8192/4
    lin(io(1, 4096, 2048))
    assert lin.output_info.first.shape == (1, 4096, 4096)
    evs = [(ev.stage, ev.op_name) for ev in lin.comm_ops]
    assert evs == [("fwd", "all_reduce")]
    assert lin.comm_ops[0].size == 4096 * 4096 * 2


def test_core_attention_flash_flops(mi355x_system):
    s = make_strategy()
    att = CoreAttention(32, 8, 128, 128, s, mi355x_system)
    b, sq = 1, 4096
    q = TensorSize([b, sq, 32 * 128])
    k = TensorSize([b, sq, 8 * 128])
    v = TensorSize([b, sq, 8 * 128])
    att(InputOutputInfo([q, k, v]))
    ci = att.get_compute_info()
    qk = 2 * b * 32 * sq * sq * 128
    assert ci.fwd_flops == pytest.approx(2 * qk)
    assert ci.bwd_grad_act_flops == pytest.approx(5 * qk)  # 4 bmm + flash recompute
    # flash cache: q,k,v + lse (o belongs to out_proj)
    lse = b * sq * 32 * 4
    assert att.get_act_info().activation_mem_cache == (
        q.mem_bytes() + k.mem_bytes() + v.mem_bytes() + lse
    )
    assert att.get_input_shapes_desc("fwd") == (
        "batch=1, seq_len=4096, head_num=32, kv_head_num=8, qk_head_dim=128, "
        "v_head_dim=128, qkv_contiguous=True"
    )


def test_causal_sparse_ratio(mi355x_system):
    s = make_strategy(attention_sparse_ratio=0.5)
    att = CoreAttention(32, 8, 128, 128, s, mi355x_system)
    q = TensorSize([1, 4096, 32 * 128])
    kv = TensorSize([1, 4096, 8 * 128])
    att(InputOutputInfo([q, kv, kv]))
    full = 2 * 2 * 32 * 4096 * 4096 * 128
    assert att.get_compute_info().fwd_flops == pytest.approx(full * 0.5)


def test_swiglu_shapes_and_cache(mi355x_system):
    s = make_strategy()
    sw = Swiglu(s, mi355x_system)
    sw(io(1, 4096, 2 * 14336))
    assert sw.output_info.first.shape == (1, 4096, 14336)
    assert sw.get_act_info().activation_mem_cache == 1 * 4096 * 2 * 14336 * 2


def test_layernorm_cache(mi355x_system):
    s = make_strategy()
    ln = LayerNorm(4096, s, mi355x_system)
    ln(io(1, 4096, 4096))
    # input + fp32 rstd per row
    assert ln.get_act_info().activation_mem_cache == 4096 * 4096 * 2 + 4096 * 4
    assert ln.get_model_info().dense_weight_bytes == 4096 * 2


def test_parallel_ce_comm(mi355x_system):
    s = make_strategy(tp_size=8)
    ce = ParallelCE(s, mi355x_system)
    ce(io(1, 4096, 128256 // 8))
    evs = [ev for ev in ce.comm_ops if ev.op_name == "all_reduce"]
    assert len(evs) == 3  # unfused CE
    assert all(ev.size == 4096 * 4 for ev in evs)


def test_attention_composite_gqa(mi355x_system, llama3_8b):
    s = make_strategy(tp_size=2, enable_sequence_parallel=True)
    att = Attention(llama3_8b, s, mi355x_system)
    att(io(1, 2048, 4096))
    assert att.output_info.first.shape == (1, 2048, 4096)
    # qkv projection local N = (16 + 2*4)*128 = 3072
    assert att.qkv_proj.output_size == (16 + 2 * 4) * 128
    leaves = [type(l).__name__ for l in att.leaf_modules()]
    assert leaves == ["LinearCol", "RotaryEmbedding", "CoreAttention", "LinearRow"]


def test_grad_and_state_sharding(mi355x_system):
    # ZeRO-1: optimizer state divided by dp*cp
    s = make_strategy(world_size=8, tp_size=1, zero_state=1)
    lin = LinearCol(1024, 1024, s, mi355x_system)
    lin(io(1, 128, 1024))
    mi = lin.get_model_info()
    numel = 1024 * 1024
    assert mi.dense_weight_bytes == numel * 2
    assert mi.dense_grad_bytes == numel * 4       # fp32 main grads
    assert mi.dense_state_bytes == pytest.approx(numel * 12 / 8)
