"""Hand-computed unit tests for the autograd-glue cost terms added on
top of the roofline (kernel-level copies/adds the trainer really runs:
qkv split copies, residual grad fan-in, main-grad hook cast+add, MoE
dispatch/combine traffic, routing-chain host latency)."""

import pytest

from simumax_amd.core.config import ModelConfig, StrategyConfig, SystemConfig
from simumax_amd import (PerfLLM, get_simu_model_config,
                         get_simu_system_config)


def build(model="llama3-8b", **over):
    mc = ModelConfig.init_from_config_file(get_simu_model_config(model))
    st = StrategyConfig(
        seq_len=4096, micro_batch_size=1, micro_batch_num=1,
        world_size=1, tp_size=1, pp_size=1, ep_size=1,
        enable_sequence_parallel=False, zero_state=0,
        use_fp32_accum_grad=True, enable_recompute=False,
        cross_entropy_loss_fusion=True, attention_sparse_ratio=0.5,
        mem_factor=1.0)
    for k, v in over.items():
        setattr(st, k, v)
    p = PerfLLM()
    p.configure(st, mc, SystemConfig.init_from_config_file(
        get_simu_system_config("mi355x")))
    p.run_estimate()
    return p


def leaf(p, cls, part=""):
    for l in p.chunks[0].leaf_modules():
        if type(l).__name__ == cls and part in l.full_name:
            return l
    raise AssertionError(f"{cls} {part}")


@pytest.fixture(scope="module")
def p8b():
    return build()


def test_residual_add_bwd_glue(p8b):
    """Residual fan-in: backward accumulates two grad paths with one
    elementwise add = 3 passes over the [1,4096,4096] bf16 tensor."""
    add = leaf(p8b, "Add", "attn_residual")
    ci = add.get_compute_info()
    assert ci.bwd_grad_act_extra_mem == 3 * 4096 * 4096 * 2
    # the add kernel's time shows up additively in bwd
    assert add.get_cost_info().bwd_grad_act_time > 0


def test_qkv_split_copy_glue(p8b):
    """Dense GQA: q/k/v are materialized from the fused qkv output
    (fwd) and dq/dk/dv cat back (bwd): 2x(q+k+v) bytes each way."""
    core = leaf(p8b, "CoreAttention")
    ci = core.get_compute_info()
    q = 4096 * 32 * 128 * 2
    kv = 4096 * 8 * 128 * 2
    assert ci.fwd_extra_mem == 2 * (q + 2 * kv)
    assert ci.bwd_grad_act_extra_mem == 2 * (q + 2 * kv)


def test_embedding_hook_glue(p8b):
    """Non-fused params accumulate via the hook: p.grad.float() temp
    (r2+w4) + main_grad.add_ (r4+r4+w4) = 18 B/elem."""
    emb = leaf(p8b, "Embedding")
    ci = emb.get_compute_info()
    # vocab is padded to a multiple of 128
    v = p8b.model_config.vocab_size
    assert ci.bwd_grad_w_extra_mem == v * 4096 * 18


def test_mla_cat_glue():
    p = build("deepseekv2-l4")
    core = leaf(p, "MLACoreAttention")
    ci = core.get_compute_info()
    q = 4096 * 128 * 192 * 2
    v = 4096 * 128 * 128 * 2
    assert ci.fwd_extra_mem == 2 * (q + q) + 2 * v
    assert ci.bwd_grad_act_extra_mem == 2 * (q + q)


def test_moe_routing_latency_priced():
    """Router's extra term is priced through bandwidth['moe_routing']
    whose latency_us is the measured host-bound chain cost."""
    p = build("mixtral-8x7b-l8")
    router = leaf(p, "Router")
    sysc = p.system
    bwf = sysc.accelerator.bandwidth["moe_routing"]
    bwb = sysc.accelerator.bandwidth["moe_routing_bwd"]
    # base + per-local-expert launch chain (measured: idle/layer-mb
    # 0.285 ms at E=8 vs 3.54 ms at E=162, scripts/moe_idle_probe.py)
    lat_f = bwf.latency_us + bwf.per_unit_us * router.extra_op_units
    lat_b = bwb.latency_us + bwb.per_unit_us * router.extra_op_units
    assert lat_f > 100
    assert 0 < lat_b < lat_f
    ci = router.get_cost_info()
    # fwd time must include at least the routing latency
    assert ci.fwd_compute_time * 1e3 >= lat_f


def test_permutation_traffic_formulas():
    """Dispatch ~5 passes over the expanded buffer; combine ~4 in + 3
    out passes (train/moe.py op sequence)."""
    p = build("mixtral-8x7b-l8")
    perm = leaf(p, "Permutation")
    ci = perm.get_compute_info()
    tokens, topk, h = 4096, 2, 4096
    in_b = tokens * h * 2
    out_b = tokens * topk * h * 2  # cap-padded expanded buffer (cap=1)
    assert ci.fwd_accessed_mem == 2 * out_b + 3 * max(in_b * topk, out_b)
    assert ci.bwd_grad_act_accessed_mem == 2 * out_b + 3 * in_b
    unperm = leaf(p, "UnPermutation")
    cu = unperm.get_compute_info()
    assert cu.fwd_accessed_mem == 4 * out_b + 3 * in_b
    assert cu.bwd_grad_act_accessed_mem == 6 * out_b + 2 * in_b
