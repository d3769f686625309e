"""CPU tests for the MLA (DeepSeek-style) trainer path: low-rank q/kv,
RoPE on positional sub-dims only, asymmetric flash SDP, shared-expert
MoE — the real-run side of the deepseekv2 validation case."""

import torch

from simumax_amd.core.config import ModelConfig
from simumax_amd.train.trainer import (TrainConfig, build_trainer,
                                       make_synthetic_batch, train_step)


def tiny_mla_cfg():
    return ModelConfig(
        model_type="moe", model_name="tiny_mla", attention_type="mla",
        hidden_size=256, head_num=8, kv_head_num=8, head_size=32,
        intermediate_size=512, moe_ffn_hidden_size=128,
        moe_shared_expert_intermediate_size=96,
        layer_num=2, dense_layers=1, expert_num=4,
        v_head_dim=32, qk_head_dim=32, qk_pos_emb_head_dim=16,
        q_lora_rank=96, kv_lora_rank=64, topk=2, vocab_size=512,
        use_swiglu=True)


def test_mla_moe_train_step_cpu():
    cfg = tiny_mla_cfg()
    tc = TrainConfig(seq_len=64, micro_batch_size=1, micro_batch_num=2)
    m, opt, red = build_trainer(cfg, tc, "cpu")
    toks, labels = make_synthetic_batch(cfg.vocab_size, 2, 1, 64, "cpu")
    l1 = train_step(m, opt, red, toks, labels, 2)
    l2 = train_step(m, opt, red, toks, labels, 2)
    assert torch.isfinite(torch.tensor([l1, l2])).all()
    red.remove_hooks()


def test_mla_attention_grads_flow():
    cfg = tiny_mla_cfg()
    from simumax_amd.train.model import MLAAttention

    att = MLAAttention(cfg, dtype=torch.float32)
    from simumax_amd.kernels.ops import build_rope_cache

    cs = build_rope_cache(64, cfg.qk_pos_emb_head_dim)
    pos = torch.arange(64, dtype=torch.int32)
    x = torch.randn(1, 64, 256, requires_grad=True)
    y = att(x, cs, pos)
    assert y.shape == (1, 64, 256)
    y.sum().backward()
    assert x.grad is not None and torch.isfinite(x.grad).all()


def test_deepseek_predicts_on_one_gpu():
    """The deepseekv2-l4 validation-case prediction must run (ep1, tp1)
    and fit the 288 GB budget."""
    import copy

    from simumax_amd import (PerfLLM, StrategyConfig, SystemConfig,
                             get_simu_model_config, get_simu_system_config)

    mc = ModelConfig.init_from_config_file(
        get_simu_model_config("deepseekv2-l4"))
    st = StrategyConfig(
        seq_len=4096, micro_batch_size=1, micro_batch_num=2,
        world_size=1, tp_size=1, pp_size=1, ep_size=1,
        enable_sequence_parallel=False, zero_state=0,
        use_fp32_accum_grad=True, enable_recompute=False,
        cross_entropy_loss_fusion=True, attention_sparse_ratio=0.5,
        mem_factor=1.0)
    p = PerfLLM()
    p.configure(st, copy.deepcopy(mc),
                SystemConfig.init_from_config_file(
                    get_simu_system_config("mi355x")))
    p.run_estimate()
    cost = p.analysis_cost()
    mem = p.analysis_mem()
    assert cost["iter_time"] > 0
    assert mem["max_peak_mem"] < 288 * 1024**3, (
        f"{mem['max_peak_mem']/2**30:.1f} GiB")


def test_mla_lite_train_step_cpu():
    """DeepSeek-V2-Lite style MLA: q_lora_rank=0 -> direct q projection
    (trainer q_proj branch, matching the simulator's)."""
    cfg = tiny_mla_cfg()
    cfg.q_lora_rank = 0
    tc = TrainConfig(seq_len=64, micro_batch_size=1, micro_batch_num=2)
    m, opt, red = build_trainer(cfg, tc, "cpu")
    assert hasattr(m.layers[1].attention, "q_proj")
    assert not hasattr(m.layers[1].attention, "q_down")
    toks, labels = make_synthetic_batch(cfg.vocab_size, 2, 1, 64, "cpu")
    l1 = train_step(m, opt, red, toks, labels, 2)
    l2 = train_step(m, opt, red, toks, labels, 2)
    assert l1 == l1 and l2 == l2


def test_deepseekv2_lite_simulator_builds():
    """The shipped deepseekv2-lite config (q_lora_rank=0) runs through
    PerfLLM end to end."""
    from simumax_amd import (PerfLLM, StrategyConfig, SystemConfig,
                             get_simu_model_config, get_simu_system_config)

    st = StrategyConfig(seq_len=4096, micro_batch_size=1, micro_batch_num=1,
                        world_size=1, tp_size=1, pp_size=1,
                        enable_sequence_parallel=False, zero_state=0,
                        use_fp32_accum_grad=True, enable_recompute=False,
                        cross_entropy_loss_fusion=True,
                        attention_sparse_ratio=0.5, mem_factor=1.0)
    p = PerfLLM()
    p.configure(st, ModelConfig.init_from_config_file(
        get_simu_model_config("deepseekv2-lite")),
        SystemConfig.init_from_config_file(get_simu_system_config("mi355x")))
    p.run_estimate()
    cost = p.analysis_cost()
    mem = p.analysis_mem()
    assert cost["iter_time"] > 0 and not mem["oom"]
    # direct q projection: no q_down leaf in the block
    names = [l.full_name for l in p.chunks[0].blocks[1].leaf_modules()]
    assert not any("q_down" in n for n in names)
    assert any("q_proj" in n for n in names)
