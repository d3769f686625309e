"""Predicted-vs-measured validation sweep (1 GPU): several Llama configs
in one process. Writes JSON lines to gpurun_out/validation.jsonl.

The in-repo analog of the reference's perf-vs-real pipeline
(tools/b200/run_megatron_perf_real_pipeline.py) with the in-repo trainer
standing in for Megatron-LM.
"""
import gc
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from simumax_amd import ModelConfig, get_simu_model_config
from simumax_amd.train.trainer import (TrainConfig, build_trainer,
                                       make_synthetic_batch, train_step)

CASES = [
    # (name, model, seq, mbs, mbc, steps)
    ("8b_seq4096_mbc4", "llama3-8b", 4096, 1, 4, 3),
    ("8b_seq4096_mbc8", "llama3-8b", 4096, 1, 8, 2),
    ("8b_seq2048_mbs2", "llama3-8b", 2048, 2, 4, 3),
    ("8b_seq8192_mbc2", "llama3-8b", 8192, 1, 2, 3),
    ("70b_l12_seq4096", "llama3-70b-l12", 4096, 1, 2, 3),
    ("mixtral_l8_moe", "mixtral-8x7b-l8", 4096, 1, 2, 3),
    ("8b_seq16384_mbc1", "llama3-8b", 16384, 1, 1, 3),
    ("qwen32b_l12_seq4096", "qwen3-32b-l12", 4096, 1, 2, 3),
    ("deepseekv2_l4_mla_moe", "deepseekv2-l4", 4096, 1, 2, 3),
    ("8b_fp8_seq4096_mbc4", "llama3-8b", 4096, 1, 4, 3, True),
    # full-block activation recompute (trainer torch.utils.checkpoint vs
    # simulator full_block/recompute_layer_num)
    ("70b_l12_rc_seq8192", "llama3-70b-l12", 8192, 1, 2, 3, False, 12),
    # 32k long context needs full recompute to fit (act ~80 GiB without)
    ("8b_seq32768_rc_mbc1", "llama3-8b", 32768, 1, 1, 2, False, 32),
    ("llama2_7b_seq4096_mbc4", "llama2-7b", 4096, 1, 4, 3),
]

OUT = "gpurun_out/validation.jsonl"


def predict(model_cfg, seq, mbs, mbc, fp8=False, rc=0, overlay=None):
    import copy

    from simumax_amd import (PerfLLM, StrategyConfig, SystemConfig,
                             get_simu_system_config)

    st = StrategyConfig(
        seq_len=seq, micro_batch_size=mbs, micro_batch_num=mbc,
        world_size=1, tp_size=1, pp_size=1, fp8=fp8,
        enable_recompute=rc > 0,
        recompute_granularity="full_block" if rc else None,
        recompute_layer_num=rc,
        enable_sequence_parallel=False, zero_state=0,
        use_fp32_accum_grad=True,
        cross_entropy_loss_fusion=True, attention_sparse_ratio=0.5,
        mem_factor=1.0)
    sysc = SystemConfig.init_from_config_file(
        get_simu_system_config("mi355x"))
    if overlay is not None:
        from simumax_amd.calib.insitu_overlay import apply_insitu_overlay

        apply_insitu_overlay(sysc, overlay)
    p = PerfLLM()
    p.configure(st, copy.deepcopy(model_cfg), sysc)
    p.run_estimate()
    return p.analysis_cost(), p.analysis_mem()


def run_case(name, model, seq, mbs, mbc, steps, fp8=False, rc=0, warmup=1):
    mc = ModelConfig.init_from_config_file(get_simu_model_config(model))
    tc = TrainConfig(seq_len=seq, micro_batch_size=mbs, micro_batch_num=mbc,
                     fp8=fp8, recompute_layers=rc)
    t0 = time.time()
    m, opt, red = build_trainer(mc, tc, "cuda:0")
    try:
        toks, labels = make_synthetic_batch(mc.vocab_size, mbc, mbs, seq,
                                            "cuda:0")
        for _ in range(warmup):
            train_step(m, opt, red, toks, labels, mbc)
        # one untimed in-situ step: THIS box's per-shape efficiencies feed
        # the prediction (same-machine calibrate-then-validate per case).
        # Recompute rows participate too: insitu tags instances recorded
        # inside the backward graph task, so checkpoint reruns cannot
        # shift the fwd keys (recompute_factor prices the rerun).
        from simumax_amd.core.consts import \
            OPTIMIZER_TRAFFIC_BYTES_PER_PARAM
        from simumax_amd.kernels import insitu

        insitu.enable()
        train_step(m, opt, red, toks, labels, mbc)
        torch.cuda.synchronize()
        insitu.disable()
        overlay = insitu.summarize()
        s_ev = torch.cuda.Event(enable_timing=True)
        e_ev = torch.cuda.Event(enable_timing=True)
        s_ev.record()
        opt.step()
        e_ev.record()
        torch.cuda.synchronize()
        overlay.setdefault("bandwidth", {})["optimizer_eff"] = (
            opt.flat_grad.numel() * OPTIMIZER_TRAFFIC_BYTES_PER_PARAM
            / (s_ev.elapsed_time(e_ev) / 1e3) / (8000.0 * 1024**3))
        torch.cuda.reset_peak_memory_stats()
        torch.cuda.synchronize()
        t1 = time.time()
        for _ in range(steps):
            train_step(m, opt, red, toks, labels, mbc)
        torch.cuda.synchronize()
        ms = (time.time() - t1) / steps * 1e3
        peak = torch.cuda.max_memory_allocated()
        cost, mem = predict(mc, seq, mbs, mbc, fp8=fp8, rc=rc,
                            overlay=overlay)
        row = dict(
            case=name, model=model, seq=seq, mbs=mbs, mbc=mbc,
            measured_ms=round(ms, 2),
            predicted_ms=round(cost["iter_time"], 2),
            timing_err_pct=round((cost["iter_time"] - ms) / ms * 100, 2),
            measured_gib=round(peak / 2**30, 2),
            predicted_gib=round(mem["max_peak_mem"] / 2**30, 2),
            mem_err_pct=round((mem["max_peak_mem"] - peak) / peak * 100, 2),
            fp8=fp8, recompute_layers=rc,
            recompute_factor_meas=(overlay or {}).get("meta", {}).get(
                "recompute_factor"),
            measured_mfu=round(
                mc.flops_per_token(seq) * mbs * mbc * seq / (ms / 1e3)
                / 2.5e15, 4),
            predicted_mfu=round(cost["mfu"], 4),
            build_s=round(t1 - t0, 1),
        )
        print(json.dumps(row), flush=True)
        with open(OUT, "a") as f:
            f.write(json.dumps(row) + "\n")
        return row
    finally:
        # free everything EVEN ON FAILURE (the post-accumulate hooks pin
        # the whole trainer from C++ storage; a leaked case once added
        # 45 GiB to the next case's measured peak)
        from simumax_amd.kernels.ops import clear_dummy_wgrads

        red.remove_hooks()
        clear_dummy_wgrads()
        del m, opt, red
        gc.collect()
        torch.cuda.empty_cache()


def main():
    os.makedirs("gpurun_out", exist_ok=True)
    only = sys.argv[1:] or None
    for case in CASES:
        if only and not any(o in case[0] for o in only):
            continue
        try:
            run_case(*case)
        except torch.cuda.OutOfMemoryError:
            print(json.dumps({"case": case[0], "error": "OOM"}), flush=True)
            torch.cuda.empty_cache()
        except Exception as e:  # keep sweeping
            print(json.dumps({"case": case[0], "error": str(e)[:200]}),
                  flush=True)


if __name__ == "__main__":
    main()
