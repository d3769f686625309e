"""Perf-vs-real validation pipeline with phased state + resume.

Parity target: tools/b200/run_megatron_perf_real_pipeline.py (1302 LoC in
the reference): perf_screen (simulator prediction) -> real run (the
in-repo Megatron-ROCm-style trainer standing in for Megatron-LM) ->
summarize (rel_err / alloc_err tables). Each phase writes into a state
JSON so a partial run resumes where it stopped (--reuse-state parity).

Run ON a GPU box:
    python -m simumax_amd.calib.perf_real_pipeline --phase all
    python -m simumax_amd.calib.perf_real_pipeline --phase summarize
"""

from __future__ import annotations

import argparse
import copy
import json
import os
import sys

DEFAULT_CASES = [
    dict(case="llama3_8b_dp1_mbc4", model="llama3-8b", seq=4096, mbs=1,
         mbc=4, steps=3),
    dict(case="llama3_8b_dp1_mbc8", model="llama3-8b", seq=4096, mbs=1,
         mbc=8, steps=2),
    dict(case="llama3_8b_seq2048_mbs2", model="llama3-8b", seq=2048, mbs=2,
         mbc=4, steps=3),
    dict(case="llama3_8b_seq8192", model="llama3-8b", seq=8192, mbs=1,
         mbc=2, steps=3),
    dict(case="llama3_70b_l12", model="llama3-70b-l12", seq=4096, mbs=1,
         mbc=2, steps=3),
]


def _load_state(path):
    if os.path.exists(path):
        with open(path) as f:
            return json.load(f)
    return {"cases": {}}


def _save_state(path, state):
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    with open(path, "w") as f:
        json.dump(state, f, indent=1)


def perf_screen(case: dict) -> dict:
    """Simulator prediction for one case (CPU-only)."""
    from simumax_amd import (ModelConfig, PerfLLM, StrategyConfig,
                             SystemConfig, get_simu_model_config,
                             get_simu_system_config)

    mc = ModelConfig.init_from_config_file(
        get_simu_model_config(case["model"]))
    st = StrategyConfig(
        seq_len=case["seq"], micro_batch_size=case["mbs"],
        micro_batch_num=case["mbc"], world_size=case.get("world", 1),
        tp_size=1, pp_size=1, enable_sequence_parallel=False, zero_state=0,
        use_fp32_accum_grad=True, enable_recompute=False,
        cross_entropy_loss_fusion=True, attention_sparse_ratio=0.5,
        mem_factor=1.0)
    p = PerfLLM()
    p.configure(st, copy.deepcopy(mc), SystemConfig.init_from_config_file(
        get_simu_system_config("mi355x")))
    p.run_estimate()
    cost = p.analysis_cost()
    mem = p.analysis_mem()
    return dict(predicted_ms=cost["iter_time"], predicted_mfu=cost["mfu"],
                predicted_bytes=mem["max_peak_mem"])


def real_run(case: dict) -> dict:
    """One real measurement (requires a GPU)."""
    import time

    import torch

    from simumax_amd import ModelConfig, get_simu_model_config
    from simumax_amd.train.trainer import (TrainConfig, build_trainer,
                                           make_synthetic_batch, train_step)

    mc = ModelConfig.init_from_config_file(
        get_simu_model_config(case["model"]))
    tc = TrainConfig(seq_len=case["seq"], micro_batch_size=case["mbs"],
                     micro_batch_num=case["mbc"])
    model, opt, red = build_trainer(mc, tc, "cuda:0")
    toks, labels = make_synthetic_batch(mc.vocab_size, case["mbc"],
                                        case["mbs"], case["seq"], "cuda:0")
    train_step(model, opt, red, toks, labels, case["mbc"])  # warmup
    torch.cuda.reset_peak_memory_stats()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(case["steps"]):
        train_step(model, opt, red, toks, labels, case["mbc"])
    torch.cuda.synchronize()
    ms = (time.time() - t0) / case["steps"] * 1e3
    out = dict(measured_ms=ms,
               measured_bytes=float(torch.cuda.max_memory_allocated()))
    from simumax_amd.kernels.ops import clear_dummy_wgrads

    red.remove_hooks()
    clear_dummy_wgrads()
    del model, opt, red, toks, labels
    import gc

    gc.collect()
    torch.cuda.empty_cache()
    return out


def summarize(state: dict) -> dict:
    rows = []
    for name, c in state["cases"].items():
        if "predicted_ms" not in c or "measured_ms" not in c:
            continue
        rows.append(dict(
            case=name,
            rel_err=round((c["predicted_ms"] - c["measured_ms"])
                          / c["measured_ms"] * 100, 2),
            alloc_err=round((c["predicted_bytes"] - c["measured_bytes"])
                            / c["measured_bytes"] * 100, 2),
            measured_ms=round(c["measured_ms"], 2),
            predicted_ms=round(c["predicted_ms"], 2),
            measured_gib=round(c["measured_bytes"] / 2**30, 2),
            predicted_gib=round(c["predicted_bytes"] / 2**30, 2),
        ))
    summary = {
        "rows": rows,
        "timing_err_range": [min((r["rel_err"] for r in rows), default=None),
                             max((r["rel_err"] for r in rows), default=None)],
        "alloc_err_range": [min((r["alloc_err"] for r in rows), default=None),
                            max((r["alloc_err"] for r in rows), default=None)],
    }
    state["summary"] = summary
    return summary


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--phase", default="all",
                    choices=["perf_screen", "run", "summarize", "all"])
    ap.add_argument("--state", default="gpurun_out/perf_real_state.json")
    ap.add_argument("--case", default=None, help="run a single case")
    args = ap.parse_args()
    sys.path.insert(0, os.getcwd())
    state = _load_state(args.state)
    cases = [c for c in DEFAULT_CASES
             if args.case is None or c["case"] == args.case]

    if args.phase in ("perf_screen", "all"):
        for c in cases:
            entry = state["cases"].setdefault(c["case"], {})
            if "predicted_ms" not in entry:
                entry.update(perf_screen(c))
                _save_state(args.state, state)
                print(f"[perf_screen] {c['case']}: "
                      f"{entry['predicted_ms']:.1f} ms", flush=True)
    if args.phase in ("run", "all"):
        for c in cases:
            entry = state["cases"].setdefault(c["case"], {})
            if "measured_ms" not in entry:
                try:
                    entry.update(real_run(c))
                except Exception as e:  # OOM etc: record and continue
                    entry["error"] = str(e)[:200]
                _save_state(args.state, state)
                print(f"[run] {c['case']}: "
                      f"{entry.get('measured_ms', 'ERR')}", flush=True)
    if args.phase in ("summarize", "all"):
        summary = summarize(state)
        _save_state(args.state, state)
        print(json.dumps(summary, indent=1))


if __name__ == "__main__":
    main()
