"""Attribute the MoE timing under-prediction: wall time vs summed device
kernel time per step for the two MoE validation cases. The difference
(device-idle: launch gaps in the routing chain + per-expert GEMM loops)
is what the simulator's moe_routing latency terms must cover.

Prints per-case: wall ms, busy ms (union of kernel intervals), idle ms,
idle per MoE layer-microbatch; writes gpurun_out/moe_idle.json."""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from torch.profiler import ProfilerActivity, profile

from simumax_amd import ModelConfig, get_simu_model_config
from simumax_amd.train.trainer import (TrainConfig, build_trainer,
                                       make_synthetic_batch, train_step)

CASES = [
    ("mixtral-8x7b-l8", 8, 8),      # (model, moe_layers, local experts)
    ("deepseekv2-l4", 4, 162),      # 160 routed + 2 shared-equivalent
]


def busy_ms(prof):
    """Union length of device kernel intervals (overlap-safe)."""
    spans = []
    for ev in prof.events():
        if ev.device_type.name == "CUDA" and ev.time_range is not None:
            spans.append((ev.time_range.start, ev.time_range.end))
    if not spans:
        # fall back: kernel list from key averages
        return sum(ev.self_device_time_total
                   for ev in prof.key_averages()) / 1e3
    spans.sort()
    total = 0
    cs, ce = spans[0]
    for s, e in spans[1:]:
        if s > ce:
            total += ce - cs
            cs, ce = s, e
        else:
            ce = max(ce, e)
    total += ce - cs
    return total / 1e3


def main():
    out = {}
    mbc = 2
    for name, n_moe, n_exp in CASES:
        mc = ModelConfig.init_from_config_file(get_simu_model_config(name))
        tc = TrainConfig(seq_len=4096, micro_batch_size=1, micro_batch_num=mbc)
        m, opt, red = build_trainer(mc, tc, "cuda:0")
        toks, labels = make_synthetic_batch(mc.vocab_size, mbc, 1, 4096,
                                            "cuda:0")
        for _ in range(2):
            train_step(m, opt, red, toks, labels, mbc)
        torch.cuda.synchronize()
        import time
        t0 = time.time()
        train_step(m, opt, red, toks, labels, mbc)
        torch.cuda.synchronize()
        wall = (time.time() - t0) * 1e3
        with profile(activities=[ProfilerActivity.CUDA]) as prof:
            train_step(m, opt, red, toks, labels, mbc)
            torch.cuda.synchronize()
        busy = busy_ms(prof)
        idle = wall - busy
        per_lmb = idle / (n_moe * mbc)
        row = dict(model=name, wall_ms=round(wall, 2), busy_ms=round(busy, 2),
                   idle_ms=round(idle, 2), moe_layers=n_moe, mbc=mbc,
                   local_experts=n_exp,
                   idle_per_layer_mb_ms=round(per_lmb, 3))
        print(row, flush=True)
        out[name] = row
        red.remove_hooks()   # C++-side hook storage pins the whole trainer
        del m, opt, red
        import gc
        gc.collect()
        torch.cuda.empty_cache()
    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/moe_idle.json", "w") as f:
        json.dump(out, f, indent=1)


if __name__ == "__main__":
    main()
