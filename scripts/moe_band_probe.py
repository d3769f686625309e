"""Band-level attribution of the mixtral MoE layer: record_function
labels around each sub-band (attention, norms, moe_mlp) during a
profiled step; compare per-band CUDA totals with the simulator's
per-leaf predictions to locate the residual timing under-prediction.

Labels wrap module calls via forward hooks; backward attribution uses
the profiler's bwd correlation (PyTorch names backward regions after
the autograd nodes of ops inside the labelled region).
"""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from torch.profiler import ProfilerActivity, profile, record_function

from simumax_amd import ModelConfig, get_simu_model_config
from simumax_amd.train.trainer import (TrainConfig, build_trainer,
                                       make_synthetic_batch, train_step)


def main():
    mc = ModelConfig.init_from_config_file(
        get_simu_model_config("mixtral-8x7b-l8"))
    tc = TrainConfig(seq_len=4096, micro_batch_size=1, micro_batch_num=2)
    m, opt, red = build_trainer(mc, tc, "cuda:0")
    toks, labels = make_synthetic_batch(mc.vocab_size, 2, 1, 4096, "cuda:0")

    # wrap each band with record_function via pre/post hooks
    handles = []

    def wrap(mod, name):
        state = {}

        def pre(m_, inp):
            state["rf"] = record_function(name)
            state["rf"].__enter__()

        def post(m_, inp, out):
            state["rf"].__exit__(None, None, None)

        handles.append(mod.register_forward_pre_hook(pre))
        handles.append(mod.register_forward_hook(post))

    for layer in m.layers:
        wrap(layer.attention, "band_attention")
        wrap(layer.input_norm, "band_norm")
        wrap(layer.pre_mlp_norm, "band_norm")
        wrap(layer.moe_mlp, "band_moe")
    for _ in range(2):
        train_step(m, opt, red, toks, labels, 2)
    torch.cuda.synchronize()
    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                 with_stack=False) as prof:
        train_step(m, opt, red, toks, labels, 2)
        torch.cuda.synchronize()
    rows = {}
    for ev in prof.key_averages():
        if ev.key.startswith("band_") or ev.key in ("aten::mm",):
            rows[ev.key] = dict(cuda_ms=round(ev.device_time_total / 1e3, 2),
                                count=ev.count)
    print(json.dumps(rows, indent=1))
    with open("gpurun_out/moe_bands.json", "w") as f:
        json.dump(rows, f, indent=1)


if __name__ == "__main__":
    main()
