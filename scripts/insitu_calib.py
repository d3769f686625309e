"""In-situ operator calibration over the validation cases.

For each case: build the trainer, warm up, then run 2 steps with the
kernels/insitu event registry enabled — every FusedLinear fwd/dgrad/wgrad,
flash-attention fwd/bwd and grouped-GEMM call is timed in the real
training stream, keyed by the simulator's shape-key strings. Also times
the flat MixedPrecisionAdam step at model scale (the microbench measured
a 512M-element buffer; the real 8–17 B-param flat step reaches higher
effective bandwidth).

Outputs gpurun_out/calib/{matmul,sdp_fwd,sdp_bwd,group_matmul}_insitu.json
and bandwidth_insitu.json, which calib/merge.py overlays with precedence.
"""
import gc
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from simumax_amd import ModelConfig, get_simu_model_config
from simumax_amd.core.consts import OPTIMIZER_TRAFFIC_BYTES_PER_PARAM
from simumax_amd.kernels import insitu
from simumax_amd.kernels.ops import clear_dummy_wgrads
from simumax_amd.train.trainer import (TrainConfig, build_trainer,
                                       make_synthetic_batch, train_step)

CASES = [
    ("llama3-8b", 4096, 1, 2),
    ("llama3-8b", 2048, 2, 2),
    ("llama3-8b", 8192, 1, 2),
    ("llama3-70b-l12", 4096, 1, 2),
    ("mixtral-8x7b-l8", 4096, 1, 2),
    ("llama3-8b", 16384, 1, 1),
    ("qwen3-32b-l12", 4096, 1, 2),
    ("deepseekv2-l4", 4096, 1, 2),
    ("llama3-8b", 4096, 1, 2, True),   # fp8 decoder linears
]

OUTDIR = "gpurun_out/calib"


def measure_routing_chain(device="cuda:0"):
    """Wall-clock the exact top-k routing chain of train/moe.py (softmax,
    topk, argsort, bincount, cumsum, bookkeeping index ops) on the
    mixtral shape: ~50 tiny host-launched kernels whose cost is launch-
    bound, not bandwidth-bound. Returns ms per invocation."""
    import math
    import time

    N, E, topk, cap_f = 4096, 8, 2, 1
    logits = torch.randn(N, E, device=device, dtype=torch.float32)
    torch.cuda.synchronize()
    t0 = time.time()
    ITER = 30
    for _ in range(ITER):
        probs = torch.softmax(logits, dim=-1)
        weight, idx = probs.topk(topk, dim=-1)
        weight = weight / weight.sum(-1, keepdim=True)
        cap = int(math.ceil(N * topk / E * cap_f))
        flat_expert = idx.reshape(-1)
        flat_token = torch.arange(N, device=device).repeat_interleave(topk)
        order = torch.argsort(flat_expert, stable=True)
        counts = torch.bincount(flat_expert, minlength=E)
        offs = torch.cumsum(counts, 0) - counts
        rank_sorted = (torch.arange(N * topk, device=device)
                       - offs[flat_expert[order]])
        keep = rank_sorted < cap
        src_tok = flat_token[order][keep]
        dst_exp = flat_expert[order][keep]
        slot_index = dst_exp * cap + rank_sorted[keep]
        w_kept = weight.reshape(-1)[order][keep].to(torch.bfloat16)
    torch.cuda.synchronize()
    fwd_ms = (time.time() - t0) / ITER * 1e3

    # backward half: grads flow to the router logits through the
    # softmax/topk/normalize + gather chain
    logits_g = logits.clone().requires_grad_(True)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(ITER):
        probs = torch.softmax(logits_g, dim=-1)
        weight, idx = probs.topk(topk, dim=-1)
        weight = weight / weight.sum(-1, keepdim=True)
        flat_expert = idx.reshape(-1)
        order = torch.argsort(flat_expert, stable=True)
        w_kept = weight.reshape(-1)[order]
        w_kept.float().sum().backward()
        logits_g.grad = None
    torch.cuda.synchronize()
    both_ms = (time.time() - t0) / ITER * 1e3
    return fwd_ms, max(both_ms - fwd_ms * 0.6, fwd_ms * 0.5)


def main():
    opt_samples = []
    for case in sys.argv[1:] and [
            c for c in CASES if c[0] in sys.argv[1:]] or CASES:
        model, seq, mbs, mbc = case[:4]
        fp8 = len(case) > 4 and case[4]
        mc = ModelConfig.init_from_config_file(get_simu_model_config(model))
        tc = TrainConfig(seq_len=seq, micro_batch_size=mbs,
                         micro_batch_num=mbc, fp8=fp8)
        m, opt, red = build_trainer(mc, tc, "cuda:0")
        toks, labels = make_synthetic_batch(mc.vocab_size, mbc, mbs, seq,
                                            "cuda:0")
        train_step(m, opt, red, toks, labels, mbc)  # warmup
        torch.cuda.synchronize()
        insitu.enable()
        for _ in range(2):
            train_step(m, opt, red, toks, labels, mbc)
        insitu.disable()
        insitu.dump(OUTDIR)

        # optimizer at model scale: events around zero_grad + step
        n_params = opt.flat_grad.numel()
        s_ev = torch.cuda.Event(enable_timing=True)
        e_ev = torch.cuda.Event(enable_timing=True)
        ts = []
        for _ in range(3):
            s_ev.record()
            opt.step()
            opt.zero_grad()
            e_ev.record()
            torch.cuda.synchronize()
            ts.append(s_ev.elapsed_time(e_ev))
        t = sorted(ts)[1]
        eff = (n_params * OPTIMIZER_TRAFFIC_BYTES_PER_PARAM
               / (t / 1e3) / (8000.0 * 1024**3))
        opt_samples.append(dict(model=model, n_params=n_params,
                                t_ms=round(t, 2), eff=round(eff, 4)))
        print(f"[optimizer] {model}: {n_params/1e9:.2f}B params "
              f"{t:.1f} ms -> eff {eff:.4f}", flush=True)

        red.remove_hooks()
        clear_dummy_wgrads()
        del m, opt, red, toks, labels
        gc.collect()
        torch.cuda.empty_cache()

    # median optimizer efficiency across models
    effs = sorted(r["eff"] for r in opt_samples)
    path = os.path.join(OUTDIR, "bandwidth_insitu.json")
    out = {}
    if os.path.exists(path):
        with open(path) as f:
            out = json.load(f)
    out["optimizer_eff"] = effs[len(effs) // 2]
    out["optimizer_samples"] = opt_samples
    rt_fwd, rt_bwd = measure_routing_chain()
    out["moe_routing_ms"] = round(rt_fwd, 4)
    out["moe_routing_bwd_ms"] = round(rt_bwd, 4)
    print(f"[routing chain] fwd {rt_fwd:.3f} ms, bwd {rt_bwd:.3f} ms "
          f"per layer invocation")
    os.makedirs(OUTDIR, exist_ok=True)
    with open(path, "w") as f:
        json.dump(out, f, indent=1)
    print(json.dumps(out, indent=1))


if __name__ == "__main__":
    main()
