"""Burst vs sustained GEMM timing probe.

The calibration sweep (`calib/sweeps._timeit`) times 10 identical
iterations back-to-back: the weight stays hot in L2 and the clocks are
at their short-burst ceiling. Inside a real training step the same GEMM
runs in a mixed kernel stream at sustained power with cold-ish inputs.
This probe quantifies that gap for the llama3-8b dominant shapes:

* burst: the exact microbench loop the sweep uses
* sustained: all shapes x stages round-robin for many rounds with 4
  rotating buffer sets each (defeats L2 activation reuse, holds the chip
  at sustained power), per-call hipEvent timing

Writes gpurun_out/sustained_probe.json: per (shape, stage)
{burst_ms, sustained_ms, ratio}. The geometric-mean ratio is the
measured in-situ derate the merge step can apply to GEMM efficiency.
"""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from simumax_amd.kernels.ops import ext

# llama3-8b per-layer GEMMs at M=4096 tokens (tp1), plus lm_head
SHAPES = [
    ("qkv", 4096, 4096, 6144),
    ("attn_out", 4096, 4096, 4096),
    ("mlp_up", 4096, 4096, 28672),
    ("mlp_down", 4096, 14336, 4096),
    ("lm_head", 4096, 4096, 128256),
]
STAGES = ["fwd", "dgrad", "wgrad"]
NBUF = 4


def make_op(m, k, n, stage, dev):
    """Return (fn(buf_idx), flops) matching FusedLinear's real calls."""
    E = ext()
    if stage == "fwd":
        xs = [torch.randn(m, k, dtype=torch.bfloat16, device=dev) for _ in range(NBUF)]
        w = torch.randn(n, k, dtype=torch.bfloat16, device=dev)
        return lambda i: torch.matmul(xs[i], w.t()), 2 * m * k * n
    if stage == "dgrad":
        ds = [torch.randn(m, n, dtype=torch.bfloat16, device=dev) for _ in range(NBUF)]
        w = torch.randn(n, k, dtype=torch.bfloat16, device=dev)
        return lambda i: torch.matmul(ds[i], w), 2 * m * k * n
    # wgrad: main_grad[n,k] += dout^T @ x (fp32 accum GemmEx)
    ds = [torch.randn(m, n, dtype=torch.bfloat16, device=dev) for _ in range(NBUF)]
    xs = [torch.randn(m, k, dtype=torch.bfloat16, device=dev) for _ in range(NBUF)]
    g = torch.zeros(n, k, dtype=torch.float32, device=dev)
    return lambda i: E.wgrad_accum(ds[i], xs[i], g), 2 * m * k * n


def burst_time(fn, iters=10):
    for _ in range(3):
        fn(0)
    torch.cuda.synchronize()
    s, e = torch.cuda.Event(True), torch.cuda.Event(True)
    s.record()
    for _ in range(iters):
        fn(0)
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters


def main():
    dev = "cuda:0"
    ops = {}
    for name, m, k, n in SHAPES:
        for stage in STAGES:
            fn, flops = make_op(m, k, n, stage, dev)
            ops[(name, stage)] = dict(fn=fn, flops=flops)

    # burst pass (microbench style)
    for key, op in ops.items():
        op["burst_ms"] = burst_time(op["fn"])

    # sustained pass: round-robin everything, rotating buffers
    keys = list(ops.keys())
    ROUNDS = 24
    SKIP = 4  # warm rounds (reach sustained clocks) excluded
    events = {k: [] for k in keys}
    torch.cuda.synchronize()
    for r in range(ROUNDS):
        for key in keys:
            op = ops[key]
            s, e = torch.cuda.Event(True), torch.cuda.Event(True)
            s.record()
            op["fn"](r % NBUF)
            e.record()
            if r >= SKIP:
                events[key].append((s, e))
    torch.cuda.synchronize()

    out = {}
    ratios = []
    for key in keys:
        ts = [s.elapsed_time(e) for s, e in events[key]]
        ts.sort()
        sus = sum(ts[: len(ts) // 2 + 1]) / (len(ts) // 2 + 1)  # robust lower half
        b = ops[key]["burst_ms"]
        fl = ops[key]["flops"]
        out["|".join(key)] = dict(
            burst_ms=round(b, 4), sustained_ms=round(sus, 4),
            ratio=round(sus / b, 4),
            burst_tflops=round(fl / b / 1e9, 1),
            sustained_tflops=round(fl / sus / 1e9, 1))
        ratios.append(sus / b)
        print(f"{key[0]:10s} {key[1]:6s} burst {b:8.3f} ms  sustained {sus:8.3f} ms  "
              f"ratio {sus/b:.3f}")
    import math

    gmean = math.exp(sum(math.log(r) for r in ratios) / len(ratios))
    out["_gmean_ratio"] = round(gmean, 4)
    print(f"gmean sustained/burst ratio: {gmean:.4f}")
    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/sustained_probe.json", "w") as f:
        json.dump(out, f, indent=1)


if __name__ == "__main__":
    main()
