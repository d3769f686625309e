"""Find the unmodeled MoE memory: record allocator history for one
mixtral-8x7b-l8 step, replay the trace to the peak point, and print the
live allocations at peak grouped by size/stack. The simulator's mixtral
prediction is ~3.9 GiB (1.8%) under measurement — this names the gap.
"""
import os
import sys
from collections import defaultdict

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from simumax_amd import ModelConfig, get_simu_model_config
from simumax_amd.train.trainer import (TrainConfig, build_trainer,
                                       make_synthetic_batch, train_step)


def main():
    mc = ModelConfig.init_from_config_file(get_simu_model_config("mixtral-8x7b-l8"))
    tc = TrainConfig(seq_len=4096, micro_batch_size=1, micro_batch_num=2)
    m, opt, red = build_trainer(mc, tc, "cuda:0")
    toks, labels = make_synthetic_batch(mc.vocab_size, 2, 1, 4096, "cuda:0")
    train_step(m, opt, red, toks, labels, 2)  # warmup
    torch.cuda.synchronize()
    torch.cuda.memory._record_memory_history(max_entries=400000)
    train_step(m, opt, red, toks, labels, 2)
    torch.cuda.synchronize()
    snap = torch.cuda.memory._snapshot()
    torch.cuda.memory._record_memory_history(enabled=None)

    traces = snap["device_traces"][0]
    live = {}
    cur = peak = 0
    peak_live = None
    for ev in traces:
        a = ev["action"]
        if a == "alloc":
            live[ev["addr"]] = ev
            cur += ev["size"]
            if cur > peak:
                peak = cur
                peak_live = dict(live)
        elif a in ("free_completed", "free_requested"):
            if a == "free_completed" and ev["addr"] in live:
                cur -= live[ev["addr"]]["size"]
                del live[ev["addr"]]
    print(f"trace peak delta: {peak/2**30:.2f} GiB over {len(traces)} events "
          f"(pre-existing steady allocs not in trace)")
    print(f"max_memory_allocated: {torch.cuda.max_memory_allocated()/2**30:.2f} GiB")

    groups = defaultdict(lambda: [0, 0, None])
    for ev in peak_live.values():
        fr = ev.get("frames") or []
        sig = None
        for f in fr:
            fn = f.get("filename", "")
            if "simumax_amd" in fn or "autograd" in fn:
                sig = f"{os.path.basename(fn)}:{f['line']}:{f['name']}"
                break
        if sig is None and fr:
            f = fr[0]
            sig = f"{os.path.basename(f.get('filename','?'))}:{f.get('line',0)}:{f.get('name','?')}"
        g = groups[(sig, ev["size"])]
        g[0] += ev["size"]
        g[1] += 1
        g[2] = fr[:6]
    rows = sorted(groups.items(), key=lambda kv: -kv[1][0])[:40]
    print("\nlive-at-peak allocations (grouped by site+size), top 40:")
    for (sig, size), (tot, n, fr) in rows:
        print(f"  {tot/2**30:8.3f} GiB  n={n:4d}  each={size/2**20:9.2f} MiB  {sig}")
        if tot > 0.5 * 2**30 and fr:
            for f in fr:
                print(f"        {os.path.basename(f.get('filename','?'))}:"
                      f"{f.get('line',0)} {f.get('name','?')}")


if __name__ == "__main__":
    main()
