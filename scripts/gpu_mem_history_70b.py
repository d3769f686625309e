"""Allocator-history probe: what is live at the 70b-l12 backward peak."""
import os, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from simumax_amd import ModelConfig, get_simu_model_config
from simumax_amd.train.trainer import TrainConfig, build_trainer, make_synthetic_batch, train_step

mc = ModelConfig.init_from_config_file(get_simu_model_config("llama3-70b-l12"))
tc = TrainConfig(seq_len=4096, micro_batch_size=1, micro_batch_num=1)
model, opt, red = build_trainer(mc, tc, "cuda:0")
toks, labels = make_synthetic_batch(mc.vocab_size, 1, 1, 4096, "cuda:0")
train_step(model, opt, red, toks, labels, 1)  # warmup
torch.cuda.synchronize()
torch.cuda.memory._record_memory_history(max_entries=200000)
train_step(model, opt, red, toks, labels, 1)
torch.cuda.synchronize()
snap = torch.cuda.memory._snapshot()
torch.cuda.memory._record_memory_history(enabled=None)

# replay the trace: live set + peak
trace = snap["device_traces"][0]
live = {}
cur = peak = 0
peak_live = None
for ev in trace:
    if ev["action"] == "alloc":
        live[ev["addr"]] = ev
        cur += ev["size"]
        if cur > peak:
            peak = cur
            peak_live = dict(live)
    elif ev["action"] in ("free_completed",):
        e = live.pop(ev["addr"], None)
        if e is not None:
            cur -= e["size"]
print(f"trace events {len(trace)}; traced peak delta {peak/2**30:.2f} GiB over {len(peak_live or {})} blocks")
if peak_live:
    blocks = sorted(peak_live.values(), key=lambda e: -e["size"])[:25]
    for e in blocks:
        frames = [f"{f['filename'].split('/')[-1]}:{f['line']}:{f.get('name','')}"
                  for f in e.get("frames", [])][:6]
        print(f"  {e['size']/2**20:9.1f} MiB  {frames}")
