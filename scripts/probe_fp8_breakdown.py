"""Bisect the fp8 over-prediction: predict with parts of the in-situ
overlay removed to attribute the gap (casts vs GEMMs vs rest)."""
import sys, os, json, copy, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from simumax_amd import ModelConfig, get_simu_model_config
from simumax_amd.train.trainer import (TrainConfig, build_trainer,
                                       make_synthetic_batch, train_step)
from simumax_amd.kernels import insitu
import scripts.validation_sweep as vs

mc = ModelConfig.init_from_config_file(get_simu_model_config("llama3-8b"))
tc = TrainConfig(seq_len=4096, micro_batch_size=1, micro_batch_num=4, fp8=True)
m, opt, red = build_trainer(mc, tc, "cuda:0")
toks, labels = make_synthetic_batch(mc.vocab_size, 4, 1, 4096, "cuda:0")
train_step(m, opt, red, toks, labels, 4)
insitu.enable()
train_step(m, opt, red, toks, labels, 4)
torch.cuda.synchronize()
insitu.disable()
overlay = insitu.summarize()
# measured
import time
torch.cuda.synchronize(); t0=time.time()
for _ in range(3):
    train_step(m, opt, red, toks, labels, 4)
torch.cuda.synchronize()
ms=(time.time()-t0)/3*1e3
print(f"measured {ms:.1f} ms")
# total in-situ cast/gemm time
for tab in ("fp8_matmul",):
    tot = sum(r["t_ms"]*r["n"] for r in overlay.get(tab,{}).values())
    print(f"insitu {tab}: total {tot:.1f} ms ({len(overlay.get(tab,{}))} keys)")
bw = overlay.get("bandwidth", {})
print("fp8_quant_eff:", bw.get("fp8_quant_eff"))

def pred(ov, note):
    c, _ = vs.predict(mc, 4096, 1, 4, fp8=True, overlay=ov)
    print(f"predicted {c['iter_time']:.1f} ms  [{note}]")

pred(overlay, "full overlay")
ov2 = copy.deepcopy(overlay); ov2.get("bandwidth",{}).pop("fp8_quant_eff", None)
pred(ov2, "no fp8_quant overlay (static 0.55)")
ov3 = copy.deepcopy(overlay); ov3.pop("fp8_matmul", None)
pred(ov3, "no fp8_matmul overlay (sweep tables)")
ov4 = copy.deepcopy(overlay); ov4.pop("matmul", None)
pred(ov4, "no bf16 matmul overlay")
ov5 = copy.deepcopy(overlay); ov5.pop("sdp_fwd", None); ov5.pop("sdp_bwd", None)
pred(ov5, "no sdp overlay")
# per-key compare: insitu fp8_matmul eff vs charged
rows = sorted(overlay.get("fp8_matmul", {}).items(),
              key=lambda kv: -kv[1]["t_ms"]*kv[1]["n"])[:6]
for k, r in rows:
    print(f"  {k[:70]:70s} t={r['t_ms']:.3f} n={r['n']} eff={r.get('eff'):.3f}")
red.remove_hooks()

import json as _json
os.makedirs("gpurun_out", exist_ok=True)
with open("gpurun_out/fp8_overlay.json", "w") as f:
    _json.dump({k: dict(v) for k, v in overlay.items()}, f, indent=1)
tot_all = {}
for (table, key), pairs in insitu._RECORDS.items():
    pass
print("overlay dumped")
