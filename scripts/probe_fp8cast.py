import sys, os, torch, json
sys.path.insert(0, "/root/repo")
from simumax_amd.kernels import fp8, insitu
from simumax_amd.kernels.ops import ext
E = ext()
t = torch.randn(4096, 4096, device="cuda", dtype=torch.bfloat16)
amax = torch.zeros((), device="cuda")
scale = torch.ones(1, device="cuda")
q = torch.empty(t.shape, dtype=torch.float8_e4m3fn, device="cuda")
qt = torch.empty((4096, 4096), dtype=torch.float8_e4m3fn, device="cuda")
for _ in range(5):
    E.fp8_cast_t(t, q, qt, amax, scale, False, 448.0)
torch.cuda.synchronize()
s = torch.cuda.Event(True); e = torch.cuda.Event(True)
s.record()
for _ in range(20):
    E.fp8_cast_t(t, q, qt, amax, scale, False, 448.0)
e.record(); torch.cuda.synchronize()
ms = s.elapsed_time(e)/20
print(f"cast_t 4096x4096: {ms*1e3:.1f} us, eff={4*t.numel()/(ms/1e3)/8e12*0.931:.3f}")
# non-square (down proj input)
t2 = torch.randn(4096, 14336, device="cuda", dtype=torch.bfloat16)
q2 = torch.empty_like(t2, dtype=torch.float8_e4m3fn)
qt2 = torch.empty((14336, 4096), dtype=torch.float8_e4m3fn, device="cuda")
for _ in range(5):
    E.fp8_cast_t(t2, q2, qt2, amax, scale, False, 448.0)
torch.cuda.synchronize()
s.record()
for _ in range(20):
    E.fp8_cast_t(t2, q2, qt2, amax, scale, False, 448.0)
e.record(); torch.cuda.synchronize()
ms = s.elapsed_time(e)/20
print(f"cast_t 4096x14336: {ms*1e3:.1f} us, eff={4*t2.numel()/(ms/1e3)/8e12*0.931:.3f}")
# numerics: q matches plain cast, qt is its transpose
E.fp8_cast_t(t, q, qt, amax, scale, False, 448.0)
q_ref = torch.empty_like(q)
E.fp8_cast(t, q_ref, amax, scale, False, 448.0)
assert torch.equal(q.view(torch.uint8), q_ref.view(torch.uint8)), "q mismatch"
assert torch.equal(qt.view(torch.uint8), q_ref.view(torch.uint8).t().contiguous()), "qt mismatch"
print("cast_t numerics OK")
# plain cast speed
for _ in range(5):
    E.fp8_cast(t, q_ref, amax, scale, False, 448.0)
torch.cuda.synchronize(); s.record()
for _ in range(20):
    E.fp8_cast(t, q_ref, amax, scale, False, 448.0)
e.record(); torch.cuda.synchronize()
ms = s.elapsed_time(e)/20
print(f"plain cast 4096x4096: {ms*1e3:.1f} us ({3*t.numel()/(ms/1e3)/1e12:.2f} TB/s actual)")

# now one fp8 trainer microstep with insitu, dump bw_fp8_quant stats
from simumax_amd import ModelConfig, get_simu_model_config
from simumax_amd.train.trainer import TrainConfig, build_trainer, make_synthetic_batch, train_step
mc = ModelConfig.init_from_config_file(get_simu_model_config("llama3-8b"))
tc = TrainConfig(seq_len=4096, micro_batch_size=1, micro_batch_num=1, fp8=True)
m, opt, red = build_trainer(mc, tc, "cuda:0")
toks, labels = make_synthetic_batch(mc.vocab_size, 1, 1, 4096, "cuda:0")
train_step(m, opt, red, toks, labels, 1)
insitu.enable()
train_step(m, opt, red, toks, labels, 1)
torch.cuda.synchronize()
insitu.disable()
tot_ms = 0.0; tot_b = 0.0; n = 0
for (table, key), pairs in insitu._RECORDS.items():
    if table == "bw_fp8_quant":
        for p in pairs:
            tot_ms += p[0].elapsed_time(p[1]); tot_b += float(key); n += 1
print(f"insitu bw_fp8_quant: n={n} total_ms={tot_ms:.2f} bytes={tot_b/1e9:.1f}GB eff={tot_b/((tot_ms-n*0.004)/1e3)/ (8000*1024**3):.3f}")
summ = insitu.summarize()
print("summary fp8_quant_eff:", summ.get("bandwidth", {}).get("fp8_quant_eff"))
red.remove_hooks()
