"""Localize the 70b-l12 memory-prediction gap: per-phase peaks."""
import os, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from simumax_amd import ModelConfig, get_simu_model_config
from simumax_amd.train.trainer import TrainConfig, build_trainer, make_synthetic_batch, train_step, accumulate_main_grads

mc = ModelConfig.init_from_config_file(get_simu_model_config("llama3-70b-l12"))
tc = TrainConfig(seq_len=4096, micro_batch_size=1, micro_batch_num=2)
model, opt, red = build_trainer(mc, tc, "cuda:0")
toks, labels = make_synthetic_batch(mc.vocab_size, 2, 2 if False else 1, 4096, "cuda:0")
train_step(model, opt, red, toks, labels, 2)  # warmup
torch.cuda.synchronize()
base = torch.cuda.memory_allocated()/2**30
print(f"live after warmup (static): {base:.2f} GiB")

def peak(phase, fn):
    torch.cuda.reset_peak_memory_stats()
    fn()
    torch.cuda.synchronize()
    print(f"{phase}: peak {torch.cuda.max_memory_allocated()/2**30:.2f} GiB "
          f"(live {torch.cuda.memory_allocated()/2**30:.2f})")

opt.zero_grad()
losses = []
def fwd_bwd(mb):
    red.reduce_this_pass = mb == 1
    loss = model(toks[mb], labels[mb])
    peak_fwd = torch.cuda.max_memory_allocated()/2**30
    print(f"  mb{mb} after fwd: live {torch.cuda.memory_allocated()/2**30:.2f} peak-so-far {peak_fwd:.2f}")
    loss.backward()
    accumulate_main_grads(opt.params)

peak("mb0 fwd+bwd", lambda: fwd_bwd(0))
peak("mb1 fwd+bwd", lambda: fwd_bwd(1))
peak("optimizer", opt.step)
