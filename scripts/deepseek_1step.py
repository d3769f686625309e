import sys, os
sys.path.insert(0, "/root/repo")
import torch
from simumax_amd import ModelConfig, get_simu_model_config
from simumax_amd.train.trainer import (TrainConfig, build_trainer,
                                       make_synthetic_batch, train_step)
mc = ModelConfig.init_from_config_file(get_simu_model_config("deepseekv2-l4"))
tc = TrainConfig(seq_len=4096, micro_batch_size=1, micro_batch_num=2)
m, opt, red = build_trainer(mc, tc, "cuda:0")
toks, labels = make_synthetic_batch(mc.vocab_size, 2, 1, 4096, "cuda:0")
train_step(m, opt, red, toks, labels, 2)
torch.cuda.synchronize()
train_step(m, opt, red, toks, labels, 2)
torch.cuda.synchronize()
