"""torch.profiler attribution of one mixtral step: which aten ops
consume the CUDA time the simulator doesn't model. Writes the top rows
by CUDA time to gpurun_out/moe_profile.txt."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from torch.profiler import ProfilerActivity, profile

from simumax_amd import ModelConfig, get_simu_model_config
from simumax_amd.train.trainer import (TrainConfig, build_trainer,
                                       make_synthetic_batch, train_step)


def main():
    mc = ModelConfig.init_from_config_file(get_simu_model_config("mixtral-8x7b-l8"))
    tc = TrainConfig(seq_len=4096, micro_batch_size=1, micro_batch_num=2)
    m, opt, red = build_trainer(mc, tc, "cuda:0")
    toks, labels = make_synthetic_batch(mc.vocab_size, 2, 1, 4096, "cuda:0")
    train_step(m, opt, red, toks, labels, 2)
    torch.cuda.synchronize()
    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                 record_shapes=True) as prof:
        train_step(m, opt, red, toks, labels, 2)
        torch.cuda.synchronize()
    table = prof.key_averages(group_by_input_shape=True).table(
        sort_by="cuda_time_total", row_limit=60, max_src_column_width=40)
    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/moe_profile.txt", "w") as f:
        f.write(table)
    print(table)


if __name__ == "__main__":
    main()
