"""In-situ calibration + measurement of the fp8 trainer path (llama3-8b
bench config): dumps fp8_matmul_insitu.json and prints the measured
step time for the fp8 perf-vs-real row."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from simumax_amd import ModelConfig, get_simu_model_config
from simumax_amd.kernels import insitu
from simumax_amd.train.trainer import (TrainConfig, build_trainer,
                                       make_synthetic_batch, train_step)


def main():
    mc = ModelConfig.init_from_config_file(get_simu_model_config("llama3-8b"))
    tc = TrainConfig(seq_len=4096, micro_batch_size=1, micro_batch_num=4,
                     fp8=True)
    m, opt, red = build_trainer(mc, tc, "cuda:0")
    toks, labels = make_synthetic_batch(mc.vocab_size, 4, 1, 4096, "cuda:0")
    for _ in range(2):
        train_step(m, opt, red, toks, labels, 4)
    torch.cuda.synchronize()
    insitu.enable()
    for _ in range(2):
        train_step(m, opt, red, toks, labels, 4)
    torch.cuda.synchronize()
    insitu.disable()
    insitu.dump("gpurun_out/calib")
    torch.cuda.reset_peak_memory_stats()
    t0 = time.time()
    for _ in range(4):
        train_step(m, opt, red, toks, labels, 4)
    torch.cuda.synchronize()
    ms = (time.time() - t0) / 4 * 1e3
    out = dict(case="8b_fp8_seq4096_mbc4", measured_ms=round(ms, 2),
               measured_gib=round(torch.cuda.max_memory_allocated() / 2**30, 2))
    print(json.dumps(out))
    with open("gpurun_out/fp8_row.json", "w") as f:
        json.dump(out, f)


if __name__ == "__main__":
    main()
