"""Train Llama-3 8B on ONE MI355X with the in-repo reference trainer
(synthetic data, bf16 params + fp32 flat Adam, gfx950 HIP kernels).

This is the real-run side of the perf-vs-real validation; multi-GPU DP
runs via `torchrun --nproc-per-node N ../bench.py --gpus N`.
"""
import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from simumax_amd import ModelConfig, get_simu_model_config
from simumax_amd.train.trainer import (TrainConfig, build_trainer,
                                       make_synthetic_batch, train_step)


def main(steps=5):
    assert torch.cuda.is_available(), "this example needs an MI355X"
    mc = ModelConfig.init_from_config_file(get_simu_model_config("llama3-8b"))
    tc = TrainConfig(seq_len=4096, micro_batch_size=1, micro_batch_num=4)
    model, opt, red = build_trainer(mc, tc, "cuda:0")
    toks, labels = make_synthetic_batch(mc.vocab_size, 4, 1, 4096, "cuda:0")
    for i in range(steps):
        loss = train_step(model, opt, red, toks, labels, 4)
        print(f"step {i}: loss {loss:.4f}, "
              f"peak {torch.cuda.max_memory_allocated()/2**30:.1f} GiB")


if __name__ == "__main__":
    main()
