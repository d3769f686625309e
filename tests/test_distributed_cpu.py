"""Multi-process DP tests on CPU (gloo, world_size 2): exercises the same
torch.distributed code path bench.py runs on N GPUs over RCCL."""

import json
import os
import sys

import pytest
import torch
import torch.multiprocessing as mp

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _worker(rank, world, port, q):
    import torch.distributed as dist

    sys.path.insert(0, REPO)
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from simumax_amd.core.config import ModelConfig
        from simumax_amd.train.trainer import (TrainConfig, build_trainer,
                                               make_synthetic_batch,
                                               train_step)

        cfg = ModelConfig(hidden_size=128, head_num=2, kv_head_num=1,
                          head_size=64, intermediate_size=256, layer_num=2,
                          vocab_size=512, use_swiglu=True)
        tc = TrainConfig(seq_len=32, micro_batch_size=2, micro_batch_num=2,
                         overlap_grad_reduce=True)
        model, opt, reducer = build_trainer(cfg, tc, "cpu")
        assert reducer.enabled
        losses = []
        for step in range(3):
            toks, labels = make_synthetic_batch(cfg.vocab_size, 2, 2, 32,
                                                "cpu", seed=rank * 100 + step)
            losses.append(train_step(model, opt, reducer, toks, labels, 2))
        # after reduced grads + identical init, params must match across ranks
        fingerprint = float(opt.flat_param.float().sum())
        q.put((rank, losses, fingerprint))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(240)
def test_dp2_gloo_train_step_param_sync():
    """2-rank DP training: overlapped bucketed all_reduce keeps parameters
    bit-identical across ranks."""
    mp.set_start_method("spawn", force=True)
    q = mp.get_context("spawn").Queue()
    port = 29511
    procs = [mp.get_context("spawn").Process(target=_worker,
                                             args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, losses, fp = q.get(timeout=200)
        results[rank] = (losses, fp)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    assert len(results) == 2
    # identical parameter state on both ranks (grads were averaged)
    assert results[0][1] == pytest.approx(results[1][1], rel=1e-5)
    # losses finite
    for losses, _ in results.values():
        assert all(l == l for l in losses)


def _worker_zero(rank, world, port, q):
    import torch.distributed as dist

    sys.path.insert(0, REPO)
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from simumax_amd.core.config import ModelConfig
        from simumax_amd.train.trainer import (TrainConfig, build_trainer,
                                               make_synthetic_batch,
                                               train_step)

        cfg = ModelConfig(hidden_size=128, head_num=2, kv_head_num=1,
                          head_size=64, intermediate_size=256, layer_num=2,
                          vocab_size=512, use_swiglu=True)
        results = {}
        for zero in (0, 1):
            tc = TrainConfig(seq_len=32, micro_batch_size=2,
                             micro_batch_num=2, overlap_grad_reduce=False,
                             zero_state=zero, grad_clip=1e9)
            model, opt, red = build_trainer(cfg, tc, "cpu")
            if zero == 1:
                # state really is sharded
                assert opt.master_numel * 2 >= opt.flat_param.numel()
                assert opt.master_numel < opt.flat_param.numel()
            toks, labels = make_synthetic_batch(cfg.vocab_size, 2, 2, 32,
                                                "cpu", seed=rank * 7 + 1)
            for _ in range(2):
                train_step(model, opt, red, toks, labels, 2)
            results[zero] = opt.flat_param.detach().float().clone()
            red.remove_hooks()
        total = min(r.numel() for r in results.values())
        diff = (results[0][:total] - results[1][:total]).abs().max()
        q.put((rank, float(diff)))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_zero1_matches_zero0_params():
    """ZeRO-1 (sharded optimizer state + param all_gather) must produce
    the same parameters as the replicated optimizer after training."""
    mp.set_start_method("spawn", force=True)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_zero, args=(r, 2, 29535, q))
             for r in range(2)]
    for p in procs:
        p.start()
    for _ in range(2):
        rank, diff = q.get(timeout=240)
        assert diff < 1e-6, f"rank {rank}: param divergence {diff}"
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0


def test_recompute_layers_bitexact():
    """Full-block activation recompute (torch.utils.checkpoint) must be
    gradient-bitexact with the plain forward (no dropout, same RNG-free
    path)."""
    import torch

    from simumax_amd.core.config import ModelConfig
    from simumax_amd.train.trainer import (TrainConfig, build_trainer,
                                           accumulate_main_grads,
                                           make_synthetic_batch)

    cfg = ModelConfig(hidden_size=128, head_num=4, kv_head_num=2,
                      head_size=32, intermediate_size=256, layer_num=2,
                      vocab_size=512, use_swiglu=True)
    grads = {}
    for rc in (0, 2):
        torch.manual_seed(0)
        tc = TrainConfig(seq_len=64, micro_batch_size=2, micro_batch_num=1,
                         recompute_layers=rc, grad_clip=1e9)
        m, opt, red = build_trainer(cfg, tc, "cpu")
        toks, labels = make_synthetic_batch(cfg.vocab_size, 1, 2, 64,
                                            "cpu", seed=3)
        opt.zero_grad()
        m(toks[0], labels[0]).backward()
        accumulate_main_grads(opt.params)
        grads[rc] = torch.cat([p.main_grad.reshape(-1) for p in opt.params])
        red.remove_hooks()
    assert torch.equal(grads[0], grads[2])
