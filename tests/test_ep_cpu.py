"""Expert-parallel trainer equivalence on CPU (gloo, world 2, EP2):
the a2a-dispatched 2-rank run must reproduce the single-process EP1
gradients — experts exactly (edp=1, no averaging), dense params up to
the 1/world DP averaging factor."""

import os
import sys

import pytest
import torch
import torch.multiprocessing as mp

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _tiny_cfg():
    from simumax_amd.core.config import ModelConfig

    return ModelConfig(hidden_size=128, head_num=2, kv_head_num=1,
                       head_size=64, intermediate_size=256,
                       layer_num=2, vocab_size=512, use_swiglu=True,
                       model_type="moe", expert_num=4, topk=2,
                       moe_ffn_hidden_size=96,
                       moe_shared_expert_intermediate_size=64)


def _worker(rank, world, port, q):
    import torch.distributed as dist

    sys.path.insert(0, REPO)
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from simumax_amd.train.trainer import (MixedPrecisionAdam,
                                               TrainConfig, build_trainer,
                                               accumulate_main_grads,
                                               make_synthetic_batch)
        from simumax_amd.train.model import LlamaForTraining

        cfg = _tiny_cfg()
        tc = TrainConfig(seq_len=32, micro_batch_size=2, micro_batch_num=2,
                         overlap_grad_reduce=False, ep_size=2)
        model, opt, red = build_trainer(cfg, tc, "cpu")
        le = cfg.expert_num // 2

        # EP1 reference model (full experts), same dense init by seed
        torch.manual_seed(1234)
        ref = LlamaForTraining(cfg, tc.seq_len, device="cpu")
        ref_opt = MixedPrecisionAdam(ref.parameters(), tc)

        # make EP2 weights an exact shard of the reference
        with torch.no_grad():
            rd = dict(ref.named_parameters())
            for name, p in model.named_parameters():
                r = rd[name]
                if getattr(p, "_is_expert", False):
                    p.copy_(r[rank * le:(rank + 1) * le])
                else:
                    p.copy_(r)

        toks, labels = make_synthetic_batch(cfg.vocab_size, world, 2, 32,
                                            "cpu", seed=7)
        # EP2: this rank trains on microbatch `rank`
        opt.zero_grad()
        red.reduce_this_pass = True
        loss = model(toks[rank], labels[rank])
        loss.backward()
        red.finalize()

        # EP1 reference: both microbatches, summed grads
        ref_opt.zero_grad()
        for mb in range(world):
            ref(toks[mb], labels[mb]).backward()
            accumulate_main_grads(ref_opt.params)

        errs = {}
        rd = dict(ref.named_parameters())
        for name, p in model.named_parameters():
            r = rd[name]
            if getattr(p, "_is_expert", False):
                want = r.main_grad[rank * le:(rank + 1) * le]
                got = p.main_grad
            else:
                want = r.main_grad
                got = p.main_grad * world  # DP averaging factor
            denom = want.abs().max().clamp(min=1e-4)
            errs[name] = float((got - want).abs().max() / denom)
        q.put((rank, float(loss), errs))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_ep2_matches_ep1_gradients():
    mp.set_start_method("spawn", force=True)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, 29517, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, loss, errs = q.get(timeout=240)
        results[rank] = (loss, errs)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    for rank, (loss, errs) in results.items():
        assert loss == loss  # finite
        bad = {n: e for n, e in errs.items() if e > 3e-2}
        assert not bad, f"rank {rank} grad mismatches: {bad}"


def _zero_ep_worker(rank, world, port, q):
    import torch.distributed as dist

    sys.path.insert(0, REPO)
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from simumax_amd.train.trainer import (TrainConfig, build_trainer,
                                               make_synthetic_batch,
                                               train_step)

        cfg = _tiny_cfg()
        results = {}
        for zero in (0, 1):
            tc = TrainConfig(seq_len=32, micro_batch_size=2,
                             micro_batch_num=2, overlap_grad_reduce=False,
                             ep_size=2, zero_state=zero, grad_clip=1e9)
            model, opt, red = build_trainer(cfg, tc, "cpu")
            if zero == 1:
                # dense shards over world(4), expert over edp(2)
                assert opt.master_numel < opt.flat_param.numel()
            toks, labels = make_synthetic_batch(cfg.vocab_size, 2, 2, 32,
                                                "cpu",
                                                seed=(rank // 2) * 7 + 1)
            for _ in range(2):
                train_step(model, opt, red, toks, labels, 2)
            results[zero] = opt.flat_param.detach().float().clone()
            red.remove_hooks()
            del model, opt, red
        err = (results[0] - results[1]).abs().max().item()
        q.put((rank, err))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(420)
def test_zero1_with_ep2_matches_zero0():
    """ZeRO-1 x EP on world 4 (ep2 x edp2): segmented optimizer state
    (dense sharded over the world, experts over edp) must reproduce the
    replicated-optimizer parameters after 2 steps."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29667
    ps = [ctx.Process(target=_zero_ep_worker, args=(r, 4, port, q))
          for r in range(4)]
    for p in ps:
        p.start()
    results = [q.get(timeout=400) for _ in range(4)]
    for p in ps:
        p.join(timeout=60)
    for rank, err in results:
        assert err < 2e-3, f"rank {rank} param drift {err}"
