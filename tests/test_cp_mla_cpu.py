"""MLA + context parallelism on CPU (gloo, world 2): the all_gather and
ring CP modes compose with multi-head latent attention (asymmetric
192/128 head dims, shared k_pe) — a2a cannot (head-scatter clashes with
the per-token positional key). Gradients must match the single-process
full-sequence MLA run."""

import os
import sys

import pytest
import torch
import torch.multiprocessing as mp

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _mla_cfg():
    from simumax_amd.core.config import ModelConfig

    return ModelConfig(hidden_size=128, head_num=4, kv_head_num=4,
                       head_size=32, intermediate_size=256, layer_num=2,
                       vocab_size=512, use_swiglu=True,
                       attention_type="mla", q_lora_rank=64,
                       kv_lora_rank=64, qk_head_dim=32,
                       qk_pos_emb_head_dim=16, v_head_dim=32)


def _worker(rank, world, port, q, cp_comm_type):
    import torch.distributed as dist

    sys.path.insert(0, REPO)
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from simumax_amd.train.model import LlamaForTraining
        from simumax_amd.train.trainer import (MixedPrecisionAdam,
                                               TrainConfig, build_trainer,
                                               accumulate_main_grads,
                                               make_synthetic_batch)

        cfg = _mla_cfg()
        S = 64
        tc = TrainConfig(seq_len=S, micro_batch_size=2, micro_batch_num=1,
                         overlap_grad_reduce=False, cp_size=2,
                         cp_comm_type=cp_comm_type)
        model, opt, red = build_trainer(cfg, tc, "cpu")

        torch.manual_seed(1234)
        ref = LlamaForTraining(cfg, S, device="cpu")
        ref_opt = MixedPrecisionAdam(ref.parameters(), tc)
        rd = dict(ref.named_parameters())
        with torch.no_grad():
            for name, p in model.named_parameters():
                p.copy_(rd[name])

        toks, labels = make_synthetic_batch(cfg.vocab_size, 1, 2, S,
                                            "cpu", seed=77)
        s_loc = S // world
        sl = slice(rank * s_loc, (rank + 1) * s_loc)
        loss = model(toks[0][:, sl], labels[0][:, sl])
        loss.backward()
        accumulate_main_grads([p for p in model.parameters()])
        red.finalize()

        ref_loss = ref(toks[0], labels[0])
        ref_loss.backward()
        accumulate_main_grads([p for p in ref.parameters()])

        lt = loss.detach().clone()
        dist.all_reduce(lt)
        lt /= world
        lerr = (lt - ref_loss.detach()).abs().item()
        gerr = 0.0
        rd = dict(ref.named_parameters())
        for name, p in model.named_parameters():
            r = rd[name].main_grad
            denom = r.abs().max().clamp(min=1e-6)
            gerr = max(gerr, ((p.main_grad - r).abs().max() / denom).item())
        q.put((rank, lerr, gerr))
    finally:
        dist.destroy_process_group()


def _run(port, mode):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    ps = [ctx.Process(target=_worker, args=(r, 2, port, q, mode))
          for r in range(2)]
    for p in ps:
        p.start()
    results = [q.get(timeout=280) for _ in range(2)]
    for p in ps:
        p.join(timeout=60)
    for rank, lerr, gerr in results:
        assert lerr < 2e-3, f"rank {rank} loss err {lerr}"
        assert gerr < 0.08, f"rank {rank} grad rel err {gerr}"


@pytest.mark.timeout(300)
def test_mla_cp2_all_gather():
    _run(29643, "all_gather")


@pytest.mark.timeout(300)
def test_mla_cp2_ring():
    _run(29646, "ring")
