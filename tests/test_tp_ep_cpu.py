"""tp x ep trainer composition on CPU (gloo, world 4 = tp2(SP) x ep2):
sequence-parallel tp ranks route their own token shards through the EP
all-to-all; gradients must reproduce the single-process run (tp-sharded
and replicated dense params, router/shared-expert tp-partial reduction,
ep-sharded experts with edp averaging)."""

import os
import sys

import pytest
import torch
import torch.multiprocessing as mp

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _tiny_cfg():
    from simumax_amd.core.config import ModelConfig

    cfg = ModelConfig(hidden_size=128, head_num=4, kv_head_num=2,
                      head_size=32, intermediate_size=256,
                      layer_num=2, vocab_size=512, use_swiglu=True,
                      model_type="moe", expert_num=4, topk=2,
                      moe_ffn_hidden_size=96,
                      moe_shared_expert_intermediate_size=64)
    # capacity drops are enforced per SOURCE rank in the distributed run
    # but globally in the single-process reference; a 4x capacity factor
    # makes routing drop-free on both sides so gradients compare exactly
    cfg.capacity = 4.0
    return cfg


def _shard_dense(name, r, t, cfg, tp=2):
    """Reference tensor -> this tp rank's shard (None = full copy)."""
    d = cfg.head_size
    hq_l, hkv_l = cfg.head_num // tp, cfg.kv_head_num // tp
    v_l = cfg.vocab_size // tp
    if "qkv_proj" in name:
        hq, hkv = cfg.head_num, cfg.kv_head_num
        q = r[:hq * d][t * hq_l * d:(t + 1) * hq_l * d]
        k = r[hq * d:(hq + hkv) * d][t * hkv_l * d:(t + 1) * hkv_l * d]
        v = r[(hq + hkv) * d:][t * hkv_l * d:(t + 1) * hkv_l * d]
        return torch.cat([q, k, v])
    if "out_proj" in name:
        return r[:, t * hq_l * d:(t + 1) * hq_l * d]
    if "lm_head" in name:
        return r[t * v_l:(t + 1) * v_l]
    return r


def _worker(rank, world, port, q):
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

        cfg = _tiny_cfg()
        tp, ep = 2, 2
        tc = TrainConfig(seq_len=32, micro_batch_size=2, micro_batch_num=1,
                         overlap_grad_reduce=False, tp_size=tp, ep_size=ep,
                         sequence_parallel=True)
        model, opt, red = build_trainer(cfg, tc, "cpu")
        le = cfg.expert_num // ep
        t = rank % tp          # tp rank (tp fastest)
        erank = rank % ep      # within consecutive ep group
        d = rank // tp         # dp replica

        torch.manual_seed(1234)
        ref = LlamaForTraining(cfg, tc.seq_len, device="cpu")
        ref_opt = MixedPrecisionAdam(ref.parameters(), tc)
        with torch.no_grad():
            rd = dict(ref.named_parameters())
            for name, p in model.named_parameters():
                r = rd[name]
                if getattr(p, "_is_expert", False):
                    p.copy_(r[erank * le:(erank + 1) * le])
                else:
                    p.copy_(_shard_dense(name, r, t, cfg, tp))

        dp = world // tp
        toks, labels = make_synthetic_batch(cfg.vocab_size, dp, 2, 32,
                                            "cpu", seed=7)
        opt.zero_grad()
        red.reduce_this_pass = True
        loss = model(toks[d], labels[d])
        loss.backward()
        red.finalize()

        # single-process reference over both dp batches (summed grads)
        ref_opt.zero_grad()
        for mb in range(dp):
            ref(toks[mb], labels[mb]).backward()
            accumulate_main_grads(ref_opt.params)

        errs = {}
        rd = dict(ref.named_parameters())
        for name, p in model.named_parameters():
            r = rd[name]
            if getattr(p, "_is_expert", False):
                want = r.main_grad[erank * le:(erank + 1) * le]
                got = p.main_grad * dp      # edp averaging factor
            else:
                want = _shard_dense(name, r.main_grad, t, cfg, tp)
                got = p.main_grad * dp      # dp averaging factor
            denom = want.abs().max().clamp(min=1e-4)
            errs[name] = float((got - want).abs().max() / denom)
        q.put((rank, float(loss), errs))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(420)
def test_tp2sp_ep2_matches_single_process():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29641
    ps = [ctx.Process(target=_worker, args=(r, 4, port, q)) for r in range(4)]
    for p in ps:
        p.start()
    results = [q.get(timeout=400) for _ in range(4)]
    for p in ps:
        p.join(timeout=60)
    for rank, loss, errs in results:
        assert loss == loss, f"rank {rank} NaN loss"
        bad = {n: e for n, e in errs.items() if e > 0.08}
        assert not bad, f"rank {rank} grad mismatches: {bad}"
