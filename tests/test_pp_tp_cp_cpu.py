"""3-way pp x tp x cp on CPU (gloo, world 8 = pp2 x tp2(SP) x cp2):
the full Megatron rank grid — tp fastest, cp strided by tp inside each
stage's block, pp outermost. Gradients must match the single-process
full-sequence run."""

import os
import sys

import pytest
import torch
import torch.multiprocessing as mp

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _tiny_cfg():
    from simumax_amd.core.config import ModelConfig

    return ModelConfig(hidden_size=128, head_num=8, kv_head_num=4,
                       head_size=16, intermediate_size=256, layer_num=2,
                       vocab_size=512, use_swiglu=True)


def _shard(name, r, t, cfg, tp=2):
    d = cfg.head_size
    hq_l, hkv_l = cfg.head_num // tp, cfg.kv_head_num // tp
    v_l = cfg.vocab_size // tp
    if "qkv_proj" in name:
        hq, hkv = cfg.head_num, cfg.kv_head_num
        qq = r[:hq * d][t * hq_l * d:(t + 1) * hq_l * d]
        k = r[hq * d:(hq + hkv) * d][t * hkv_l * d:(t + 1) * hkv_l * d]
        v = r[(hq + hkv) * d:][t * hkv_l * d:(t + 1) * hkv_l * d]
        return torch.cat([qq, k, v])
    if "out_proj" in name:
        return r[:, t * hq_l * d:(t + 1) * hq_l * d]
    if "gate_up" in name or "fc1" in name:
        half = r.shape[0] // 2
        per = half // tp
        return torch.cat([r[t * per:(t + 1) * per],
                          r[half + t * per:half + (t + 1) * per]])
    if "down_proj" in name or "fc2" in name:
        per = r.shape[1] // tp
        return r[:, t * per:(t + 1) * per]
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
        from simumax_amd.train.pp import (build_pp_trainer, pp_train_step,
                                          stage_layer_range)
        from simumax_amd.train.trainer import (MixedPrecisionAdam,
                                               TrainConfig,
                                               accumulate_main_grads,
                                               make_synthetic_batch)

        cfg = _tiny_cfg()
        mbc, S, tp, cp, pp = 2, 64, 2, 2, 2
        tc = TrainConfig(seq_len=S, micro_batch_size=2, micro_batch_num=mbc,
                         overlap_grad_reduce=False, pp_size=pp, tp_size=tp,
                         cp_size=cp, sequence_parallel=False, lr=0.0,
                         grad_clip=1e9)
        model, opt, red, ps = build_pp_trainer(cfg, tc, "cpu")
        stage = ps.stage
        t = rank % tp
        c = (rank // tp) % cp

        torch.manual_seed(1234)
        ref = LlamaForTraining(cfg, S, device="cpu")
        ref_opt = MixedPrecisionAdam(ref.parameters(), tc)
        lo, hi = stage_layer_range(cfg.layer_num, pp, stage)
        rd = dict(ref.named_parameters())

        def ref_name(name):
            if name.startswith("layers."):
                parts = name.split(".")
                parts[1] = str(int(parts[1]) + lo)
                return ".".join(parts)
            return name

        with torch.no_grad():
            for name, p in model.named_parameters():
                p.copy_(_shard(ref_name(name), rd[ref_name(name)], t, cfg))

        toks, labels = make_synthetic_batch(cfg.vocab_size, mbc, 2, S,
                                            "cpu", seed=5)
        s_loc = S // cp
        sl = slice(c * s_loc, (c + 1) * s_loc)
        toks_l = toks[:, :, sl].contiguous()
        labels_l = labels[:, :, sl].contiguous()
        pp_train_step(model, opt, red, toks_l, labels_l, mbc,
                      ps.pp_prev, ps.pp_next, (2, s_loc, cfg.hidden_size),
                      torch.bfloat16)

        ref_opt.zero_grad()
        for m in range(mbc):
            ref(toks[m], labels[m]).backward()
            accumulate_main_grads(ref_opt.params)

        errs = {}
        rg = {n: p.main_grad for n, p in ref.named_parameters()}
        for name, p in model.named_parameters():
            want = _shard(ref_name(name), rg[ref_name(name)], t, cfg)
            denom = want.abs().max().clamp(min=1e-4)
            errs[name] = float((p.main_grad - want).abs().max() / denom)
        q.put((rank, errs))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(480)
def test_pp2_tp2_cp2():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29675
    ps = [ctx.Process(target=_worker, args=(r, 8, port, q))
          for r in range(8)]
    for p in ps:
        p.start()
    results = [q.get(timeout=460) for _ in range(8)]
    for p in ps:
        p.join(timeout=60)
    for rank, errs in results:
        bad = {n: e for n, e in errs.items() if e > 0.08}
        assert not bad, f"rank {rank} mismatches: {bad}"
