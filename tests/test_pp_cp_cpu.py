"""pp x cp composition on CPU (gloo, world 4 = pp2 x cp2): cp rides
inside each stage's dp block (consecutive ranks), stage peers pair at
rank +- dp preserving the cp coordinate, attention a2a's the tp-local
heads over the cp pair, and the stage-local reducer's dp group performs
the dp_cp average. Gradients must match the single-process
full-sequence run."""

import os
import sys

import pytest
import torch
import torch.multiprocessing as mp

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _tiny_cfg():
    from simumax_amd.core.config import ModelConfig

    return ModelConfig(hidden_size=128, head_num=4, kv_head_num=2,
                       head_size=32, intermediate_size=256, layer_num=2,
                       vocab_size=512, use_swiglu=True)


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
        mbc, S, cp = 2, 64, 2
        tc = TrainConfig(seq_len=S, micro_batch_size=2, micro_batch_num=mbc,
                         overlap_grad_reduce=False, pp_size=2, cp_size=cp,
                         lr=0.0, grad_clip=1e9)
        model, opt, red, ps = build_pp_trainer(cfg, tc, "cpu")
        stage, prev_rank, next_rank = ps.stage, ps.pp_prev, ps.pp_next
        c = rank % cp

        torch.manual_seed(1234)
        ref = LlamaForTraining(cfg, S, device="cpu")
        ref_opt = MixedPrecisionAdam(ref.parameters(), tc)
        lo, hi = stage_layer_range(cfg.layer_num, 2, stage)
        rd = dict(ref.named_parameters())

        def ref_name(name):
            if name.startswith("layers."):
                parts = name.split(".")
                parts[1] = str(int(parts[1]) + lo)
                return ".".join(parts)
            return name

        with torch.no_grad():
            for name, p in model.named_parameters():
                p.copy_(rd[ref_name(name)])

        toks, labels = make_synthetic_batch(cfg.vocab_size, mbc, 2, S,
                                            "cpu", seed=5)
        s_loc = S // cp
        sl = slice(c * s_loc, (c + 1) * s_loc)
        toks_l = toks[:, :, sl].contiguous()
        labels_l = labels[:, :, sl].contiguous()
        hidden_shape = (2, s_loc, cfg.hidden_size)
        loss = pp_train_step(model, opt, red, toks_l, labels_l, mbc,
                             prev_rank, next_rank, hidden_shape,
                             torch.bfloat16)

        ref_opt.zero_grad()
        ref_losses = []
        for m in range(mbc):
            l = ref(toks[m], labels[m])
            l.backward()
            accumulate_main_grads(ref_opt.params)
            ref_losses.append(float(l))

        errs = {}
        rg = {n: p.main_grad for n, p in ref.named_parameters()}
        for name, p in model.named_parameters():
            want = rg[ref_name(name)]
            denom = want.abs().max().clamp(min=1e-4)
            errs[name] = float((p.main_grad - want).abs().max() / denom)
        if stage == 1:
            # per-shard mean losses average to the full-seq mean over cp
            lt = torch.tensor([loss])
            dist.all_reduce(lt, group=ps.dp_group)
            mine = lt.item() / cp
            ref_mean = sum(ref_losses) / mbc
            errs["loss"] = abs(mine - ref_mean) / max(abs(ref_mean), 1e-6)
        q.put((rank, errs))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(420)
def test_pp2_cp2_matches_single_process_gradients():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29662
    ps = [ctx.Process(target=_worker, args=(r, 4, port, q))
          for r in range(4)]
    for p in ps:
        p.start()
    results = [q.get(timeout=400) for _ in range(4)]
    for p in ps:
        p.join(timeout=60)
    for rank, errs in results:
        bad = {n: e for n, e in errs.items() if e > 0.08}
        assert not bad, f"rank {rank} mismatches: {bad}"
