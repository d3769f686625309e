"""tp x cp composition on CPU (gloo, world 4 = tp2(SP) x cp2): Megatron
rank order (tp fastest), cp groups strided by tp, Ulysses a2a on
tp-local heads, dp_cp grad averaging. Gradients must match the
single-process full-sequence run."""

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


def _shard_dense(name, r, t, cfg, tp=2):
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

        cfg = _tiny_cfg()
        S = 64
        tp, cp = 2, 2
        tc = TrainConfig(seq_len=S, micro_batch_size=2, micro_batch_num=1,
                         overlap_grad_reduce=False, tp_size=tp, cp_size=cp,
                         cp_comm_type=cp_comm_type, sequence_parallel=True)
        model, opt, red = build_trainer(cfg, tc, "cpu")
        t = rank % tp                 # tp rank (fastest)
        c = rank // tp                # cp rank

        torch.manual_seed(1234)
        ref = LlamaForTraining(cfg, S, device="cpu")
        MixedPrecisionAdam(ref.parameters(), tc)
        rd = dict(ref.named_parameters())
        with torch.no_grad():
            for name, p in model.named_parameters():
                p.copy_(_shard_dense(name, rd[name], t, cfg, tp))

        toks, labels = make_synthetic_batch(cfg.vocab_size, 1, 2, S,
                                            "cpu", seed=77)
        s_loc = S // cp
        sl = slice(c * s_loc, (c + 1) * s_loc)
        loss = model(toks[0][:, sl], labels[0][:, sl])
        loss.backward()
        accumulate_main_grads([p for p in model.parameters()])
        red.finalize()

        ref_loss = ref(toks[0], labels[0])
        ref_loss.backward()
        accumulate_main_grads([p for p in ref.parameters()])

        lt = loss.detach().clone()
        dist.all_reduce(lt)
        # sum over world = tp * (sum over cp); full-seq loss = cp-mean
        lt /= world
        lerr = (lt - ref_loss.detach()).abs().item()

        gerr = 0.0
        rd = dict(ref.named_parameters())
        for name, p in model.named_parameters():
            want = _shard_dense(name, rd[name].main_grad, t, cfg, tp)
            denom = want.abs().max().clamp(min=1e-6)
            gerr = max(gerr, ((p.main_grad - want).abs().max()
                              / denom).item())
        q.put((rank, lerr, gerr))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(420)
@pytest.mark.parametrize("mode,port", [("a2a", 29651), ("ring", 29654)])
def test_tp2sp_cp2(mode, port):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    ps = [ctx.Process(target=_worker, args=(r, 4, port, q, mode))
          for r in range(4)]
    for p in ps:
        p.start()
    results = [q.get(timeout=400) for _ in range(4)]
    for p in ps:
        p.join(timeout=60)
    for rank, lerr, gerr in results:
        assert lerr < 3e-3, f"rank {rank} loss err {lerr}"
        assert gerr < 0.08, f"rank {rank} grad rel err {gerr}"
