"""Pipeline-parallel trainer equivalence on CPU (gloo, world 2, PP2):
the 1F1B schedule with manual cross-stage autograd stitching must
reproduce the single-process gradients exactly (activations/grads cross
the wire in their native dtype, so the math is identical)."""

import os
import sys

import pytest
import torch
import torch.multiprocessing as mp

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _tiny_cfg():
    from simumax_amd.core.config import ModelConfig

    return ModelConfig(hidden_size=128, head_num=4, kv_head_num=2,
                       head_size=32, intermediate_size=256, layer_num=4,
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
        mbc = 3
        # grad_clip disabled: pp_train_step runs the optimizer (which
        # clips in place) while the reference grads are read unclipped
        tc = TrainConfig(seq_len=32, micro_batch_size=2, micro_batch_num=mbc,
                         overlap_grad_reduce=False, pp_size=2, lr=0.0,
                         grad_clip=1e9)
        model, opt, red, ps = build_pp_trainer(cfg, tc, "cpu")
        stage, prev_rank, next_rank = ps.stage, ps.pp_prev, ps.pp_next

        torch.manual_seed(1234)
        ref = LlamaForTraining(cfg, tc.seq_len, device="cpu")
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

        toks, labels = make_synthetic_batch(cfg.vocab_size, mbc, 2, 32,
                                            "cpu", seed=5)
        hidden_shape = (2, 32, cfg.hidden_size)
        loss = pp_train_step(model, opt, red, toks, labels, mbc,
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
            errs["loss"] = abs(loss - sum(ref_losses) / mbc) / max(
                abs(sum(ref_losses) / mbc), 1e-6)
        # the model-parallel global grad norm (summed across stages) must
        # equal the single-process full-model norm — this is what the
        # in-place Megatron-style clip uses
        gn = float(opt._global_grad_norm())
        ref_gn = float(ref_opt.flat_grad.norm())
        errs["global_grad_norm"] = abs(gn - ref_gn) / max(ref_gn, 1e-6)
        q.put((rank, errs))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_pp2_matches_single_process_gradients():
    mp.set_start_method("spawn", force=True)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, 29523, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, errs = q.get(timeout=240)
        results[rank] = errs
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    for rank, errs in results.items():
        bad = {n: e for n, e in errs.items() if e > 1e-3}
        assert not bad, f"rank {rank} mismatches: {bad}"


def _tp_shard_name_map(cfg, tp, rank_tp, lo):
    """Return fn(name, ref_tensor) -> sharded view for tp x pp."""
    d = cfg.head_size
    hq, hkv = cfg.head_num, cfg.kv_head_num
    hq_l, hkv_l = hq // tp, hkv // tp
    i_l = cfg.intermediate_size // tp
    v_l = cfg.vocab_size // tp
    I = cfg.intermediate_size

    def shard(name, r):
        if "qkv_proj" in name:
            return torch.cat([
                r[:hq * d][rank_tp * hq_l * d:(rank_tp + 1) * hq_l * d],
                r[hq * d:(hq + hkv) * d][rank_tp * hkv_l * d:(rank_tp + 1) * hkv_l * d],
                r[(hq + hkv) * d:][rank_tp * hkv_l * d:(rank_tp + 1) * hkv_l * d]])
        if "out_proj" in name:
            return r[:, rank_tp * hq_l * d:(rank_tp + 1) * hq_l * d]
        if "fc1" in name:
            return torch.cat([r[:I][rank_tp * i_l:(rank_tp + 1) * i_l],
                              r[I:][rank_tp * i_l:(rank_tp + 1) * i_l]])
        if "fc2" in name:
            return r[:, rank_tp * i_l:(rank_tp + 1) * i_l]
        if "lm_head" in name:
            return r[rank_tp * v_l:(rank_tp + 1) * v_l]
        return r

    def ref_name(name):
        if name.startswith("layers."):
            parts = name.split(".")
            parts[1] = str(int(parts[1]) + lo)
            return ".".join(parts)
        return name

    return shard, ref_name


def _worker_tp_pp(rank, world, port, q):
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
        mbc = 2
        tc = TrainConfig(seq_len=32, micro_batch_size=2, micro_batch_num=mbc,
                         overlap_grad_reduce=False, pp_size=2, tp_size=2,
                         lr=0.0, grad_clip=1e9)
        model, opt, red, ps = build_pp_trainer(cfg, tc, "cpu")
        lo, hi = stage_layer_range(cfg.layer_num, 2, ps.stage)
        shard, ref_name = _tp_shard_name_map(cfg, 2, ps.tp_rank, lo)

        torch.manual_seed(1234)
        ref = LlamaForTraining(cfg, tc.seq_len, device="cpu")
        ref_opt = MixedPrecisionAdam(ref.parameters(), tc)
        rd = dict(ref.named_parameters())
        with torch.no_grad():
            for name, p in model.named_parameters():
                p.copy_(shard(name, rd[ref_name(name)]))

        toks, labels = make_synthetic_batch(cfg.vocab_size, mbc, 2, 32,
                                            "cpu", seed=9)
        loss = pp_train_step(model, opt, red, toks, labels, mbc,
                             ps.pp_prev, ps.pp_next,
                             (2, 32, cfg.hidden_size), torch.bfloat16)

        ref_opt.zero_grad()
        for m in range(mbc):
            ref(toks[m], labels[m]).backward()
            accumulate_main_grads(ref_opt.params)

        rg = {n: p.main_grad for n, p in ref.named_parameters()}
        errs = {}
        for name, p in model.named_parameters():
            want = shard(name, rg[ref_name(name)])
            denom = want.abs().max().clamp(min=1e-4)
            errs[name] = float((p.main_grad - want).abs().max() / denom)
        gn = float(opt._global_grad_norm())
        ref_gn = float(ref_opt.flat_grad.norm())
        errs["global_grad_norm"] = abs(gn - ref_gn) / max(ref_gn, 1e-6)
        q.put((rank, errs))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(400)
def test_tp2_pp2_matches_single_process_gradients():
    """Composed tp2 x pp2 on 4 ranks: sharded + staged grads and the
    two-level global grad norm must match the single-process run."""
    mp.set_start_method("spawn", force=True)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_tp_pp, args=(r, 4, 29527, q))
             for r in range(4)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(4):
        rank, errs = q.get(timeout=300)
        results[rank] = errs
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    for rank, errs in results.items():
        bad = {n: e for n, e in errs.items() if e > 4e-2}
        assert not bad, f"rank {rank} mismatches: {bad}"


def _worker_pp_dp(rank, world, port, q):
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
        mbc = 2
        tc = TrainConfig(seq_len=32, micro_batch_size=2, micro_batch_num=mbc,
                         overlap_grad_reduce=False, pp_size=2, lr=0.0,
                         grad_clip=1e9)
        model, opt, red, ps = build_pp_trainer(cfg, tc, "cpu")
        assert ps.dp_size == 2
        lo, hi = stage_layer_range(cfg.layer_num, 2, ps.stage)

        torch.manual_seed(1234)
        ref = LlamaForTraining(cfg, tc.seq_len, device="cpu")
        ref_opt = MixedPrecisionAdam(ref.parameters(), tc)
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

        # 2 dp columns x mbc microbatches of distinct data
        toks, labels = make_synthetic_batch(cfg.vocab_size, 2 * mbc, 2, 32,
                                            "cpu", seed=13)
        my_toks = toks[ps.dp_rank * mbc:(ps.dp_rank + 1) * mbc]
        my_labels = labels[ps.dp_rank * mbc:(ps.dp_rank + 1) * mbc]
        pp_train_step(model, opt, red, my_toks, my_labels, mbc,
                      ps.pp_prev, ps.pp_next, (2, 32, cfg.hidden_size),
                      torch.bfloat16)

        # reference: ALL 2*mbc microbatches, then compare against the
        # dp-averaged (1/2) distributed grads
        ref_opt.zero_grad()
        for m in range(2 * mbc):
            ref(toks[m], labels[m]).backward()
            accumulate_main_grads(ref_opt.params)

        rg = {n: p.main_grad for n, p in ref.named_parameters()}
        errs = {}
        for name, p in model.named_parameters():
            want = rg[ref_name(name)]
            got = p.main_grad * 2  # dp averaging factor
            denom = want.abs().max().clamp(min=1e-4)
            errs[name] = float((got - want).abs().max() / denom)
        q.put((rank, errs))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(400)
def test_pp2_dp2_data_sharding():
    """pp2 x dp2 on 4 ranks: per-column data shards, dense grads
    dp-averaged within each stage."""
    mp.set_start_method("spawn", force=True)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_pp_dp, args=(r, 4, 29529, q))
             for r in range(4)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(4):
        rank, errs = q.get(timeout=300)
        results[rank] = errs
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    for rank, errs in results.items():
        bad = {n: e for n, e in errs.items() if e > 4e-2}
        assert not bad, f"rank {rank} mismatches: {bad}"


def _worker_vpp(rank, world, port, q, mbc=2):
    import torch.distributed as dist

    sys.path.insert(0, REPO)
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from simumax_amd.train.model import LlamaForTraining
        from simumax_amd.train.pp import build_vpp_trainer, vpp_train_step
        from simumax_amd.train.trainer import (MixedPrecisionAdam,
                                               TrainConfig,
                                               accumulate_main_grads,
                                               make_synthetic_batch)

        cfg = _tiny_cfg()          # 4 layers -> nv=4 with pp2 x vp2
        pp, vp = 2, 2
        tc = TrainConfig(seq_len=32, micro_batch_size=2, micro_batch_num=mbc,
                         overlap_grad_reduce=False, pp_size=pp, lr=0.0,
                         grad_clip=1e9)
        model, opt, red, ps = build_vpp_trainer(cfg, tc, vp, "cpu")

        torch.manual_seed(1234)
        ref = LlamaForTraining(cfg, tc.seq_len, device="cpu")
        ref_opt = MixedPrecisionAdam(ref.parameters(), tc)
        rd = dict(ref.named_parameters())

        def ref_name(name):
            # chunks.<c>.layers.<i>.rest -> layers.<v>.rest (per=1)
            parts = name.split(".")
            c = int(parts[1])
            v = c * pp + ps.stage
            if parts[2] == "layers":
                rest = ".".join(parts[4:])
                return f"layers.{v}.{rest}"
            return ".".join(parts[2:])

        with torch.no_grad():
            for name, p in model.named_parameters():
                p.copy_(rd[ref_name(name)])

        toks, labels = make_synthetic_batch(cfg.vocab_size, mbc, 2, 32,
                                            "cpu", seed=17)
        loss = vpp_train_step(model, opt, red, toks, labels, mbc, ps,
                              (2, 32, cfg.hidden_size), torch.bfloat16)

        ref_opt.zero_grad()
        ref_losses = []
        for m in range(mbc):
            l = ref(toks[m], labels[m])
            l.backward()
            accumulate_main_grads(ref_opt.params)
            ref_losses.append(float(l.detach()))

        rg = {n: p.main_grad for n, p in ref.named_parameters()}
        errs = {}
        for name, p in model.named_parameters():
            want = rg[ref_name(name)]
            denom = want.abs().max().clamp(min=1e-4)
            errs[name] = float((p.main_grad - want).abs().max() / denom)
        if ps.stage == pp - 1:
            mref = sum(ref_losses) / mbc
            errs["loss"] = abs(loss - mref) / max(abs(mref), 1e-6)
        q.put((rank, errs))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(400)
@pytest.mark.parametrize("mbc,port", [(2, 29531), (4, 29533)])
def test_vpp2_matches_single_process_gradients(mbc, port):
    """Interleaved VPP (pp2 x vp2, 4 virtual stages on 2 ranks): the
    Megatron schedule-table replay must reproduce the single-process
    gradients exactly — mbc=4 exercises the steady-state channel
    orderings beyond the warmup-dominated mbc=2 case."""
    mp.set_start_method("spawn", force=True)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_vpp, args=(r, 2, port, q, mbc))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, errs = q.get(timeout=300)
        results[rank] = errs
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    for rank, errs in results.items():
        bad = {n: e for n, e in errs.items() if e > 1e-3}
        assert not bad, f"rank {rank} mismatches: {bad}"


def _worker_pp_zero(rank, world, port, q):
    import torch.distributed as dist

    sys.path.insert(0, REPO)
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from simumax_amd.train.pp import build_pp_trainer, pp_train_step
        from simumax_amd.train.trainer import (TrainConfig,
                                               make_synthetic_batch)

        cfg = _tiny_cfg()
        mbc = 2
        results = {}
        for zero in (0, 1):
            tc = TrainConfig(seq_len=32, micro_batch_size=2,
                             micro_batch_num=mbc, overlap_grad_reduce=False,
                             pp_size=2, zero_state=zero, grad_clip=1e9)
            model, opt, red, ps = build_pp_trainer(cfg, tc, "cpu")
            if zero == 1:
                # sharded over the stage's dp pair, never across stages
                assert opt.master_numel * 2 >= opt.flat_param.numel()
                assert opt.master_numel < opt.flat_param.numel()
            toks, labels = make_synthetic_batch(cfg.vocab_size, 2 * mbc, 2,
                                                32, "cpu", seed=13)
            my_toks = toks[ps.dp_rank * mbc:(ps.dp_rank + 1) * mbc]
            my_labels = labels[ps.dp_rank * mbc:(ps.dp_rank + 1) * mbc]
            for _ in range(2):
                pp_train_step(model, opt, red, my_toks, my_labels, mbc,
                              ps.pp_prev, ps.pp_next,
                              (2, 32, cfg.hidden_size), torch.bfloat16)
            results[zero] = opt.flat_param.detach().float().clone()
            red.remove_hooks()
            del model, opt, red
        err = (results[0] - results[1]).abs().max().item()
        q.put((rank, err))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(420)
def test_zero1_with_pp2_dp2_matches_zero0():
    """ZeRO-1 x PP on world 4 (pp2 x dp2): per-stage optimizer state
    shards over the stage's dp pair and must reproduce the replicated
    parameters after 2 steps."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29671
    ps = [ctx.Process(target=_worker_pp_zero, args=(r, 4, port, q))
          for r in range(4)]
    for p in ps:
        p.start()
    results = [q.get(timeout=400) for _ in range(4)]
    for p in ps:
        p.join(timeout=60)
    for rank, err in results:
        assert err < 2e-3, f"rank {rank} param drift {err}"
