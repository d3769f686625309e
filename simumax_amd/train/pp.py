"""Pipeline parallelism (1F1B) for the reference trainer.

Mirrors the simulator's PP model (perf/perf_llm.py schedule_1f1b,
sim/schedule.py): the layer stack is split into pp contiguous stages
(stage 0 holds the embedding, the last stage final-norm + lm_head + CE);
microbatches flow through the classic warmup / steady-1F1B / cooldown
schedule with isend/recv at stage boundaries (non-blocking sends — see
_isend — so the steady state cannot rendezvous-deadlock). Cross-stage
autograd is stitched manually: each stage keeps (input, output) per
in-flight microbatch, backward receives the output grad from the next
stage and sends its input grad to the previous one.

Rank layout matches core/utils.get_rank_group (pp outermost):
stage = rank // dp, peers at rank ± dp. Combine with DP freely; TP/EP
composition is a round-2 item.
"""

from __future__ import annotations

import torch
import torch.distributed as dist
import torch.nn as nn

from ..core.config import ModelConfig
from ..kernels import ops as K
from .model import LlamaDecoderLayer


def stage_layer_range(layer_num, pp, stage):
    assert layer_num % pp == 0, "pp trainer needs layer_num % pp == 0"
    per = layer_num // pp
    return stage * per, (stage + 1) * per


class PipelineStageModel(nn.Module):
    """One PP stage of LlamaForTraining (dense/GQA path), optionally with
    tensor parallelism inside the stage (tp x pp composition)."""

    def __init__(self, cfg: ModelConfig, seq_len: int, stage: int, pp: int,
                 dtype=torch.bfloat16, rope_base=500000.0, device=None,
                 tp_group=None, tp_size=1, tp_rank=0, cp_group=None,
                 cp_rank=0, cp_size=1, cp_comm_type="a2a",
                 cp_sharding="contiguous"):
        super().__init__()
        self.cfg = cfg
        self.stage = stage
        self.pp = pp
        self.tp_group = tp_group
        self.tp_size = tp_size
        self.cp_rank = cp_rank
        self.cp_size = cp_size
        self.cp_sharding = cp_sharding
        lo, hi = stage_layer_range(cfg.layer_num, pp, stage)
        if stage == 0:
            self.embedding = nn.Embedding(cfg.vocab_size, cfg.hidden_size,
                                          dtype=dtype, device=device)
        self.layers = nn.ModuleList(
            [LlamaDecoderLayer(cfg, dtype, device, layer_idx=i,
                               tp_group=tp_group, tp_size=tp_size,
                               cp_group=cp_group, cp_size=cp_size,
                               cp_rank=cp_rank, cp_comm_type=cp_comm_type,
                               cp_sharding=cp_sharding)
             for i in range(lo, hi)])
        if stage == pp - 1:
            assert cfg.vocab_size % tp_size == 0
            self.vocab_local = cfg.vocab_size // tp_size
            self.vocab_start = tp_rank * self.vocab_local
            self.final_norm = K.RMSNorm(cfg.hidden_size, dtype=dtype,
                                        device=device)
            self.lm_head = K.FusedLinear(cfg.hidden_size, self.vocab_local,
                                         dtype=dtype, device=device)
            if tp_size > 1:
                self.lm_head.weight._is_tp_shard = True
        cs = K.build_rope_cache(seq_len, cfg.head_size, base=rope_base,
                                device=device or "cpu")
        self.register_buffer("rope_cs", cs, persistent=False)

    def forward(self, x, labels=None):
        # x: tokens [B, S] on stage 0, hidden [B, S, H] elsewhere
        if self.stage == 0:
            B, S = x.shape
            x = self.embedding(x)
        else:
            B, S, _ = x.shape
        # global positions of the local (CP) seq shard
        if self.cp_size > 1:
            from .cp import cp_positions

            pos = cp_positions(S * self.cp_size, self.cp_size,
                               self.cp_rank,
                               self.cp_sharding == "zigzag",
                               x.device).repeat(B)
        else:
            pos = (torch.arange(S, device=x.device,
                                dtype=torch.int32).repeat(B))
        for layer in self.layers:
            x = layer(x, self.rope_cs, pos)
        if self.stage == self.pp - 1:
            x = self.final_norm(x)
            if self.tp_size > 1:
                from .tp import copy_to_tp, vocab_parallel_ce

                logits = self.lm_head(copy_to_tp(x, self.tp_group))
                loss = vocab_parallel_ce(logits.reshape(B * S, -1),
                                         labels.reshape(-1), self.tp_group,
                                         self.vocab_start)
            else:
                logits = self.lm_head(x)
                loss = K.fused_cross_entropy(
                    logits.reshape(B * S, -1), labels.reshape(-1))
            return loss.mean()
        return x


def _isend(t, dst, pending):
    """Non-blocking send (blocking rendezvous sends deadlock the 1F1B
    steady state: a stage can sit in send-fwd while its peer sits in
    send-grad). The tensor must stay alive until the wait."""
    t = t.contiguous()
    pending.append((dist.isend(t, dst), t))


def _recv(shape, dtype, src, device):
    t = torch.empty(shape, dtype=dtype, device=device)
    dist.recv(t, src)
    return t


def pp_train_step(model: PipelineStageModel, opt, reducer, toks, labels,
                  mbc, prev_rank, next_rank, hidden_shape, dtype):
    """One optimizer step of 1F1B over mbc microbatches.

    toks/labels: [mbc, B, S] (every rank gets the full set; stage 0 reads
    toks, the last stage labels). Returns the mean loss on the last stage
    (0.0 elsewhere).
    """
    stage, pp = model.stage, model.pp
    dev = toks.device
    opt.zero_grad()

    warmup = min(pp - stage - 1, mbc)
    in_flight = []           # (input_leaf_or_None, output)
    pending = []             # outstanding isends (handle, tensor)
    losses = []
    nf = nb = 0

    def fwd_one():
        nonlocal nf
        m = nf
        if stage == 0:
            inp = None
            out = model(toks[m])
        else:
            h = _recv(hidden_shape, dtype, prev_rank, dev)
            inp = h.requires_grad_(True)
            out = model(inp, labels[m] if stage == pp - 1 else None)
        if stage == pp - 1:
            losses.append(out)
        else:
            _isend(out.detach(), next_rank, pending)
        in_flight.append((inp, out))
        nf += 1

    def bwd_one(last):
        nonlocal nb
        reducer.reduce_this_pass = last
        inp, out = in_flight.pop(0)
        if stage == pp - 1:
            out.backward()
        else:
            g = _recv(out.shape, dtype, next_rank, dev)
            out.backward(gradient=g)
        if stage > 0:
            _isend(inp.grad, prev_rank, pending)
        nb += 1

    for _ in range(warmup):
        fwd_one()
    while nb < mbc:
        if nf < mbc:
            fwd_one()
        bwd_one(last=(nb == mbc - 1))
    for h, _t in pending:
        h.wait()
    reducer.finalize()
    opt.step()
    if losses:
        return float(torch.stack([l.detach() for l in losses]).mean())
    return 0.0


def build_pp_trainer(model_cfg: ModelConfig, cfg, device="cpu"):
    """PP (optionally x TP) analog of train.trainer.build_trainer
    (pp = cfg.pp_size, tp = cfg.tp_size). Returns
    (model, opt, reducer, parallel_state)."""
    from .parallel_state import init_parallel_state
    from .trainer import DataParallelGradReducer, MixedPrecisionAdam

    torch.manual_seed(1234)
    model_cfg.maybe_pad_vocab_size(cfg.tp_size)
    ps = init_parallel_state(tp_size=cfg.tp_size, pp_size=cfg.pp_size)
    cp_group, cp_rank = None, 0
    if getattr(cfg, "cp_size", 1) > 1:
        # pp x (tp x) cp: cp rides inside each stage's dp block, strided
        # by tp (Megatron tp-cp-dp order); stage blocks are multiples of
        # tp*cp so the groups never straddle a stage, and the stage-peer
        # pairing at rank +- tp*cp*dp preserves the (tp, cp) coordinate
        from .cp import get_cp_groups

        cp_group, cp_rank = get_cp_groups(cfg.cp_size, cfg.tp_size)
    model = PipelineStageModel(model_cfg, cfg.seq_len, ps.stage, cfg.pp_size,
                               device=device, tp_group=ps.tp_group,
                               tp_size=cfg.tp_size, tp_rank=ps.tp_rank,
                               cp_group=cp_group, cp_rank=cp_rank,
                               cp_size=getattr(cfg, "cp_size", 1),
                               cp_comm_type=getattr(cfg, "cp_comm_type",
                                                    "a2a"),
                               cp_sharding=getattr(cfg, "cp_sharding",
                                                   "contiguous"))
    # ZeRO-1 under PP: each stage's fp32 state shards over the stage's
    # own dp replicas (never across stages — they hold different params)
    opt = MixedPrecisionAdam(model.parameters(), cfg,
                             zero_group=ps.dp_group)
    if ps.tp_group is not None or ps.pp_norm_group is not None:
        for p in opt.params:
            p._replicated_tp = not getattr(p, "_is_tp_shard", False)
        opt.set_model_parallel_norm(ps.tp_group, "_replicated_tp",
                                    pp_group=ps.pp_norm_group)
    reducer = DataParallelGradReducer(opt, cfg.overlap_grad_reduce,
                                      cfg.bucket_bytes,
                                      dp_group=ps.dp_group,
                                      dp_size=ps.dp_size,
                                      tp_group=ps.tp_group)
    return model, opt, reducer, ps


# ---------------------------------------------------------------------
# Interleaved VPP (virtual pipeline) training — Megatron sync-VPP
# semantics mirrored from perf/vpp.py's schedule helpers (the same
# chunk_id_of / mb_id_of table the analytic scheduler and the event
# simulator replay).
# ---------------------------------------------------------------------
class VppChunkModel(nn.Module):
    """One VIRTUAL stage v = chunk*pp + stage: a slice of the layer
    stack, embedding on v==0, head on v==nv-1."""

    def __init__(self, cfg: ModelConfig, seq_len: int, v: int, nv: int,
                 dtype=torch.bfloat16, rope_base=500000.0, device=None):
        super().__init__()
        assert cfg.layer_num % nv == 0, "vpp needs layer_num % (pp*vp) == 0"
        per = cfg.layer_num // nv
        self.v = v
        self.nv = nv
        if v == 0:
            self.embedding = nn.Embedding(cfg.vocab_size, cfg.hidden_size,
                                          dtype=dtype, device=device)
        self.layers = nn.ModuleList(
            [LlamaDecoderLayer(cfg, dtype, device, layer_idx=i)
             for i in range(v * per, (v + 1) * per)])
        if v == nv - 1:
            self.final_norm = K.RMSNorm(cfg.hidden_size, dtype=dtype,
                                        device=device)
            self.lm_head = K.FusedLinear(cfg.hidden_size, cfg.vocab_size,
                                         dtype=dtype, device=device)
        cs = K.build_rope_cache(seq_len, cfg.head_size, base=rope_base,
                                device=device or "cpu")
        self.register_buffer("rope_cs", cs, persistent=False)

    def forward(self, x, labels=None):
        if self.v == 0:
            B, S = x.shape
            x = self.embedding(x)
        else:
            B, S, _ = x.shape
        pos = (torch.arange(S, device=x.device, dtype=torch.int32).repeat(B))
        for layer in self.layers:
            x = layer(x, self.rope_cs, pos)
        if self.v == self.nv - 1:
            x = self.final_norm(x)
            logits = self.lm_head(x)
            loss = K.fused_cross_entropy(
                logits.reshape(B * S, -1), labels.reshape(-1))
            return loss.mean()
        return x


class VppStageModel(nn.Module):
    """All vp chunks living on one pipeline stage."""

    def __init__(self, cfg: ModelConfig, seq_len: int, stage: int, pp: int,
                 vp: int, dtype=torch.bfloat16, device=None):
        super().__init__()
        self.stage = stage
        self.pp = pp
        self.vp = vp
        self.nv = pp * vp
        self.chunks = nn.ModuleList(
            [VppChunkModel(cfg, seq_len, c * pp + stage, self.nv,
                           dtype=dtype, device=device) for c in range(vp)])


def vpp_train_step(model: VppStageModel, opt, reducer, toks, labels, mbc,
                   ps, hidden_shape, dtype):
    """One optimizer step of the Megatron interleaved schedule
    (perf/vpp.py stream: warmup (pp-stage-1)*2 + (vp-1)*pp, then 1F1B
    over k-indices decoded by chunk_id_of/mb_id_of). Virtual-stage
    boundaries v->v+1 cross to the next pipeline stage for the same
    chunk, wrapping from the last stage to stage 0 of the next chunk."""
    from ..perf.vpp import chunk_id_of, mb_id_of

    stage, pp, vp, nv = model.stage, model.pp, model.vp, model.nv
    assert mbc % pp == 0, "interleaved schedule requires mbc % pp == 0"
    dev = toks.device
    stage_span = (dist.get_world_size() // pp if dist.is_initialized() else 1)
    r = dist.get_rank() if dist.is_initialized() else 0

    def rank_of_stage(s):
        return s * stage_span + (r % stage_span)

    opt.zero_grad()
    total = mbc * vp
    warm = min((pp - stage - 1) * 2 + (vp - 1) * pp, total)
    inflight = {}            # (chunk, mb) -> (input_leaf, output)
    pending = []
    losses = []
    nf = nb = 0

    def fwd_one():
        nonlocal nf
        k = nf
        c = chunk_id_of(k, pp, vp, True)
        m = mb_id_of(k, pp, vp)
        v = c * pp + stage
        if v == 0:
            inp = None
            out = model.chunks[c](toks[m])
        else:
            src = rank_of_stage(stage - 1 if stage > 0 else pp - 1)
            h = _recv(hidden_shape, dtype, src, dev)
            inp = h.requires_grad_(True)
            out = model.chunks[c](inp, labels[m] if v == nv - 1 else None)
        if v == nv - 1:
            losses.append(out)
        else:
            dst = rank_of_stage(stage + 1 if stage < pp - 1 else 0)
            _isend(out.detach(), dst, pending)
        inflight[(c, m)] = (inp, out)
        nf += 1

    def bwd_one(last):
        nonlocal nb
        k = nb
        c = chunk_id_of(k, pp, vp, False)
        m = mb_id_of(k, pp, vp)
        v = c * pp + stage
        reducer.reduce_this_pass = last
        inp, out = inflight.pop((c, m))
        if v == nv - 1:
            out.backward()
        else:
            src = rank_of_stage(stage + 1 if stage < pp - 1 else 0)
            g = _recv(out.shape, dtype, src, dev)
            out.backward(gradient=g)
        if v > 0:
            dst = rank_of_stage(stage - 1 if stage > 0 else pp - 1)
            _isend(inp.grad, dst, pending)
        nb += 1

    for _ in range(warm):
        fwd_one()
    while nb < total:
        if nf < total:
            fwd_one()
        bwd_one(last=(nb == total - 1))
    for h, _t in pending:
        h.wait()
    reducer.finalize()
    opt.step()
    if losses:
        return float(torch.stack([l.detach() for l in losses]).mean())
    return 0.0


def build_vpp_trainer(model_cfg: ModelConfig, cfg, vp: int, device="cpu"):
    """Interleaved-VPP analog of build_pp_trainer (tp inside stages is a
    round-2 composition)."""
    from .parallel_state import init_parallel_state
    from .trainer import DataParallelGradReducer, MixedPrecisionAdam

    torch.manual_seed(1234)
    model_cfg.maybe_pad_vocab_size(1)
    ps = init_parallel_state(tp_size=1, pp_size=cfg.pp_size)
    model = VppStageModel(model_cfg, cfg.seq_len, ps.stage, cfg.pp_size, vp,
                          device=device)
    opt = MixedPrecisionAdam(model.parameters(), cfg)
    if ps.pp_norm_group is not None:
        opt.set_model_parallel_norm(None, pp_group=ps.pp_norm_group)
    reducer = DataParallelGradReducer(opt, cfg.overlap_grad_reduce,
                                      cfg.bucket_bytes,
                                      dp_group=ps.dp_group,
                                      dp_size=ps.dp_size)
    return model, opt, reducer, ps
