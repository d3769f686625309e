"""Training loop for the real-run side of perf-vs-real validation.

Megatron-semantics mixed precision:
* bf16 parameters
* fp32 main_grad buffers (post-accumulate-grad hooks, matching
  use_fp32_accum_grad in the strategy schema)
* fp32 master weights + Adam moments (12 B/param optimizer state)
* DP: flat fp32 grad all_reduce over RCCL, optionally bucket-overlapped
  with backward

The memory/time structure of one step is exactly what PerfLLM predicts
with zero_state=0 for the same strategy config.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.distributed as dist

from ..core.config import ModelConfig
from .model import LlamaForTraining


@dataclass
class TrainConfig:
    seq_len: int = 4096
    micro_batch_size: int = 1
    micro_batch_num: int = 1
    lr: float = 1e-4
    adam_betas = (0.9, 0.95)
    adam_eps: float = 1e-8
    grad_clip: float = 1.0
    overlap_grad_reduce: bool = True
    bucket_bytes: int = 100 * 1024**2
    ep_size: int = 1
    tp_size: int = 1
    pp_size: int = 1
    cp_size: int = 1   # context parallel (seq_len = FULL sequence)
    cp_comm_type: str = "a2a"   # "a2a" (Ulysses, flash path) | "all_gather"
    cp_sharding: str = "contiguous"   # | "zigzag" (all_gather/ring only)
    fp8: bool = False  # e4m3/e5m2 GEMMs via _scaled_mm (decoder linears)
    recompute_layers: int = 0  # full-block activation recompute for the
                               # first N layers (torch.utils.checkpoint;
                               # simulator full_block + recompute_layer_num)
    sequence_parallel: bool = False
    zero_state: int = 0   # 1 = ZeRO-1 distributed optimizer (sharded state)


_EP_GROUPS = {}


def get_ep_groups(ep_size):
    """EP group = ep_size consecutive ranks; edp group = same offset
    strided by ep (order matches core/utils.get_rank_group: ep fastest).
    dist.new_group must be called by every rank in the same order, so
    groups are built for all slices and cached."""
    if ep_size <= 1 or not dist.is_initialized():
        return None, None
    key = (ep_size, dist.get_world_size())
    if key not in _EP_GROUPS:
        world = dist.get_world_size()
        assert world % ep_size == 0
        ep_groups, edp_groups = {}, {}
        for start in range(0, world, ep_size):
            g = dist.new_group(list(range(start, start + ep_size)))
            for r in range(start, start + ep_size):
                ep_groups[r] = g
        for off in range(ep_size):
            ranks = list(range(off, world, ep_size))
            g = dist.new_group(ranks)
            for r in ranks:
                edp_groups[r] = g
        _EP_GROUPS[key] = (ep_groups, edp_groups)
    ep_groups, edp_groups = _EP_GROUPS[key]
    r = dist.get_rank()
    return ep_groups[r], edp_groups[r]


class MixedPrecisionAdam:
    """fp32 master + moments over bf16 params with fp32 main grads.

    Megatron-style FLAT buffers: one contiguous fp32 tensor each for
    main_grad / master / m / v and one flat bf16 tensor backing the model
    parameters (p.data and p.main_grad become views). The whole optimizer
    step is ~8 full-size kernels instead of ~3 small kernels per param —
    matching the calibrated optimizer bandwidth."""

    def __init__(self, params, cfg: TrainConfig, zero_group=None,
                 expert_zero_group=None):
        params = [p for p in params if p.requires_grad]
        # dense params first, expert params last: the flat grad buffer then
        # splits into one contiguous SEGMENT per reduction group; under
        # ZeRO-1 each segment shards over ITS group (dense over dp[_cp],
        # experts over edp — Megatron distributed-optimizer semantics)
        self.params = ([p for p in params if not getattr(p, "_is_expert", False)]
                       + [p for p in params if getattr(p, "_is_expert", False)])
        dense_total = sum(p.numel() for p in self.params
                          if not getattr(p, "_is_expert", False))
        expert_total = sum(p.numel() for p in self.params
                           if getattr(p, "_is_expert", False))
        self.cfg = cfg
        dev = self.params[0].device
        self.zero = (cfg.zero_state == 1 and dist.is_initialized()
                     and (dist.get_world_size(zero_group) > 1
                          or (expert_total and expert_zero_group is not None
                              and dist.get_world_size(expert_zero_group)
                              > 1)))
        self.zero_group = zero_group if self.zero else None
        self.expert_zero_group = expert_zero_group if self.zero else None

        def seg(total, group, world_default):
            """(padded_len, this rank's local slice) for one segment.
            group=None means the WORLD for the dense segment (Megatron
            default) but UNSHARDED for experts (edp may be trivial)."""
            if (not self.zero or total == 0
                    or (group is None and not world_default)):
                return total, slice(0, total)
            zw = dist.get_world_size(group)
            zr = dist.get_rank(group)
            pad = (total + zw - 1) // zw * zw
            return pad, slice(zr * pad // zw, (zr + 1) * pad // zw)

        d_pad, d_sh = seg(dense_total, zero_group, True)
        e_pad, e_sh = seg(expert_total, expert_zero_group, False)
        self.dense_numel = dense_total
        self.expert_offset = d_pad          # experts start after the pad
        self.expert_numel = expert_total
        pad = d_pad + e_pad
        self.flat_param = torch.empty(pad, dtype=self.params[0].dtype,
                                      device=dev)
        self.flat_grad = torch.zeros(pad, dtype=torch.float32, device=dev)
        self.offsets = {}
        off = 0
        for p in self.params:
            if getattr(p, "_is_expert", False) and off < d_pad:
                off = d_pad                 # jump over the dense padding
            n = p.numel()
            self.flat_param[off:off + n].view_as(p).copy_(p.data)
            p.data = self.flat_param[off:off + n].view_as(p)
            p.main_grad = self.flat_grad[off:off + n].view(p.shape)
            self.offsets[id(p)] = (off, off + n)
            off += n
        if self.zero:
            self.flat_param[dense_total:d_pad].zero_()
            self.flat_param[d_pad + expert_total:].zero_()
        # per-segment shard state: (flat slice, master, m, v, group, zw)
        self.segments = []
        for base, (seg_pad, sh), grp in (
            (0, (d_pad, d_sh), self.zero_group),
            (d_pad, (e_pad, e_sh), self.expert_zero_group),
        ):
            if seg_pad == 0:
                continue
            fsl = slice(base + sh.start, base + sh.stop)
            master = self.flat_param[fsl].float()
            sharded = self.zero and (sh.stop - sh.start) < seg_pad
            self.segments.append(dict(
                fsl=fsl, base=base, pad=seg_pad,
                master=master, m=torch.zeros_like(master),
                v=torch.zeros_like(master), group=grp,
                zw=(dist.get_world_size(grp) if sharded else 1)))
        self.t = 0
        self.master_numel = sum(seg["master"].numel()
                                for seg in self.segments)
        # model-parallel grad-norm clipping (Megatron semantics: the clip
        # uses the GLOBAL grad norm): sum shard-unique ||g||^2 over the
        # model-parallel group, count replicated params once
        self._norm_group = None
        self._pp_norm_group = None
        self._replicated_slices = []  # flat slices counted once

    def set_model_parallel_norm(self, group, replicated_flag="__none__",
                                pp_group=None):
        """group: ranks holding DISTINCT shards of THIS stage's params
        (tp or ep groups); replicated_flag marks params replicated across
        it (counted once). pp_group: one rank per pipeline stage — the
        per-stage totals are then summed across stages."""
        self._norm_group = group
        self._pp_norm_group = pp_group
        self._replicated_slices = []
        for p in self.params:
            if getattr(p, replicated_flag, False):
                self._replicated_slices.append(self.offsets[id(p)])

    @staticmethod
    def _sq_sum(t):
        """Temp-free squared sum; chunked because rocBLAS dot is
        int32-indexed (an 8B-param flat grad exceeds 2^31 elements)."""
        total = t.new_zeros(())
        CH = 1 << 30
        for off in range(0, t.numel(), CH):
            sl = t[off:off + CH]
            total += torch.dot(sl, sl)
        return total

    def _global_grad_norm(self):
        # torch.dot computes the squared norm with NO temporary — g.pow(2)
        # would materialize a full fp32 copy of the flat grad (~30 GiB on an
        # 8B model) and move the step's peak-allocated point into the
        # optimizer, which is exactly the fresh-box -6.7% memory anomaly in
        # BENCH_r01.json.
        g = self.flat_grad
        total_sq = self._sq_sum(g)
        if self._norm_group is None and getattr(self, "_pp_norm_group",
                                                None) is None:
            return total_sq.sqrt()
        if self._norm_group is not None:
            rep_sq = g.new_zeros(())
            for lo, hi in self._replicated_slices:
                rep_sq += self._sq_sum(g[lo:hi])
            uni_sq = total_sq - rep_sq
            dist.all_reduce(uni_sq, group=self._norm_group)
            stage_sq = uni_sq + rep_sq
        else:
            stage_sq = total_sq
        if getattr(self, "_pp_norm_group", None) is not None:
            dist.all_reduce(stage_sq, group=self._pp_norm_group)
        return stage_sq.sqrt()

    def zero_grad(self):
        self.flat_grad.zero_()

    @torch.no_grad()
    def step(self):
        """In-place flat Adam; one transient fp32 denom buffer (as
        Megatron's fused optimizer; a foreach_div path would transiently
        allocate 2x the fp32 state = +64 GiB on an 8B model)."""
        self.t += 1
        b1, b2 = self.cfg.adam_betas
        # global grad-norm clip (Megatron clip_grad; global across the
        # model-parallel group when configured)
        norm = self._global_grad_norm()
        scale = self.cfg.grad_clip / (norm + 1e-6)
        bc1 = 1 - b1 ** self.t
        bc2 = 1 - b2 ** self.t
        # fold bias corrections: m/(sqrt(v)/sqrt(bc2)+eps)/bc1
        #   = sqrt(bc2)/bc1 * m/(sqrt(v)+eps*sqrt(bc2))  (exact)
        sqrt_bc2 = bc2 ** 0.5
        eps2 = self.cfg.adam_eps * sqrt_bc2
        step_size = -self.cfg.lr * sqrt_bc2 / bc1
        CHUNK = 1 << 29  # 512M elems = 2 GiB fp32
        for seg in self.segments:
            g = self.flat_grad[seg["fsl"]]
            if scale < 1.0:
                g.mul_(scale)
            m, v, master = seg["m"], seg["v"], seg["master"]
            m.mul_(b1).add_(g, alpha=1 - b1)
            v.mul_(b2).addcmul_(g, g, value=1 - b2)
            # chunked denom: bounds the fp32 sqrt temporary to CHUNK
            # elements (a flat v.sqrt() would transiently allocate the
            # full 4B/param)
            for off in range(0, v.numel(), CHUNK):
                sl = slice(off, min(off + CHUNK, v.numel()))
                denom = v[sl].sqrt().add_(eps2)
                master[sl].addcdiv_(m[sl], denom, value=step_size)
            seg_flat = self.flat_param[seg["base"]:seg["base"] + seg["pad"]]
            if self.zero and seg["zw"] > 1:
                self.flat_param[seg["fsl"]].copy_(master)
                local = self.flat_param[seg["fsl"]].contiguous()
                if dist.get_backend(seg["group"]) == "nccl":
                    dist.all_gather_into_tensor(seg_flat, local,
                                                group=seg["group"])
                else:
                    shards = [torch.empty_like(local)
                              for _ in range(seg["zw"])]
                    dist.all_gather(shards, local, group=seg["group"])
                    seg_flat.copy_(torch.cat(shards))
            else:
                self.flat_param[seg["fsl"]].copy_(master)


class DataParallelGradReducer:
    """Bucketed fp32 main_grad all_reduce over contiguous slices of the
    optimizer's FLAT grad buffer, overlapped with backward (Megatron
    DistributedDataParallel grad-buffer semantics)."""

    def __init__(self, opt, overlap: bool, bucket_bytes: int,
                 edp_group=None, edp_size=1, dp_group=None, dp_size=None,
                 tp_group=None, cp_size=1):
        self.params = opt.params
        self.flat_grad = opt.flat_grad
        self.dense_numel = opt.dense_numel
        # dense grads reduce over the dp group (= world when tp=ep=1);
        # expert grads are replicated only across edp and reduce there
        self.edp_group = edp_group
        self.edp_size = edp_size
        # cp x ep: each rank's loss is a mean over its SEQ SHARD, so raw
        # expert grads carry a cp x token weight; the dense path absorbs
        # it in the world-spanning average, the expert path must divide
        # by edp*cp to keep both on the dp-mean convention
        self.exp_div = edp_size * cp_size
        self.dp_group = dp_group
        self.tp_group = tp_group
        # sequence-parallel norms produce PARTIAL weight grads (each tp
        # rank saw only its seq shard): summed over tp before DP averaging
        self._tp_partial = [p for p in self.params
                            if getattr(p, "_needs_tp_grad_reduce", False)]
        world = dist.get_world_size() if dist.is_initialized() else 1
        self.dp_size = dp_size if dp_size is not None else world
        self.overlap = (overlap and dist.is_initialized()
                        and self.dp_size > 1)
        self.enabled = dist.is_initialized() and self.dp_size > 1
        self.bucket_bytes = bucket_bytes
        self.handles = []
        # Megatron no_sync semantics: only the LAST microbatch's backward
        # triggers the bucketed all_reduce
        self.reduce_this_pass = True
        # param offsets in the flat buffer (the optimizer's layout — the
        # dense segment may be padded under ZeRO, so offsets cannot be
        # recomputed by cumulative numel)
        offs = opt.offsets
        self.offsets = offs
        self.expert_offset = opt.expert_offset
        # Hooks are registered UNCONDITIONALLY: releasing p.grad right after
        # each accumulation is a memory-correctness requirement — without
        # the hook the weight-sized placeholder wgrads pile up in p.grad
        # until backward ends (~25 GiB on llama3-70b-l12).
        # reverse order (grads become ready back-to-front); reversed
        # consecutive params are a contiguous flat slice. Buckets are built
        # PER PARTITION so none straddles the dense/expert boundary.
        buckets = []
        for part in (
            [p for p in self.params if getattr(p, "_is_expert", False)],
            [p for p in self.params if not getattr(p, "_is_expert", False)],
        ):
            cur, cur_bytes = [], 0
            for p in reversed(part):
                cur.append(p)
                cur_bytes += p.numel() * 4
                if cur_bytes >= bucket_bytes:
                    buckets.append(cur)
                    cur, cur_bytes = [], 0
            if cur:
                buckets.append(cur)
        self._hook_handles = []
        for bucket in buckets:
            lo = min(offs[id(p)][0] for p in bucket)
            hi = max(offs[id(p)][1] for p in bucket)
            is_exp = getattr(bucket[0], "_is_expert", False)
            remaining = {id(p) for p in bucket}
            for p in bucket:
                self._hook_handles.append(p.register_post_accumulate_grad_hook(
                    self._make_hook((lo, hi), remaining, is_exp)))
        self._buckets = buckets

    def remove_hooks(self):
        """Detach the post-accumulate hooks. The hook closures reference the
        reducer (and through it every parameter) from C++-side hook storage
        that the cyclic GC cannot traverse — a trainer is only freed after
        calling this."""
        for h in self._hook_handles:
            h.remove()
        self._hook_handles.clear()

    def _make_hook(self, span, remaining, is_expert=False):
        def hook(p):
            if not getattr(p, "_fused_wgrad", False) or not p.is_cuda:
                # add_ casts bf16->fp32 inside the kernel; .float() here
                # would allocate a param-sized fp32 temporary per grad
                p.main_grad.add_(p.grad)
            p.grad = None
            remaining.discard(id(p))
            if not remaining:
                if self.overlap and self.reduce_this_pass:
                    sl = self.flat_grad[span[0]:span[1]]
                    if is_expert:
                        if self.exp_div > 1:
                            sl.div_(self.exp_div)
                        if self.edp_size > 1:
                            self.handles.append(dist.all_reduce(
                                sl, group=self.edp_group, async_op=True))
                    else:
                        sl.div_(self.dp_size)
                        self.handles.append(dist.all_reduce(
                            sl, group=self.dp_group, async_op=True))
                # rearm for the next backward pass
                remaining.update(self._bucket_ids(span))
        return hook

    def _bucket_ids(self, span):
        ids = set()
        for p in self.params:
            lo, hi = self.offsets[id(p)]
            if lo >= span[0] and hi <= span[1]:
                ids.add(id(p))
        return ids

    def finalize(self):
        if self._tp_partial and self.tp_group is not None:
            for p in self._tp_partial:
                dist.all_reduce(p.main_grad, group=self.tp_group)
        if self.overlap:
            for h in self.handles:
                h.wait()
            self.handles.clear()
        elif self.enabled:
            dense = self.flat_grad[:self.dense_numel]
            dense.div_(self.dp_size)
            dist.all_reduce(dense, group=self.dp_group)
            if self.expert_offset < self.flat_grad.numel():
                exp = self.flat_grad[self.expert_offset:]
                if self.exp_div > 1:
                    exp.div_(self.exp_div)
                if self.edp_size > 1:
                    dist.all_reduce(exp, group=self.edp_group)


def accumulate_main_grads(params):
    for p in params:
        if p.grad is not None:
            if not getattr(p, "_fused_wgrad", False) or not p.is_cuda:
                p.main_grad.add_(p.grad)
            p.grad = None


def build_trainer(model_cfg: ModelConfig, cfg: TrainConfig, device="cuda",
                  tp_size: int = 1):
    torch.manual_seed(1234)
    if cfg.ep_size > 1 and cfg.tp_size > 1:
        # tp x ep composition: tp ranks must hold DISTINCT token shards
        # (sequence parallel), so each rank routes its own tokens and the
        # EP a2a carries no duplicates (Megatron etp=1 semantics)
        assert cfg.sequence_parallel, "tp x ep requires sequence_parallel"
    if cfg.cp_size > 1:
        assert cfg.pp_size == 1, "trainer CP does not compose with PP yet"
        # cp x ep: MoE treats cp ranks like dp members with distinct
        # (seq-shard) tokens — the consecutive ep groups span the cp pair
        # and edp averaging covers the dp replicas (Megatron dp_cp-hosted
        # expert parallelism)
        assert cfg.ep_size == 1 or cfg.tp_size == 1, \
            "cp x ep x tp not supported"
    if cfg.zero_state == 1:
        assert cfg.pp_size == 1, \
            "trainer ZeRO-1 composes with DP/TP/EP (not PP yet)"
    tp_size = max(tp_size, cfg.tp_size)
    # Megatron-style vocab padding (keeps CE vocab a GPU-friendly multiple
    # and makes GEMM shape keys match the calibration tables)
    model_cfg.maybe_pad_vocab_size(tp_size)
    ep_group, edp_group = get_ep_groups(cfg.ep_size)
    edp_size = 1
    if cfg.ep_size > 1 and dist.is_initialized():
        edp_size = dist.get_world_size() // cfg.ep_size
    from .cp import get_cp_groups
    from .tp import get_tp_groups

    tp_group, dp_group, tp_rank = get_tp_groups(tp_size)
    cp_group, cp_rank = get_cp_groups(cfg.cp_size, tp_size)
    dp_size = None
    if tp_group is not None:
        # the "dp" reduce group from get_tp_groups spans cp too (dp_cp):
        # cp ranks hold different seq shards of the SAME batch, so their
        # grads average exactly like data parallelism
        dp_size = dist.get_world_size() // tp_size
    # cp ranks average grads with the dp group (dp_cp): the default
    # world-spanning reducer already does that when tp = ep = 1
    model = LlamaForTraining(model_cfg, cfg.seq_len, device=device,
                             ep_group=ep_group, ep_size=cfg.ep_size,
                             tp_group=tp_group, tp_size=tp_size,
                             tp_rank=tp_rank, sp=cfg.sequence_parallel,
                             cp_group=cp_group, cp_rank=cp_rank,
                             cp_size=cfg.cp_size,
                             cp_comm_type=cfg.cp_comm_type,
                             cp_sharding=cfg.cp_sharding, fp8=cfg.fp8,
                             recompute_layers=cfg.recompute_layers)
    # ZeRO-1 shards the fp32 optimizer state over the DATA-parallel group
    # (Megatron distributed optimizer); with tp > 1 that is dp_group, not
    # the world
    # ZeRO-1: dense state shards over the dp(_cp) group; expert state
    # over edp (each ep rank's experts are unique — only their dp
    # replicas share state)
    opt = MixedPrecisionAdam(model.parameters(), cfg,
                             zero_group=dp_group if tp_size > 1 else None,
                             expert_zero_group=edp_group)
    if tp_group is not None:
        # tp shards are unique; norms/embedding are replicated across tp
        for p in opt.params:
            p._replicated_tp = not getattr(p, "_is_tp_shard", False)
        opt.set_model_parallel_norm(tp_group, "_replicated_tp")
    elif ep_group is not None:
        for p in opt.params:
            p._replicated_ep = not getattr(p, "_is_expert", False)
        opt.set_model_parallel_norm(ep_group, "_replicated_ep")
    reducer = DataParallelGradReducer(opt, cfg.overlap_grad_reduce,
                                      cfg.bucket_bytes,
                                      edp_group=edp_group, edp_size=edp_size,
                                      dp_group=dp_group, dp_size=dp_size,
                                      tp_group=tp_group,
                                      cp_size=cfg.cp_size)
    return model, opt, reducer


def train_step(model, opt, reducer, tokens, labels, micro_batch_num=1):
    """One optimizer step = micro_batch_num fwd+bwd + grad reduce + adam."""
    opt.zero_grad()
    total_loss = 0.0
    for mb in range(micro_batch_num):
        reducer.reduce_this_pass = mb == micro_batch_num - 1
        loss = model(tokens[mb], labels[mb])
        loss.backward()   # per-param hooks accumulate + free grads inline
        total_loss += loss.item()
    reducer.finalize()
    opt.step()
    return total_loss / micro_batch_num


def make_synthetic_batch(vocab_size, micro_batch_num, micro_batch_size,
                         seq_len, device, seed=0):
    g = torch.Generator(device="cpu").manual_seed(seed)
    toks = torch.randint(0, vocab_size,
                         (micro_batch_num, micro_batch_size, seq_len),
                         generator=g)
    labels = torch.randint(0, vocab_size,
                           (micro_batch_num, micro_batch_size, seq_len),
                           generator=g)
    return toks.to(device), labels.to(device)
