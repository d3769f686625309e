"""Megatron-style tensor parallelism for the reference trainer.

Mirrors the simulator's TP cost/memory model (ops/dense.py LinearCol /
LinearRow / ParallelCE): attention heads and MLP intermediate split over
the TP group, one all_reduce after each row-parallel GEMM in forward and
one behind each column-parallel GEMM in backward, lm_head column-split
with vocab-parallel cross entropy (two [rows] fp32 all_reduces — the
`ce` comm model).

Process-group layout matches core/utils.get_rank_group: tp is the
fastest-varying dimension, dp = world / tp. TP and EP are mutually
exclusive in the trainer for now (the simulator supports both).
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from ..kernels.ops import FusedLinear

_TP_GROUPS = {}


def get_tp_groups(tp_size):
    """tp group = tp_size consecutive ranks; dp group = same offset
    strided by tp. Returns (tp_group, dp_group, tp_rank)."""
    if tp_size <= 1 or not dist.is_initialized():
        return None, None, 0
    key = (tp_size, dist.get_world_size())
    if key not in _TP_GROUPS:
        world = dist.get_world_size()
        assert world % tp_size == 0
        tp_groups, dp_groups = {}, {}
        for start in range(0, world, tp_size):
            g = dist.new_group(list(range(start, start + tp_size)))
            for r in range(start, start + tp_size):
                tp_groups[r] = g
        for off in range(tp_size):
            ranks = list(range(off, world, tp_size))
            g = dist.new_group(ranks)
            for r in ranks:
                dp_groups[r] = g
        _TP_GROUPS[key] = (tp_groups, dp_groups)
    tp_groups, dp_groups = _TP_GROUPS[key]
    r = dist.get_rank()
    return tp_groups[r], dp_groups[r], r % tp_size


class _CopyToTP(torch.autograd.Function):
    """Identity forward; all_reduce the gradient over the tp group
    (the f operator in the Megatron paper — placed before every
    column-parallel GEMM)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, grad):
        g = grad.contiguous()
        dist.all_reduce(g, group=ctx.group)
        return g, None


class _ReduceFromTP(torch.autograd.Function):
    """all_reduce forward; identity backward (the g operator — placed
    after every row-parallel GEMM)."""

    @staticmethod
    def forward(ctx, x, group):
        x = x.contiguous()
        dist.all_reduce(x, group=group)
        return x

    @staticmethod
    def backward(ctx, grad):
        return grad, None


def copy_to_tp(x, group):
    return _CopyToTP.apply(x, group) if group is not None else x


def reduce_from_tp(x, group):
    return _ReduceFromTP.apply(x, group) if group is not None else x


class ColumnParallelLinear(torch.nn.Module):
    """out features split over tp; input replicated. Forward is local;
    backward all_reduces the input gradient."""

    def __init__(self, in_features, out_features, tp_group, tp_size,
                 dtype=torch.bfloat16, device=None):
        super().__init__()
        assert out_features % tp_size == 0
        self.tp_group = tp_group
        self.linear = FusedLinear(in_features, out_features // tp_size,
                                  dtype=dtype, device=device)
        self.linear.weight._is_tp_shard = True

    @property
    def weight(self):
        return self.linear.weight

    def forward(self, x):
        return self.linear(copy_to_tp(x, self.tp_group))


class RowParallelLinear(torch.nn.Module):
    """in features split over tp; output all_reduced."""

    def __init__(self, in_features, out_features, tp_group, tp_size,
                 dtype=torch.bfloat16, device=None):
        super().__init__()
        assert in_features % tp_size == 0
        self.tp_group = tp_group
        self.linear = FusedLinear(in_features // tp_size, out_features,
                                  dtype=dtype, device=device)
        self.linear.weight._is_tp_shard = True

    @property
    def weight(self):
        return self.linear.weight

    def forward(self, x):
        return reduce_from_tp(self.linear(x), self.tp_group)


class _VocabParallelCE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels, group, vocab_start):
        lf = logits.float()
        row_max = lf.max(dim=-1).values
        dist.all_reduce(row_max, op=dist.ReduceOp.MAX, group=group)
        shifted = lf - row_max[:, None]
        exp = shifted.exp()
        exp_sum = exp.sum(dim=-1)
        dist.all_reduce(exp_sum, group=group)
        local = labels - vocab_start
        in_shard = (local >= 0) & (local < logits.shape[-1])
        safe = (local.clamp(0, logits.shape[-1] - 1))[:, None]
        target = lf.gather(1, safe).squeeze(1) * in_shard
        dist.all_reduce(target, group=group)
        loss = exp_sum.log() + row_max - target
        ctx.save_for_backward(exp, exp_sum, safe, in_shard)
        ctx.dtype = logits.dtype
        return loss

    @staticmethod
    def backward(ctx, dloss):
        exp, exp_sum, safe, in_shard = ctx.saved_tensors
        p = exp / exp_sum[:, None]
        p.scatter_add_(1, safe, -in_shard.float()[:, None])
        return (p * dloss[:, None]).to(ctx.dtype), None, None, None


def vocab_parallel_ce(logits, labels, tp_group, vocab_start):
    return _VocabParallelCE.apply(logits, labels, tp_group, vocab_start)


# ---------------------------------------------------------------------
# Sequence parallelism (Megatron SP): activations between the TP regions
# are sequence-sharded; the f/g all_reduces become all_gather /
# reduce-scatter pairs. gloo has no reduce_scatter, so the CPU test path
# emulates it with all_reduce + narrow (RCCL uses the native collective).
# ---------------------------------------------------------------------
def _reduce_scatter_seq(x, group):
    """x [B, S, H] -> [B, S/tp, H] (sum of shards)."""
    tp = dist.get_world_size(group)
    r = dist.get_rank(group)
    shard = x.shape[1] // tp
    if dist.get_backend(group) == "nccl":
        xt = x.contiguous().view(x.shape[0], tp, shard, -1)             .transpose(0, 1).contiguous()
        out = torch.empty_like(xt[0])
        dist.reduce_scatter_tensor(out, xt, group=group)
        return out
    x = x.contiguous()
    dist.all_reduce(x, group=group)
    return x[:, r * shard:(r + 1) * shard].contiguous()


def _all_gather_seq(x, group):
    """x [B, S/tp, H] -> [B, S, H]."""
    tp = dist.get_world_size(group)
    parts = [torch.empty_like(x) for _ in range(tp)]
    dist.all_gather(parts, x.contiguous(), group=group)
    return torch.cat(parts, dim=1)


class _GatherSeq(torch.autograd.Function):
    """fwd all_gather over seq; bwd reduce_scatter (the SP g-bar op
    before column-parallel GEMMs)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return _all_gather_seq(x, group)

    @staticmethod
    def backward(ctx, grad):
        return _reduce_scatter_seq(grad, ctx.group), None


class _ScatterSeq(torch.autograd.Function):
    """fwd reduce_scatter over seq; bwd all_gather (after row-parallel
    GEMMs)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return _reduce_scatter_seq(x, group)

    @staticmethod
    def backward(ctx, grad):
        return _all_gather_seq(grad, ctx.group), None


def gather_seq(x, group):
    return _GatherSeq.apply(x, group) if group is not None else x


def scatter_seq(x, group):
    return _ScatterSeq.apply(x, group) if group is not None else x


class _SliceSeq(torch.autograd.Function):
    """Plain seq slice (after the replicated embedding); backward
    all_gathers so every rank reconstructs the full-seq gradient."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        tp = dist.get_world_size(group)
        r = dist.get_rank(group)
        shard = x.shape[1] // tp
        return x[:, r * shard:(r + 1) * shard].contiguous()

    @staticmethod
    def backward(ctx, grad):
        return _all_gather_seq(grad, ctx.group), None


def slice_seq(x, group):
    return _SliceSeq.apply(x, group) if group is not None else x
