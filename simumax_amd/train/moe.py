"""MoE MLP for the reference trainer (mixtral-style top-k routing with
capacity dispatch), matching the simulator's MoE cost/memory accounting:

* router GEMM -> softmax -> top-k (Router)
* capacity-padded dispatch: cap = ceil(tokens*topk/E * capacity); tokens
  beyond an expert's capacity are dropped (Permutation with
  moe_pad_expert_input_to_capacity)
* per-expert GEMMs as ONE torch.bmm over [E, cap, *] — the exact op the
  group_matmul calibration sweep times (calib/sweeps.sweep_grouped)
* weighted combine scatter (UnPermutation)

Expert parallelism (ep_size > 1): experts are sharded over the EP
process group; capacity padding makes every rank's per-expert
contribution a FIXED size, so dispatch/combine are single equal-split
`all_to_all_single` calls (autograd via _AllToAll below — a2a is its own
adjoint for symmetric equal splits). Expert weights carry `_is_expert`
and their grads reduce over the edp group (DataParallelGradReducer).
Mirrors the simulator's Permutation/UnPermutation comm model
(ops/moe.py).
"""

from __future__ import annotations

import math

import torch
import torch.distributed as dist
import torch.nn as nn




class _PermuteGlue(nn.Module):
    """Dispatch/combine index glue as a module so in-situ calibration can
    event-time BOTH directions via backward hooks; bytes follow the
    simulator's Permutation/UnPermutation accessed-mem conventions
    (tests/test_glue_model.py::test_permutation_traffic_formulas)."""

    def __init__(self, kind):
        super().__init__()
        self.kind = kind          # "dispatch" | "combine"
        self._h = []

    def _bytes(self, in_b, out_b, topk, stage):
        if self.kind == "dispatch":
            if stage == "fwd":
                return 2 * out_b + 3 * max(in_b * topk, out_b)
            return 2 * out_b + 3 * in_b
        if stage == "fwd":
            return 4 * out_b + 3 * in_b
        return 6 * out_b + 2 * in_b

    def arm_insitu(self, in_b, out_b, topk):
        """Register timing hooks once (idempotent)."""
        from ..kernels import insitu

        if self._h or not insitu.ENABLED:
            return
        state = {}

        def fpre(m, inp):
            state["f"] = insitu.start(
                "bw_permute_fwd", str(self._bytes(in_b, out_b, topk, "fwd")))

        def fpost(m, inp, out):
            state["f"]()

        def bpre(m, gout):
            state["b"] = insitu.start(
                "bw_permute_bwd", str(self._bytes(in_b, out_b, topk, "bwd")))

        def bpost(m, gin, gout):
            state["b"]()

        self._h = [self.register_forward_pre_hook(fpre),
                   self.register_forward_hook(fpost),
                   self.register_full_backward_pre_hook(bpre),
                   self.register_full_backward_hook(bpost)]


class _DispatchGlue(_PermuteGlue):
    def __init__(self):
        super().__init__("dispatch")

    def forward(self, xf, src_tok, slot_index, total_slots):
        xp = torch.zeros(total_slots, xf.shape[1], dtype=xf.dtype,
                         device=xf.device)
        xp.index_copy_(0, slot_index, xf.index_select(0, src_tok))
        return xp


class _CombineGlue(_PermuteGlue):
    def __init__(self):
        super().__init__("combine")

    def forward(self, y_flat, src_tok, slot_index, w_kept, n_tokens):
        out = y_flat.new_zeros(n_tokens, y_flat.shape[1])
        out.index_add_(0, src_tok,
                       y_flat.index_select(0, slot_index) * w_kept[:, None])
        return out


class _AllToAll(torch.autograd.Function):
    """Equal-split all_to_all_single with autograd (backward = a2a of the
    incoming grads — exact adjoint for symmetric splits)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        out = torch.empty_like(x)
        dist.all_to_all_single(out, x.contiguous(), group=group)
        return out

    @staticmethod
    def backward(ctx, grad):
        g = torch.empty_like(grad)
        dist.all_to_all_single(g, grad.contiguous(), group=ctx.group)
        return g, None


def all_to_all(x, group):
    return _AllToAll.apply(x, group)


def _grouped_key(ng, m, k, n, stage):
    """The simulator's group_matmul shape key (core/module.py
    GroupLinearBase.get_input_shapes_desc) for the trainer's [E,M,K]@[E,K,N]
    grouped op. NOTE: N/K in the key are the LAYER's output/input sizes."""
    s = (f"ng={ng}, M={m}, N={n}, K={k}, dtype=bf16, out_dtype=bf16, "
         f"main_grad_dtype=fp32")
    if stage == "fwd":
        s += (", stage=fwd, grad=False, accumulate=False, "
              "use_split_accumulator=False, single_output=True")
    elif stage == "bwd_grad_act":
        s += (", stage=bwd_grad_act, grad=True, accumulate=False, "
              "use_split_accumulator=True, single_output=False")
    else:
        s += (", stage=bwd_grad_w, grad=True, accumulate=True, "
              "use_split_accumulator=True, single_output=False")
    return s


# weights live in the natural Linear layout [E, out(N), in(K)].
# Dispatch: the batched MFMA kernels (csrc/grouped_gemm.hip) win for
# many small experts (dgrad 2.6x, wgrad 1.2x vs the loop at E=160);
# per-expert hipBLASLt mm wins for few fat experts and for fwd
# (scripts/grouped_kernel_test.py has the shape study). torch.bmm's
# strided-batched BACKWARD and torch._grouped_mm both memory-fault on
# this stack (scripts/grouped_mm_probe.py), hence no bmm anywhere.
GROUPED_KERNEL_MIN_E = 32


def grouped_fwd_op(x, w):
    """x [E,M,K] @ w[E,N,K]^T -> [E,M,N]"""
    E = x.shape[0]
    out = torch.empty(E, x.shape[1], w.shape[1], dtype=x.dtype,
                      device=x.device)
    for e in range(E):
        torch.mm(x[e], w[e].t(), out=out[e])
    return out


def grouped_dgrad_op(dout, w):
    """dout [E,M,N] @ w[E,N,K] -> dx [E,M,K]"""
    from ..kernels.ops import ext

    E, N, K = w.shape[0], w.shape[1], w.shape[2]
    if (dout.is_cuda and E >= GROUPED_KERNEL_MIN_E
            and K % 128 == 0 and N % 32 == 0):
        return ext().grouped_dgrad(dout, w)
    dx = torch.empty(E, dout.shape[1], K, dtype=dout.dtype,
                     device=dout.device)
    for e in range(E):
        torch.mm(dout[e], w[e], out=dx[e])
    return dx


def grouped_wgrad_op(dout, x, g):
    """g fp32 [E,N,K] += dout[E,M,N]^T @ x[E,M,K]"""
    from ..kernels.ops import ext

    E, N, K = g.shape[0], g.shape[1], g.shape[2]
    if (dout.is_cuda and E >= GROUPED_KERNEL_MIN_E
            and N % 128 == 0 and K % 128 == 0):
        ext().grouped_wgrad(dout, x, g)
        return
    if dout.is_cuda:
        for e in range(E):
            ext().wgrad_accum(dout[e], x[e], g[e])
    else:
        for e in range(E):
            g[e] += dout[e].t().float() @ x[e].float()


class _GroupedLinearFn(torch.autograd.Function):
    """Grouped expert GEMM with fused fp32 wgrad accumulation; see the
    dispatch notes above."""

    @staticmethod
    def forward(ctx, x, w):
        from ..kernels import insitu

        ctx.save_for_backward(x, w)
        timing = insitu.ENABLED and x.is_cuda
        if timing:
            stop = insitu.start("group_matmul", _grouped_key(
                x.shape[0], x.shape[1], w.shape[2], w.shape[1], "fwd"))
        out = grouped_fwd_op(x, w)
        if timing:
            stop()
        return out

    @staticmethod
    def backward(ctx, dout):
        from ..kernels import insitu

        x, w = ctx.saved_tensors
        E = x.shape[0]
        dout = dout.contiguous()
        fused = x.is_cuda and hasattr(w, "main_grad")
        timing = insitu.ENABLED and x.is_cuda
        if timing:
            stop_dx = insitu.start("group_matmul", _grouped_key(
                E, x.shape[1], w.shape[2], w.shape[1], "bwd_grad_act"))
        dx = grouped_dgrad_op(dout, w)
        if timing:
            stop_dx()
            stop_dw = insitu.start("group_matmul", _grouped_key(
                E, x.shape[1], w.shape[2], w.shape[1], "bwd_grad_w"))
        if fused:
            grouped_wgrad_op(dout, x, w.main_grad)
            # fresh unreferenced buffer -> AccumulateGrad steals it (no clone)
            dw = torch.empty_like(w)
        else:
            dw = torch.empty_like(w)
            for e in range(E):
                torch.mm(dout[e].t(), x[e], out=dw[e])
        if timing:
            stop_dw()
        return dx, dw


def grouped_linear(x, w):
    return _GroupedLinearFn.apply(x, w)


class MoEMLP(nn.Module):
    def __init__(self, cfg, dtype=torch.bfloat16, device=None, ep_group=None,
                 ep_size=1):
        super().__init__()
        h = cfg.hidden_size
        self.E = cfg.expert_num
        self.topk = cfg.topk
        self.I = cfg.moe_ffn_hidden_size
        self.capacity = getattr(cfg, "capacity", 1) or 1
        self.ep_group = ep_group
        self.ep = ep_size
        assert self.E % self.ep == 0
        self.le = self.E // self.ep          # local experts
        self.dispatch_glue = _DispatchGlue()
        self.combine_glue = _CombineGlue()
        self.router = nn.Linear(h, self.E, bias=False, dtype=dtype,
                                device=device)
        # LOCAL expert weights in the natural Linear layout [le, out, in]
        self.w1 = nn.Parameter(torch.empty(self.le, 2 * self.I, h, dtype=dtype,
                                           device=device))
        self.w2 = nn.Parameter(torch.empty(self.le, h, self.I, dtype=dtype,
                                           device=device))
        nn.init.normal_(self.w1, std=0.02)
        nn.init.normal_(self.w2, std=0.02)
        self.w1._fused_wgrad = True
        self.w2._fused_wgrad = True
        self.w1._is_expert = True
        self.w2._is_expert = True
        shared_i = getattr(cfg, "moe_shared_expert_intermediate_size", 0)
        self.shared = None
        if shared_i:
            from ..kernels.ops import FusedLinear

            self.shared_fc1 = FusedLinear(h, 2 * shared_i, dtype=dtype,
                                          device=device)
            self.shared_fc2 = FusedLinear(shared_i, h, dtype=dtype,
                                          device=device)
            self.shared = True

    def forward(self, x):
        from ..kernels import ops as K

        B, S, H = x.shape
        N = B * S
        xf = x.reshape(N, H)
        logits = self.router(xf).float()                     # [N, E]
        probs = torch.softmax(logits, dim=-1)
        weight, idx = probs.topk(self.topk, dim=-1)          # [N, k]
        weight = weight / weight.sum(-1, keepdim=True)

        cap = int(math.ceil(N * self.topk / self.E * self.capacity))
        flat_expert = idx.reshape(-1)                        # [N*k]
        flat_token = (torch.arange(N, device=x.device)
                      .repeat_interleave(self.topk))         # [N*k]
        # position of each (token, expert) slot within its expert's queue
        order = torch.argsort(flat_expert, stable=True)
        counts = torch.bincount(flat_expert, minlength=self.E)
        # rank within expert for sorted order
        offs = torch.cumsum(counts, 0) - counts
        rank_sorted = (torch.arange(N * self.topk, device=x.device)
                       - offs[flat_expert[order]])
        keep = rank_sorted < cap                             # capacity drop
        src_tok = flat_token[order][keep]
        dst_exp = flat_expert[order][keep]
        dst_slot = rank_sorted[keep]
        slot_index = dst_exp * cap + dst_slot                # [M]
        w_kept = weight.reshape(-1)[order][keep].to(x.dtype)

        # dispatch: [E*cap, H] padded buffer, ordered by GLOBAL expert
        # (Permutation permute1)
        in_b = N * H * xf.element_size()
        out_b = self.E * cap * H * xf.element_size()
        self.dispatch_glue.arm_insitu(in_b, out_b, self.topk)
        xp = self.dispatch_glue(xf, src_tok, slot_index, self.E * cap)

        if self.ep > 1:
            # EP dispatch a2a: the block for dest rank d = experts
            # [d*le, (d+1)*le) is contiguous; equal splits by capacity.
            recv = all_to_all(xp, self.ep_group)             # [ep*le*cap, H]
            # [src][le][cap] -> expert-major [le, ep*cap, H] (permute2)
            xg = (recv.view(self.ep, self.le, cap, H)
                  .transpose(0, 1).reshape(self.le, self.ep * cap, H)
                  .contiguous())
        else:
            xg = xp.view(self.E, cap, H)

        # grouped GEMMs (dispatch in grouped_*_op) + swiglu
        h1 = grouped_linear(xg, self.w1)                     # [le, M, 2I]
        a = K.swiglu(h1.reshape(-1, 2 * self.I)).reshape(h1.shape[0], -1, self.I)
        y = grouped_linear(a, self.w2)                       # [le, M, H]

        if self.ep > 1:
            # reverse permute2 + combine a2a back to source ranks
            yb = (y.view(self.le, self.ep, cap, H).transpose(0, 1)
                  .reshape(self.ep * self.le * cap, H).contiguous())
            y_flat = all_to_all(yb, self.ep_group)           # [E*cap, H]
        else:
            y_flat = y.reshape(self.E * cap, H)
        self.combine_glue.arm_insitu(in_b, out_b, self.topk)
        out = self.combine_glue(y_flat, src_tok, slot_index, w_kept, N)
        out = out.view(B, S, H)
        if self.shared:
            out = out + self.shared_fc2(
                K.swiglu(self.shared_fc1(x)))
        return out
