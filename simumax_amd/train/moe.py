"""MoE MLP for the reference trainer (mixtral-style top-k routing with
capacity dispatch), matching the simulator's MoE cost/memory accounting:

* router GEMM -> softmax -> top-k (Router)
* capacity-padded dispatch: cap = ceil(tokens*topk/E * capacity); tokens
  beyond an expert's capacity are dropped (Permutation with
  moe_pad_expert_input_to_capacity)
* per-expert GEMMs as ONE torch.bmm over [E, cap, *] — the exact op the
  group_matmul calibration sweep times (calib/sweeps.sweep_grouped)
* weighted combine scatter (UnPermutation)

Single-rank experts (EP1); the EP all-to-all path is simulator-side this
round (multi-GPU EP training is a round-2 item).
"""

from __future__ import annotations

import math

import torch
import torch.nn as nn


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
    def __init__(self, cfg, dtype=torch.bfloat16, device=None):
        super().__init__()
        h = cfg.hidden_size
        self.E = cfg.expert_num
        self.topk = cfg.topk
        self.I = cfg.moe_ffn_hidden_size
        self.capacity = getattr(cfg, "capacity", 1) or 1
        self.router = nn.Linear(h, self.E, bias=False, dtype=dtype,
                                device=device)
        # grouped weights in the natural Linear layout [E, out, in]
        self.w1 = nn.Parameter(torch.empty(self.E, 2 * self.I, h, dtype=dtype,
                                           device=device))
        self.w2 = nn.Parameter(torch.empty(self.E, h, self.I, dtype=dtype,
                                           device=device))
        nn.init.normal_(self.w1, std=0.02)
        nn.init.normal_(self.w2, std=0.02)
        self.w1._fused_wgrad = True
        self.w2._fused_wgrad = True
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

        # dispatch: [E*cap, H] padded buffer (Permutation)
        xp = torch.zeros(self.E * cap, H, dtype=x.dtype, device=x.device)
        xp.index_copy_(0, slot_index, xf.index_select(0, src_tok))
        xp = xp.view(self.E, cap, H)

        # grouped GEMMs (per-expert mm loop, the calibrated path) + swiglu
        h1 = grouped_linear(xp, self.w1)                     # [E, cap, 2I]
        a = K.swiglu(h1.reshape(-1, 2 * self.I)).reshape(self.E, cap, self.I)
        y = grouped_linear(a, self.w2)                       # [E, cap, H]

        # combine: weighted scatter back (UnPermutation)
        y_flat = y.reshape(self.E * cap, H)
        out = torch.zeros_like(xf)
        out.index_add_(0, src_tok,
                       y_flat.index_select(0, slot_index) * w_kept[:, None])
        out = out.view(B, S, H)
        if self.shared:
            out = out + self.shared_fc2(
                K.swiglu(self.shared_fc1(x)))
        return out
