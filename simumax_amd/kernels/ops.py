"""PyTorch autograd wrappers over the gfx950 HIP kernels.

Policy: on a GPU box the HIP extension MUST be present — ops raise if it
is missing (no silent eager fallback). On CPU (unit tests, shape checks)
pure-torch reference implementations run instead; GPU numerics tests
compare the HIP kernels against these references in fp32.
"""

from __future__ import annotations

import importlib.util
import os

import torch

from . import insitu

_EXT = None
_TRIED = False


def _load_ext():
    global _EXT, _TRIED
    if _TRIED:
        return _EXT
    _TRIED = True
    so = os.path.join(os.path.dirname(__file__), "simumax_hip.so")
    if os.path.exists(so):
        spec = importlib.util.spec_from_file_location("simumax_hip", so)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        _EXT = mod
    return _EXT


def ext():
    e = _load_ext()
    if e is None and torch.cuda.is_available():
        raise RuntimeError(
            "simumax_hip.so missing on a GPU machine — build it with "
            "`python simumax_amd/kernels/build.py` (no eager fallback on GPU)"
        )
    return e


def _use_hip(*tensors):
    if tensors[0].is_cuda:
        ext()  # raises if missing
        return True
    return False


# --------------------------------------------------------------------------
# RMSNorm
# --------------------------------------------------------------------------
class _RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        x = x.contiguous()  # saved for backward; the kernel requires it
        if _use_hip(x):
            if insitu.ENABLED:
                stop = insitu.start("bw_rmsnorm_fwd", str(2 * x.numel() * x.element_size()))
                y, rstd = ext().rmsnorm_fwd(x.contiguous(), weight.contiguous(), eps)
                stop()
            else:
                y, rstd = ext().rmsnorm_fwd(x.contiguous(), weight.contiguous(), eps)
        else:
            xf = x.float()
            rstd = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
            y = (xf * rstd * weight.float()).to(x.dtype)
            rstd = rstd.squeeze(-1).reshape(-1)
        ctx.save_for_backward(x, weight, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, rstd = ctx.saved_tensors
        if _use_hip(x):
            if insitu.ENABLED:
                stop = insitu.start("bw_rmsnorm_bwd", str(3 * x.numel() * x.element_size()))
                dx, dw = ext().rmsnorm_bwd(dy.contiguous(), x, weight, rstd)
                stop()
            else:
                dx, dw = ext().rmsnorm_bwd(dy.contiguous(), x, weight, rstd)
        else:
            H = x.shape[-1]
            xf = x.float().reshape(-1, H)
            dyf = dy.float().reshape(-1, H)
            wf = weight.float()
            rs = rstd.reshape(-1, 1)
            dot = (dyf * wf * xf).sum(-1, keepdim=True)
            dx = (rs * wf * dyf - xf * dot * rs.pow(3) / H).to(x.dtype).reshape(x.shape)
            dw = (dyf * xf * rs).sum(0)
        return dx, dw.to(weight.dtype), None


class RMSNorm(torch.nn.Module):
    def __init__(self, hidden_size, eps=1e-5, device=None, dtype=torch.bfloat16):
        super().__init__()
        self.weight = torch.nn.Parameter(
            torch.ones(hidden_size, device=device, dtype=dtype))
        self.eps = eps

    def forward(self, x):
        return _RMSNormFn.apply(x, self.weight, self.eps)


# --------------------------------------------------------------------------
# RoPE
# --------------------------------------------------------------------------
def build_rope_cache(max_seq, dim, base=500000.0, device="cpu"):
    """cos/sin table [max_seq, dim/2, 2] fp32 (half-rotation convention)."""
    inv = 1.0 / (base ** (torch.arange(0, dim, 2, dtype=torch.float32,
                                       device=device) / dim))
    t = torch.arange(max_seq, dtype=torch.float32, device=device)
    freqs = torch.outer(t, inv)  # [S, dim/2]
    return torch.stack([freqs.cos(), freqs.sin()], dim=-1).contiguous()


class _RoPEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, cs, pos):
        ctx.save_for_backward(cs, pos)
        if _use_hip(x):
            if insitu.ENABLED:
                stop = insitu.start("bw_rope", str(2 * x.numel() * x.element_size()))
                y = ext().rope(x.contiguous(), cs, pos, 1.0)
                stop()
                return y
            return ext().rope(x.contiguous(), cs, pos, 1.0)
        return _rope_torch(x, cs, pos, 1.0)

    @staticmethod
    def backward(ctx, dy):
        cs, pos = ctx.saved_tensors
        if _use_hip(dy):
            if insitu.ENABLED:
                stop = insitu.start("bw_rope", str(2 * dy.numel() * dy.element_size()))
                dx = ext().rope(dy.contiguous(), cs, pos, -1.0)
                stop()
                return dx, None, None
            return ext().rope(dy.contiguous(), cs, pos, -1.0), None, None
        return _rope_torch(dy, cs, pos, -1.0), None, None


def _rope_torch(x, cs, pos, sign):
    # x [tokens, heads, D]
    D = x.shape[-1]
    half = D // 2
    c = cs[pos.long()][:, None, :, 0]
    s = cs[pos.long()][:, None, :, 1] * sign
    xf = x.float()
    x1, x2 = xf[..., :half], xf[..., half:]
    return torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1).to(x.dtype)


def apply_rope(x, cs, pos):
    """x: [tokens, heads, D]; cs: [max_pos, D/2, 2] fp32; pos: [tokens] int32."""
    return _RoPEFn.apply(x, cs, pos)


# --------------------------------------------------------------------------
# SwiGLU
# --------------------------------------------------------------------------
class _SwigluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ctx.save_for_backward(x)
        if _use_hip(x):
            if insitu.ENABLED:
                b = x.numel() * x.element_size()
                stop = insitu.start("bw_swiglu", str(b + b // 2))
                y = ext().swiglu_fwd(x.contiguous())
                stop()
                return y
            return ext().swiglu_fwd(x.contiguous())
        g, u = x.float().chunk(2, dim=-1)
        return (torch.nn.functional.silu(g) * u).to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        if _use_hip(x):
            if insitu.ENABLED:
                b = x.numel() * x.element_size()
                stop = insitu.start("bw_swiglu_bwd", str(2 * b + b // 2))
                dx = ext().swiglu_bwd(dy.contiguous(), x)
                stop()
                return dx
            return ext().swiglu_bwd(dy.contiguous(), x)
        g, u = x.float().chunk(2, dim=-1)
        dyf = dy.float()
        sig = torch.sigmoid(g)
        silu = g * sig
        dg = dyf * u * (sig + silu * (1 - sig))
        du = dyf * silu
        return torch.cat([dg, du], dim=-1).to(x.dtype)


def swiglu(x):
    return _SwigluFn.apply(x)


# --------------------------------------------------------------------------
# Fused cross entropy (single vocab shard; TP variant reduces in Python)
# --------------------------------------------------------------------------
class _CEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels):
        if _use_hip(logits):
            if insitu.ENABLED:
                stop = insitu.start("bw_ce_fusion", str(logits.numel() * logits.element_size()))
                loss, row_max, row_sum = ext().ce_fwd(logits.contiguous(), labels)
                stop()
            else:
                loss, row_max, row_sum = ext().ce_fwd(logits.contiguous(), labels)
        else:
            lf = logits.float()
            row_max = lf.max(-1).values
            row_sum = (lf - row_max[:, None]).exp().sum(-1)
            ll = lf.gather(1, labels[:, None]).squeeze(1)
            loss = row_sum.log() + row_max - ll
        ctx.save_for_backward(logits, labels, row_max, row_sum)
        return loss

    @staticmethod
    def backward(ctx, dloss):
        logits, labels, row_max, row_sum = ctx.saved_tensors
        if _use_hip(logits):
            if insitu.ENABLED:
                stop = insitu.start("bw_ce_fusion_bwd", str(2 * logits.numel() * logits.element_size()))
                d = ext().ce_bwd(logits, labels, dloss.contiguous(), row_max, row_sum)
                stop()
            else:
                d = ext().ce_bwd(logits, labels, dloss.contiguous(), row_max, row_sum)
        else:
            p = (logits.float() - row_max[:, None]).exp() / row_sum[:, None]
            p.scatter_add_(1, labels[:, None],
                           -torch.ones_like(labels, dtype=p.dtype)[:, None])
            d = (p * dloss[:, None]).to(logits.dtype)
        return d, None


def fused_cross_entropy(logits, labels):
    """logits [rows, V] bf16, labels [rows] int64 -> per-row loss fp32."""
    return _CEFn.apply(logits, labels)


# --------------------------------------------------------------------------
# Linear with fused fp32 wgrad accumulation (hipBLAS GemmEx, TE-style)
# --------------------------------------------------------------------------
_DUMMY_WGRADS = {}


def clear_dummy_wgrads():
    """Release the shared dummy-wgrad buffers (benchmark sweeps that build
    several models in one process must call this between models)."""
    _DUMMY_WGRADS.clear()


def _dummy_wgrad(shape, device, dtype):
    """One shared placeholder grad per shape (TE get_dummy_wgrad analog —
    the reference memory model tracks these as te_dummy_wgrad_shapes)."""
    key = (tuple(shape), str(device), dtype)
    if key not in _DUMMY_WGRADS:
        _DUMMY_WGRADS[key] = torch.empty(shape, device=device, dtype=dtype)
    return _DUMMY_WGRADS[key]


class _FusedLinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight):
        ctx.save_for_backward(x, weight)
        if insitu.ENABLED and x.is_cuda:
            b, m = (x.shape[0], x.shape[1]) if x.ndim == 3 else (1, x.shape[0])
            stop = insitu.start("matmul", insitu.gemm_key(
                b, m, x.shape[-1], weight.shape[0], "TN", False, "bf16"))
            out = torch.matmul(x, weight.t())
            stop()
            return out
        return torch.matmul(x, weight.t())

    @staticmethod
    def backward(ctx, dout):
        x, weight = ctx.saved_tensors
        timing = insitu.ENABLED and x.is_cuda
        if timing:
            b, m = (x.shape[0], x.shape[1]) if x.ndim == 3 else (1, x.shape[0])
            stop = insitu.start("matmul", insitu.gemm_key(
                b, m, weight.shape[0], weight.shape[1], "NN", False, "bf16"))
        dx = torch.matmul(dout, weight)
        if timing:
            stop()
        if x.is_cuda and hasattr(weight, "main_grad"):
            d2 = dout.reshape(-1, dout.shape[-1]).contiguous()
            x2 = x.reshape(-1, x.shape[-1]).contiguous()
            if timing:
                stop = insitu.start("matmul", insitu.gemm_key(
                    1, weight.shape[0], x2.shape[0], weight.shape[1],
                    "NT", True, "fp32"))
            ext().wgrad_accum(d2, x2, weight.main_grad)
            if timing:
                stop()
            # fresh unreferenced buffer: autograd's AccumulateGrad STEALS it
            # (no weight-sized clone kernel; the shared-dummy variant gets
            # cloned because its use_count is too high). The DP reducer's
            # post-accumulate hook frees it right away.
            dw = torch.empty_like(weight)
        else:
            dw = torch.matmul(dout.reshape(-1, dout.shape[-1]).t(),
                              x.reshape(-1, x.shape[-1]))
        return dx, dw


class FusedLinear(torch.nn.Module):
    """nn.Linear(bias=False) whose wgrad accumulates straight into the fp32
    main_grad buffer (single GemmEx, no convert-and-add pass)."""

    def __init__(self, in_features, out_features, dtype=torch.bfloat16,
                 device=None):
        super().__init__()
        w = torch.empty(out_features, in_features, dtype=dtype, device=device)
        torch.nn.init.normal_(w, std=0.02)
        self.weight = torch.nn.Parameter(w)
        self.weight._fused_wgrad = True

    def forward(self, x):
        return _FusedLinearFn.apply(x, self.weight)


# --------------------------------------------------------------------------
# Flash attention (gfx950 HIP kernel; see csrc/attention.hip)
# --------------------------------------------------------------------------
class _FlashAttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal):
        if _use_hip(q):
            if insitu.ENABLED:
                stop = insitu.start("sdp_fwd", insitu.sdp_key(
                    q.shape[0], q.shape[1], q.shape[2], k.shape[2],
                    q.shape[3], v.shape[3],
                    contiguous=q.shape[3] == v.shape[3]))
                o, lse = ext().fa_fwd(q, k, v, causal)
                stop()
            else:
                o, lse = ext().fa_fwd(q, k, v, causal)
            ctx.save_for_backward(q, k, v, o, lse)
            ctx.causal = causal
            return o
        o = _sdpa_torch(q, k, v, causal)
        ctx.save_for_backward(q, k, v, o, torch.empty(0))
        ctx.causal = causal
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        if _use_hip(q):
            if insitu.ENABLED:
                stop = insitu.start("sdp_bwd", insitu.sdp_key(
                    q.shape[0], q.shape[1], q.shape[2], k.shape[2],
                    q.shape[3], v.shape[3],
                    contiguous=q.shape[3] == v.shape[3]))
                dq, dk, dv = ext().fa_bwd(do.contiguous(), q, k, v, o, lse,
                                          ctx.causal)
                stop()
            else:
                dq, dk, dv = ext().fa_bwd(do.contiguous(), q, k, v, o, lse,
                                          ctx.causal)
            return dq, dk, dv, None
        # CPU fallback: autograd through the reference math
        with torch.enable_grad():
            q2 = q.detach().requires_grad_(True)
            k2 = k.detach().requires_grad_(True)
            v2 = v.detach().requires_grad_(True)
            o2 = _sdpa_torch(q2, k2, v2, ctx.causal)
            o2.backward(do)
        return q2.grad, k2.grad, v2.grad, None


def _sdpa_torch(q, k, v, causal):
    # q [B,S,Hq,D], k/v [B,S,Hkv,D] -> o [B,S,Hq,D]
    B, S, Hq, D = q.shape
    Hkv = k.shape[2]
    rep = Hq // Hkv
    qt = q.permute(0, 2, 1, 3).float()
    kt = k.permute(0, 2, 1, 3).float().repeat_interleave(rep, dim=1)
    vt = v.permute(0, 2, 1, 3).float().repeat_interleave(rep, dim=1)
    scores = qt @ kt.transpose(-1, -2) / (D ** 0.5)
    if causal:
        mask = torch.triu(torch.ones(S, S, dtype=torch.bool,
                                     device=q.device), 1)
        scores = scores.masked_fill(mask, float("-inf"))
    o = torch.softmax(scores, dim=-1) @ vt
    return o.permute(0, 2, 1, 3).to(q.dtype)


def flash_attention(q, k, v, causal=True):
    """q [B,S,Hq,D], k/v [B,S,Hkv,D] (GQA) -> [B,S,Hq,D]."""
    return _FlashAttnFn.apply(q.contiguous(), k.contiguous(), v.contiguous(),
                              causal)
