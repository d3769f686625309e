"""fp8 (OCP e4m3/e5m2) linear path for the trainer — gfx950's 5 PF dense
fp8 MFMA rate through hipBLASLt's _scaled_mm.

TransformerEngine-style recipe with per-tensor DELAYED scaling: each
tensor role (activation, gradient) keeps a running amax buffer; the fused
gfx950 cast kernel (kernels/csrc/fp8_cast.hip) converts bf16 -> fp8 in
ONE pass using last call's amax as the scale while recording this call's
amax for the next — a naive dynamic recipe needs 3 passes (amax reduce,
scale-mul, cast) and measured 41% SLOWER than bf16 end to end. Weights
re-quantize only when the parameter version changes (once per optimizer
step, shared by all microbatches), caching both the row-major fp8 weight
and the column-major copy _scaled_mm wants for dgrad.

Numerics: activations/weights e4m3, gradients e5m2 (wider exponent),
fp32 accumulation, bf16 GEMM outputs. The simulator prices this path via
the fp8_matmul efficiency table (ops/dense.py fp8 branch; reference
dense_module.py:2365-2453).

torch._scaled_mm contract (ROCm/hipBLASLt): A row-major [M,K], B
column-major [K,N], both fp8, per-tensor fp32 scale tensors on device.
"""

from __future__ import annotations

import torch

from . import insitu

E4M3_MAX = 448.0
E5M2_MAX = 57344.0


def _ext():
    from .ops import ext

    return ext()


def _quant_dynamic(t, fmt):
    """Two-pass fallback (CPU tensors / first call priming)."""
    dt = torch.float8_e4m3fn if fmt == "e4m3" else torch.float8_e5m2
    fmax = E4M3_MAX if fmt == "e4m3" else E5M2_MAX
    amax = t.abs().amax().float().clamp(min=1e-12)
    scale = fmax / amax
    q = (t.float() * scale).clamp(-fmax, fmax).to(dt)
    return q, (amax / fmax).reshape(1)


def _quant_delayed(t, fmt, amax_buf, primed):
    """One-pass fused cast using last call's amax; records this call's
    amax into amax_buf. Returns (fp8 tensor, descale [1])."""
    if not t.is_cuda or t.numel() % 8 != 0:
        return _quant_dynamic(t, fmt)
    if not primed[0]:
        # prime the amax with a one-off reduction so call 1 is well-scaled
        amax_buf.copy_(t.abs().amax().float().reshape(()))
        primed[0] = True
    dt = torch.float8_e4m3fn if fmt == "e4m3" else torch.float8_e5m2
    fmax = E4M3_MAX if fmt == "e4m3" else E5M2_MAX
    a_prev = amax_buf.clamp(min=1e-12).reshape(1)
    scale = fmax / a_prev
    descale = a_prev / fmax
    amax_buf.zero_()
    out = torch.empty(t.shape, dtype=dt, device=t.device)
    if insitu.ENABLED:
        # simulator charges the cast at 2x the bf16 bytes (ops/dense.py
        # fp8 branch); record with the same convention so the overlaid
        # fp8_quant efficiency composes
        stop = insitu.start("bw_fp8_quant", str(4 * t.numel()))
        _ext().fp8_cast(t, out, amax_buf, scale, fmt == "e5m2", fmax)
        stop()
    else:
        _ext().fp8_cast(t, out, amax_buf, scale, fmt == "e5m2", fmax)
    return out, descale


def _quant_delayed_t(t, fmt, amax_buf, primed):
    """Cast + transpose in one fused pass: returns (q [M,N], q_t [N,M],
    descale). Falls back to cast + torch transpose off the fast path."""
    if (not t.is_cuda or t.dim() != 2 or t.size(0) % 64 or t.size(1) % 64):
        q, d = _quant_delayed(t, fmt, amax_buf, primed) if t.is_cuda else \
            _quant_dynamic(t, fmt)
        return q, q.t().contiguous(), d
    if not primed[0]:
        amax_buf.copy_(t.abs().amax().float().reshape(()))
        primed[0] = True
    dt = torch.float8_e4m3fn if fmt == "e4m3" else torch.float8_e5m2
    fmax = E4M3_MAX if fmt == "e4m3" else E5M2_MAX
    a_prev = amax_buf.clamp(min=1e-12).reshape(1)
    scale = fmax / a_prev
    descale = a_prev / fmax
    amax_buf.zero_()
    q = torch.empty(t.shape, dtype=dt, device=t.device)
    qt = torch.empty((t.size(1), t.size(0)), dtype=dt, device=t.device)
    if insitu.ENABLED:
        stop = insitu.start("bw_fp8_quant", str(4 * t.numel()))
        _ext().fp8_cast_t(t, q, qt, amax_buf, scale, fmt == "e5m2", fmax)
        stop()
    else:
        _ext().fp8_cast_t(t, q, qt, amax_buf, scale, fmt == "e5m2", fmax)
    return q, qt, descale


class _Fp8LinearFn(torch.autograd.Function):
    """y = x @ w^T with all three GEMMs (fwd / dgrad / wgrad) in fp8."""

    @staticmethod
    def forward(ctx, x, weight, mod):
        ishape = x.shape
        x2 = x.reshape(-1, ishape[-1]).contiguous()
        # cast_transpose: the transposed image is wgrad's column-major B
        # operand (a torch fp8 .t().contiguous() costs more than the GEMM)
        xq, xq_t, xs = _quant_delayed_t(x2, "e4m3", mod.x_amax, mod._x_primed)
        wq, ws, w_cm = mod._weight_quant()
        if insitu.ENABLED and x.is_cuda:
            b, m = (x.shape[0], x.shape[1]) if x.ndim == 3 else (1, x.shape[0])
            stop = insitu.start("fp8_matmul", insitu.gemm_key(
                b, m, x.shape[-1], weight.shape[0], "TN", False, "bf16"))
            y = torch._scaled_mm(xq, wq.t(), scale_a=xs, scale_b=ws,
                                 out_dtype=torch.bfloat16)
            stop()
        else:
            y = torch._scaled_mm(xq, wq.t(), scale_a=xs, scale_b=ws,
                                 out_dtype=torch.bfloat16)
        ctx.save_for_backward(xq_t, xs, ws, w_cm)
        ctx.mod = mod
        ctx.ishape = ishape
        # simulator shape-key (b, m) convention: 3D tensors key as (B, S)
        ctx.bm = ((x.shape[0], x.shape[1]) if x.ndim == 3
                  else (1, x.shape[0]))
        return y.reshape(*ishape[:-1], weight.shape[0])

    @staticmethod
    def backward(ctx, dy):
        xq_t, xs, ws, w_cm = ctx.saved_tensors
        mod = ctx.mod
        dy2 = dy.reshape(-1, dy.shape[-1]).contiguous()
        gq, gqt, gs = _quant_delayed_t(dy2, "e5m2", mod.g_amax,
                                       mod._g_primed)
        timing = insitu.ENABLED and dy.is_cuda
        M, N = gq.shape
        K = w_cm.shape[1]
        kb, km = ctx.bm
        if timing:
            stop = insitu.start("fp8_matmul", insitu.gemm_key(
                kb, km, N, K, "NN", False, "bf16"))
        # dgrad: dx[M,K] = dy[M,N] @ w[N,K] (w_cm: cached col-major weight)
        dx = torch._scaled_mm(gq, w_cm, scale_a=gs, scale_b=ws,
                              out_dtype=torch.bfloat16)
        if timing:
            stop()
        # wgrad: dw[N,K] = dy^T[N,M] @ x[M,K]; both operands come from the
        # fused cast_transpose (gqt row-major [N,M]; xq_t.t() col-major
        # [M,K]). bf16 out — the post-accumulate hook adds into the fp32
        # main_grad (Megatron grad_reduce_in_bf16 semantics)
        if timing:
            stop = insitu.start("fp8_matmul", insitu.gemm_key(
                1, N, M, K, "NT", True, "fp32"))  # wgrad keys flatten (b,m)
        dw = torch._scaled_mm(gqt, xq_t.t(), scale_a=gs, scale_b=xs,
                              out_dtype=torch.bfloat16)
        if timing:
            stop()
        return dx.reshape(ctx.ishape), dw, None


class Fp8Linear(torch.nn.Module):
    """Drop-in fp8 linear (bias-free, Megatron-style) with delayed
    scaling and per-step weight-quant caching."""

    def __init__(self, in_features, out_features, dtype=torch.bfloat16,
                 device=None):
        super().__init__()
        w = torch.empty(out_features, in_features, dtype=dtype, device=device)
        torch.nn.init.normal_(w, std=0.02)
        self.weight = torch.nn.Parameter(w)
        self.register_buffer("x_amax", torch.zeros((), dtype=torch.float32,
                                                   device=device),
                             persistent=False)
        self.register_buffer("g_amax", torch.zeros((), dtype=torch.float32,
                                                   device=device),
                             persistent=False)
        self._x_primed = [False]
        self._g_primed = [False]
        self._wcache = None          # (version, wq, ws, w_cm)

    def _weight_quant(self):
        ver = self.weight._version
        if self._wcache is not None and self._wcache[0] == ver:
            return self._wcache[1], self._wcache[2], self._wcache[3]
        wq, ws = _quant_dynamic(self.weight.detach(), "e4m3")
        w_cm = wq.t().contiguous().t()
        self._wcache = (ver, wq, ws, w_cm)
        return wq, ws, w_cm

    def forward(self, x):
        return _Fp8LinearFn.apply(x, self.weight, self)


def fp8_available():
    try:
        if not torch.cuda.is_available():
            return False
        a = torch.randn(16, 32, device="cuda", dtype=torch.bfloat16)
        b = torch.randn(16, 32, device="cuda", dtype=torch.bfloat16)
        aq, asc = _quant_dynamic(a, "e4m3")
        bq, bsc = _quant_dynamic(b, "e4m3")
        torch._scaled_mm(aq, bq.t(), scale_a=asc, scale_b=bsc,
                         out_dtype=torch.bfloat16)
        return True
    except (RuntimeError, AttributeError):
        return False
