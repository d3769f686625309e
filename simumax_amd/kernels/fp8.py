"""fp8 (OCP e4m3/e5m2) linear path for the trainer — gfx950's 5 PF dense
fp8 MFMA rate through hipBLASLt's _scaled_mm.

TransformerEngine-style recipe, simplified to per-tensor DYNAMIC scaling
(amax of the current tensor, no history window): activations and weights
quantize to e4m3 in forward; gradients to e5m2 in backward (wider exponent
range); all GEMMs accumulate in fp32 and emit bf16. The simulator prices
this path with the fp8_matmul efficiency table (ops/dense.py
QuantizedColLinear/QuantizedRowLinear; reference dense_module.py:2365-2453).

torch._scaled_mm contract (ROCm/hipBLASLt): A row-major [M,K], B
column-major [K,N] (i.e. pass w.t() of a row-major [N,K] weight), both fp8,
per-tensor fp32 scales, out_dtype bf16.
"""

from __future__ import annotations

import torch

E4M3_MAX = 448.0
E5M2_MAX = 57344.0


def _quant(t, fmt):
    """Per-tensor dynamic scaling quantize. Returns (fp8 tensor, descale)."""
    dt = torch.float8_e4m3fn if fmt == "e4m3" else torch.float8_e5m2
    fmax = E4M3_MAX if fmt == "e4m3" else E5M2_MAX
    amax = t.abs().amax().float().clamp(min=1e-12)
    scale = fmax / amax
    q = (t.float() * scale).clamp(-fmax, fmax).to(dt)
    return q, (1.0 / scale).view(1)


class _Fp8LinearFn(torch.autograd.Function):
    """y = x @ w^T with all three GEMMs (fwd / dgrad / wgrad) in fp8."""

    @staticmethod
    def forward(ctx, x, weight):
        ishape = x.shape
        x2 = x.reshape(-1, ishape[-1])
        xq, xs = _quant(x2, "e4m3")
        wq, ws = _quant(weight, "e4m3")
        y = torch._scaled_mm(xq, wq.t(), scale_a=xs, scale_b=ws,
                             out_dtype=torch.bfloat16)
        ctx.save_for_backward(xq, xs, wq, ws)
        ctx.ishape = ishape
        return y.reshape(*ishape[:-1], weight.shape[0])

    @staticmethod
    def backward(ctx, dy):
        xq, xs, wq, ws = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.shape[-1])
        gq, gs = _quant(dy2, "e5m2")
        # _scaled_mm wants mat2 COLUMN-major: materialize the transposed
        # copies (TE keeps a cached transposed weight the same way)
        # dgrad: dx[M,K] = dy[M,N] @ w[N,K]
        w_cm = wq.t().contiguous().t()
        dx = torch._scaled_mm(gq, w_cm, scale_a=gs, scale_b=ws,
                              out_dtype=torch.bfloat16)
        # wgrad: dw[N,K] = dy^T[N,M] @ x[M,K]
        gqt = gq.t().contiguous()
        x_cm = xq.t().contiguous().t()
        dw = torch._scaled_mm(gqt, x_cm, scale_a=gs, scale_b=xs,
                              out_dtype=torch.float32)
        return dx.reshape(ctx.ishape), dw


class Fp8Linear(torch.nn.Module):
    """Drop-in fp8 linear (bias-free, Megatron-style)."""

    def __init__(self, in_features, out_features, dtype=torch.bfloat16,
                 device=None):
        super().__init__()
        w = torch.empty(out_features, in_features, dtype=dtype, device=device)
        torch.nn.init.normal_(w, std=0.02)
        self.weight = torch.nn.Parameter(w)

    def forward(self, x):
        return _Fp8LinearFn.apply(x, self.weight)


def fp8_available():
    try:
        if not torch.cuda.is_available():
            return False
        a = torch.randn(16, 32, device="cuda", dtype=torch.bfloat16)
        b = torch.randn(16, 32, device="cuda", dtype=torch.bfloat16)
        aq, asc = _quant(a, "e4m3")
        bq, bsc = _quant(b, "e4m3")
        torch._scaled_mm(aq, bq.t(), scale_a=asc, scale_b=bsc,
                         out_dtype=torch.bfloat16)
        return True
    except (RuntimeError, AttributeError):
        return False
