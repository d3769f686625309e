"""Quick eff probe for the fused memory-bound kernels."""
import sys, os, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from simumax_amd.kernels.ops import ext, build_rope_cache
E = ext()
def t(fn, it=20):
    for _ in range(5): fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(True); e = torch.cuda.Event(True); s.record()
    for _ in range(it): fn()
    e.record(); torch.cuda.synchronize()
    return s.elapsed_time(e)/it
x = torch.randn(4096, 4096, device="cuda", dtype=torch.bfloat16)
w = torch.randn(4096, device="cuda", dtype=torch.bfloat16)
y, rstd = E.rmsnorm_fwd(x, w, 1e-5)
dy = torch.randn_like(x)
dwz = lambda: E.rmsnorm_bwd(dy, x, w, rstd)
ms = t(dwz)
print(f"rmsnorm_bwd 4096x4096: {ms*1e3:.1f} us eff={3*x.numel()*2/(ms/1e3)/(8000*1024**3):.3f}")
q = torch.randn(4096, 32, 128, device="cuda", dtype=torch.bfloat16)
cs = build_rope_cache(8192, 128, device="cuda")
pos = torch.arange(4096, dtype=torch.int32, device="cuda")
ms = t(lambda: E.rope(q, cs, pos, 1.0))
print(f"rope 4096x32x128: {ms*1e3:.1f} us eff={2*q.numel()*2/(ms/1e3)/(8000*1024**3):.3f}")
