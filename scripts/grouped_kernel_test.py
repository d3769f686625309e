"""Numerics + timing for the hand-written grouped GEMM kernels
(grouped_gemm.hip) vs per-expert torch.mm loops, at the mixtral (E=8,
fat experts) and DeepSeek (E=160, skinny experts) training shapes."""
import sys

import torch

sys.path.insert(0, "/root/repo")
from simumax_amd.kernels.ops import ext


def relerr(a, b):
    b = b.float()
    return ((a.float() - b).abs().max() / b.abs().max().clamp(min=1e-6)).item()


def timeit(fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    s, e = torch.cuda.Event(True), torch.cuda.Event(True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters


def check(E, M, N, K, tag):
    E_ = ext()
    dev = "cuda:0"
    torch.manual_seed(0)
    x = torch.randn(E, M, K, device=dev, dtype=torch.bfloat16) / 8
    w = torch.randn(E, N, K, device=dev, dtype=torch.bfloat16) / 8
    dout = torch.randn(E, M, N, device=dev, dtype=torch.bfloat16) / 8

    # fwd
    c = E_.grouped_fwd(x, w)
    cref = torch.stack([x[e].float() @ w[e].float().t() for e in range(E)])
    e1 = relerr(c, cref)
    # dgrad
    dx = E_.grouped_dgrad(dout, w)
    dxref = torch.stack([dout[e].float() @ w[e].float() for e in range(E)])
    e2 = relerr(dx, dxref)
    # wgrad
    g = torch.randn(E, N, K, device=dev, dtype=torch.float32)
    g0 = g.clone()
    E_.grouped_wgrad(dout, x, g)
    gref = g0 + torch.stack([dout[e].float().t() @ x[e].float()
                             for e in range(E)])
    e3 = relerr(g, gref)
    print(f"[{tag}] E={E} M={M} N={N} K={K} relerr fwd {e1:.2e} "
          f"dgrad {e2:.2e} wgrad {e3:.2e}")
    assert e1 < 2e-2 and e2 < 2e-2 and e3 < 2e-2, (e1, e2, e3)

    # timing vs per-expert loop (loop uses w.transpose view for fwd)
    fl = 2 * E * M * N * K
    t_f = timeit(lambda: E_.grouped_fwd(x, w))
    t_fl = timeit(lambda: [torch.mm(x[e], w[e].t()) for e in range(E)])
    t_d = timeit(lambda: E_.grouped_dgrad(dout, w))
    t_dl = timeit(lambda: [torch.mm(dout[e], w[e]) for e in range(E)])
    t_w = timeit(lambda: E_.grouped_wgrad(dout, x, g))
    t_wl = timeit(lambda: [E_.wgrad_accum(dout[e], x[e], g[e])
                           for e in range(E)])
    print(f"    fwd   kernel {t_f:7.3f} ms ({fl/t_f/1e9:5.0f} TF/s)  "
          f"loop {t_fl:7.3f} ms ({fl/t_fl/1e9:5.0f} TF/s)")
    print(f"    dgrad kernel {t_d:7.3f} ms ({fl/t_d/1e9:5.0f} TF/s)  "
          f"loop {t_dl:7.3f} ms ({fl/t_dl/1e9:5.0f} TF/s)")
    print(f"    wgrad kernel {t_w:7.3f} ms ({fl/t_w/1e9:5.0f} TF/s)  "
          f"loop {t_wl:7.3f} ms ({fl/t_wl/1e9:5.0f} TF/s)", flush=True)


def main():
    check(8, 1024, 3584, 4096, "small-sanity")        # odd M guard path
    check(8, 1024, 28672, 4096, "mixtral-gl1")
    check(8, 1024, 4096, 14336, "mixtral-gl2")
    check(160, 154, 3072, 5120, "deepseek-gl1")
    check(160, 154, 5120, 1536, "deepseek-gl2")


if __name__ == "__main__":
    main()
