"""Operator-efficiency sweeps for MI355X (run on the GPU box).

Rebuild of the reference calibration harness
(simu_tools/efficency_test/test_gemm_efficiency.py,
test_fa_efficiency.py, test_ce_permute_efficiency.py,
run_one_click_benchmark.py) on PyTorch-ROCm + the gfx950 HIP kernels:

* GEMM shapes are enumerated through PerfLLM.analysis_op_info itself, so
  the measured shape-key strings match the cost-model lookup keys exactly
  (the reference's self-referential trick).
* matmul efficiency times what the trainer actually runs: hipBLASLt via
  torch (TN fwd, NN dgrad, NT wgrad; the fp32-accumulate wgrad key times
  the composite bf16 GEMM + fp32 convert-accumulate).
* sdp_fwd/sdp_bwd time the in-repo flash-attention HIP kernels.
* bandwidth table: HBM stream, fused CE, optimizer (in-place Adam traffic).

Results accumulate into gpurun_out/calib/*.json (resumable: existing keys
are skipped unless EFFICIENCY_OVERWRITE=1, reference parity
test_gemm_efficiency.py:226-249).
"""

from __future__ import annotations

import json
import os
import re
import sys
import time

import torch

PEAK_BF16 = 2500e12
HBM_PEAK_GBPS = 8000.0
OVERWRITE = os.environ.get("EFFICIENCY_OVERWRITE", "0") == "1"

OUT_DIR = os.environ.get("CALIB_OUT", "gpurun_out/calib")


def _timeit(fn, warmup=3, iters=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    start = torch.cuda.Event(True)
    end = torch.cuda.Event(True)
    start.record()
    for _ in range(iters):
        fn()
    end.record()
    torch.cuda.synchronize()
    return start.elapsed_time(end) / iters  # ms


def _load(path):
    """Resume support: seed from the tracked calib_raw/ baseline (which
    travels in the repo snapshot) so a fresh GPU box only measures shapes
    not already in the committed tables."""
    out = {}
    seed = os.path.join(os.path.dirname(os.path.dirname(
        os.path.dirname(os.path.abspath(__file__)))), "calib_raw",
        os.path.basename(path))
    for q in (seed, path):
        if os.path.exists(q):
            with open(q) as f:
                out.update(json.load(f))
    return out


def _save(path, data):
    os.makedirs(os.path.dirname(path), exist_ok=True)
    with open(path, "w") as f:
        json.dump(data, f, indent=1, sort_keys=True)


def parse_gemm_key(desc):
    m = re.match(
        r"b=(\d+), m=(\d+), k=(\d+), n=(\d+), layout=(\w+), "
        r"accumulate=(\w+), out_dtype=(\w+)", desc)
    assert m, desc
    b, mm, k, n = (int(m.group(i)) for i in range(1, 5))
    return b, mm, k, n, m.group(5), m.group(6) == "True", m.group(7)


def time_gemm(desc, device="cuda"):
    b, m, k, n, layout, accumulate, out_dtype = parse_gemm_key(desc)
    dt = torch.bfloat16
    if layout == "TN":       # fwd: x[b,m,k] @ W[n,k]^T
        x = torch.randn(b, m, k, device=device, dtype=dt)
        w = torch.randn(n, k, device=device, dtype=dt)
        fn = lambda: torch.nn.functional.linear(x, w)
    elif layout == "NN":     # dgrad: dout[b,m,k(=N_out)] @ W[k,n]
        d = torch.randn(b, m, k, device=device, dtype=dt)
        w = torch.randn(k, n, device=device, dtype=dt)
        fn = lambda: torch.matmul(d, w)
    elif layout == "NT":     # wgrad: dout^T[ m(=out), k(=tokens)] @ x[k, n]
        dout = torch.randn(k, m, device=device, dtype=dt)
        x = torch.randn(k, n, device=device, dtype=dt)
        if accumulate and out_dtype == "fp32":
            # trainer path: fused GemmEx into fp32 main_grad (kernels/ops.py)
            from simumax_amd.kernels.ops import ext
            E = ext()
            main_grad = torch.zeros(m, n, device=device, dtype=torch.float32)
            # dout [tokens=k, out=m], x [tokens=k, in=n] -> main_grad [m, n]
            fn = lambda: E.wgrad_accum(dout, x, main_grad)
    else:
        raise ValueError(layout)
    iters = 10 if b * m * k * n < 2**40 else 4
    t_ms = _timeit(fn, iters=iters)
    flops = 2 * b * m * k * n
    return flops / (t_ms / 1e3) / PEAK_BF16, t_ms


def sweep_gemms(shape_keys, path):
    table = _load(path)
    for desc in shape_keys:
        if desc in table and not OVERWRITE:
            continue
        try:
            eff, t_ms = time_gemm(desc)
        except torch.cuda.OutOfMemoryError:
            torch.cuda.empty_cache()
            print(f"[gemm] OOM {desc}", flush=True)
            continue
        table[desc] = eff
        print(f"[gemm] {desc} -> eff {eff:.4f} ({t_ms:.3f} ms)", flush=True)
        _save(path, table)
    return table


def parse_sdp_key(desc):
    m = re.match(
        r"batch=(\d+), seq_len=(\d+), head_num=(\d+), kv_head_num=(\d+), "
        r"qk_head_dim=(\d+), v_head_dim=(\d+), qkv_contiguous=(\w+)", desc)
    assert m, desc
    return tuple(int(m.group(i)) for i in range(1, 7))


def sweep_sdp(shape_keys, fwd_path, bwd_path, sparse_ratio=0.5):
    from simumax_amd.kernels.ops import ext

    E = ext()
    fwd_tab, bwd_tab = _load(fwd_path), _load(bwd_path)
    for desc in shape_keys:
        if desc in fwd_tab and desc in bwd_tab and not OVERWRITE:
            continue
        b, s, hq, hkv, dqk, dv = parse_sdp_key(desc)
        if dqk not in (128, 192) or dv != 128:
            print(f"[sdp] skip unsupported shape {desc}", flush=True)
            continue
        dt = torch.bfloat16
        try:
            q = torch.randn(b, s, hq, dqk, device="cuda", dtype=dt)
            k = torch.randn(b, s, hkv, dqk, device="cuda", dtype=dt)
            v = torch.randn(b, s, hkv, dv, device="cuda", dtype=dt)
            o, lse = E.fa_fwd(q, k, v, True)
            do = torch.randn_like(o)
            t_fwd = _timeit(lambda: E.fa_fwd(q, k, v, True), iters=5)
            t_bwd = _timeit(lambda: E.fa_bwd(do, q, k, v, o, lse, True), iters=5)
        except torch.cuda.OutOfMemoryError:
            torch.cuda.empty_cache()
            print(f"[sdp] OOM {desc}", flush=True)
            continue
        sparse = 1.0 - sparse_ratio
        qk_fl = 2 * b * hq * s * s * dqk
        pv_fl = 2 * b * hq * s * s * dv
        f_fwd = (qk_fl + pv_fl) * sparse
        f_bwd = (2 * qk_fl + 2 * pv_fl + qk_fl) * sparse
        fwd_tab[desc] = f_fwd / (t_fwd / 1e3) / PEAK_BF16
        bwd_tab[desc] = f_bwd / (t_bwd / 1e3) / PEAK_BF16
        print(f"[sdp] {desc} -> fwd {fwd_tab[desc]:.4f} bwd {bwd_tab[desc]:.4f}",
              flush=True)
        _save(fwd_path, fwd_tab)
        _save(bwd_path, bwd_tab)
    return fwd_tab, bwd_tab


def time_fp8_gemm(desc, device="cuda"):
    """fp8 e4m3 GEMM via torch._scaled_mm (hipBLASLt fp8 path on gfx950);
    NT-accumulate wgrad shapes are skipped (grads stay bf16 on this stack)."""
    b, m, k, n, layout, accumulate, out_dtype = parse_gemm_key(desc)
    if layout == "NT" or b != 1:
        return None
    f8 = torch.float8_e4m3fn
    if layout == "TN":     # x[m,k] @ w[n,k]^T
        a = torch.randn(m, k, device=device).to(f8)
        w = torch.randn(n, k, device=device).to(f8)
        bmat = w.t()
    else:                  # NN: d[m,k] @ w[k,n]
        a = torch.randn(m, k, device=device).to(f8)
        bmat = torch.randn(n, k, device=device).to(f8).t()  # column-major [k,n]
    sa = torch.ones((), device=device)
    sb = torch.ones((), device=device)
    fn = lambda: torch._scaled_mm(a, bmat, scale_a=sa, scale_b=sb,
                                  out_dtype=torch.bfloat16)
    t_ms = _timeit(fn, iters=10)
    flops = 2 * m * k * n
    return flops / (t_ms / 1e3) / (2 * PEAK_BF16), t_ms  # vs 5 PF fp8 peak


def sweep_fp8_gemms(shape_keys, path):
    table = _load(path)
    for desc in shape_keys:
        if desc in table and not OVERWRITE:
            continue
        try:
            r = time_fp8_gemm(desc)
        except (RuntimeError, torch.cuda.OutOfMemoryError) as e:
            print(f"[fp8] skip {desc}: {str(e)[:80]}", flush=True)
            torch.cuda.empty_cache()
            continue
        if r is None:
            continue
        table[desc] = r[0]
        print(f"[fp8] {desc} -> eff {r[0]:.4f} ({r[1]:.3f} ms)", flush=True)
        _save(path, table)
    return table


def sweep_fp8_grouped(shape_keys, path):
    """fp8 e4m3 grouped GEMM (per-expert _scaled_mm loop) for the
    fp8_group_matmul table; keys are the bf16 grouped keys with
    dtype=fp8. Efficiency vs the 5 PF dense fp8 peak."""
    from simumax_amd.kernels.fp8 import _quant_dynamic as _quant

    tab = _load(path)
    for bkey in shape_keys:
        desc = bkey.replace(" dtype=bf16,", " dtype=fp8,")
        if desc in tab and not OVERWRITE:
            continue
        ng, m, n, k = parse_group_key(bkey)
        try:
            xs, ws, sx, sw = [], [], [], []
            for _ in range(ng):
                x = torch.randn(m, k, device="cuda", dtype=torch.bfloat16) / 8
                w = torch.randn(n, k, device="cuda", dtype=torch.bfloat16) / 8
                xq, xsc = _quant(x, "e4m3")
                wq, wsc = _quant(w, "e4m3")
                # wq row-major [N,K]: wq.t() is [K,N] column-major, the
                # layout _scaled_mm requires for mat2
                xs.append(xq); ws.append(wq)
                sx.append(xsc); sw.append(wsc)

            def fn():
                for i in range(ng):
                    torch._scaled_mm(xs[i], ws[i].t(), scale_a=sx[i],
                                     scale_b=sw[i], out_dtype=torch.bfloat16)
            t_ms = _timeit(fn, warmup=2, iters=5)
            flops = 2.0 * ng * m * n * k
            eff = flops / (t_ms / 1e3) / (2 * PEAK_BF16)
        except (RuntimeError, torch.cuda.OutOfMemoryError) as e:
            print(f"[fp8group] skip {desc}: {str(e)[:70]}", flush=True)
            torch.cuda.empty_cache()
            continue
        tab[desc] = round(eff, 4)
        print(f"[fp8group] {desc[:60]} -> eff {eff:.4f}", flush=True)
        _save(path, tab)
        del xs, ws
        torch.cuda.empty_cache()
    return tab


def parse_group_key(desc):
    m = re.match(r"ng=(\d+), M=(\d+), N=(\d+), K=(\d+), dtype=(\w+)", desc)
    assert m, desc
    return int(m.group(1)), int(m.group(2)), int(m.group(3)), int(m.group(4))


def sweep_grouped(shape_keys, path):
    """Time the trainer's grouped-GEMM realization per stage via the SAME
    dispatch the trainer runs (train/moe.py grouped_*_op: batched MFMA
    kernels when E >= GROUPED_KERNEL_MIN_E, per-expert hipBLASLt loops
    otherwise)."""
    from simumax_amd.train.moe import (grouped_dgrad_op, grouped_fwd_op,
                                       grouped_wgrad_op)

    tab = _load(path)
    for desc in shape_keys:
        if desc in tab and not OVERWRITE:
            continue
        ng, m, n, k = parse_group_key(desc)
        try:
            x = torch.randn(ng, m, k, device="cuda", dtype=torch.bfloat16)
            w = torch.randn(ng, n, k, device="cuda", dtype=torch.bfloat16)
            if "stage=fwd" in desc:
                fn = lambda: grouped_fwd_op(x, w)
            elif "stage=bwd_grad_act" in desc:
                d = torch.randn(ng, m, n, device="cuda", dtype=torch.bfloat16)
                fn = lambda: grouped_dgrad_op(d, w)
            else:  # wgrad: fused fp32 accumulate
                d = torch.randn(ng, m, n, device="cuda", dtype=torch.bfloat16)
                mg = torch.zeros(ng, n, k, device="cuda", dtype=torch.float32)
                fn = lambda: grouped_wgrad_op(d, x, mg)
            t_ms = _timeit(fn, iters=8)
        except torch.cuda.OutOfMemoryError:
            torch.cuda.empty_cache()
            print(f"[group] OOM {desc}", flush=True)
            continue
        flops = 2 * ng * m * k * n
        tab[desc] = flops / (t_ms / 1e3) / PEAK_BF16
        print(f"[group] {desc} -> eff {tab[desc]:.4f} ({t_ms:.3f} ms)", flush=True)
        _save(path, tab)
    return tab


def sweep_bandwidth(path):
    """HBM stream + fused-op bandwidth efficiencies + optimizer traffic."""
    from simumax_amd.kernels.ops import ext

    E = ext()
    out = _load(path)
    GiB = 1024**3

    # stream copy: read + write
    n = 2 * GiB
    x = torch.empty(n // 2, device="cuda", dtype=torch.bfloat16)
    y = torch.empty_like(x)
    t = _timeit(lambda: y.copy_(x))
    stream_gbps = 2 * n / (t / 1e3) / GiB
    out["stream_gbps"] = stream_gbps
    out["default_eff"] = stream_gbps / HBM_PEAK_GBPS

    # kernel-launch floor: tiny elementwise op
    tiny = torch.empty(256, device="cuda")
    t_launch = _timeit(lambda: tiny.add_(1.0), warmup=10, iters=200)
    out["launch_us"] = t_launch * 1e3

    # rmsnorm fwd/bwd achieved vs modeled bytes (model: 2*b fwd, 3*b bwd)
    rows, H = 8192, 4096
    xb = torch.randn(rows, H, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(H, device="cuda", dtype=torch.bfloat16)
    yb, rstd = E.rmsnorm_fwd(xb, w, 1e-5)
    b_bytes = rows * H * 2
    t = _timeit(lambda: E.rmsnorm_fwd(xb, w, 1e-5))
    out["rmsnorm_fwd_eff"] = 2 * b_bytes / (t / 1e3) / (HBM_PEAK_GBPS * GiB)
    dy = torch.randn_like(xb)
    t = _timeit(lambda: E.rmsnorm_bwd(dy, xb, w, rstd))
    out["rmsnorm_bwd_eff"] = 3 * b_bytes / (t / 1e3) / (HBM_PEAK_GBPS * GiB)

    # swiglu
    xs = torch.randn(rows, 2 * 14336, device="cuda", dtype=torch.bfloat16)
    t = _timeit(lambda: E.swiglu_fwd(xs))
    sw_bytes = xs.numel() * 2 + xs.numel()
    out["swiglu_fwd_eff"] = sw_bytes / (t / 1e3) / (HBM_PEAK_GBPS * GiB)

    # fused CE (key ce_fusion): model fwd bytes = logits
    rows_ce, V = 4096, 128256
    logits = torch.randn(rows_ce, V, device="cuda", dtype=torch.bfloat16)
    labels = torch.randint(0, V, (rows_ce,), device="cuda")
    t = _timeit(lambda: E.ce_fwd(logits, labels), iters=5)
    lb = rows_ce * V * 2
    out["ce_fusion_fwd_eff"] = lb / (t / 1e3) / (HBM_PEAK_GBPS * GiB)
    loss, rm, rs = E.ce_fwd(logits, labels)
    dl = torch.randn_like(loss)
    t = _timeit(lambda: E.ce_bwd(logits, labels, dl, rm, rs), iters=5)
    out["ce_fusion_bwd_eff"] = 2 * lb / (t / 1e3) / (HBM_PEAK_GBPS * GiB)

    # optimizer: the trainer's exact step sequence; traffic model shared
    # via core.consts.OPTIMIZER_TRAFFIC_BYTES_PER_PARAM
    from simumax_amd.core.consts import OPTIMIZER_TRAFFIC_BYTES_PER_PARAM

    numel = 2 * GiB // 4
    master = torch.zeros(numel, device="cuda", dtype=torch.float32)
    m_ = torch.zeros_like(master)
    v_ = torch.zeros_like(master)
    g_ = torch.randn_like(master)
    pb = torch.zeros(numel, device="cuda", dtype=torch.bfloat16)

    def adam():
        g_.zero_()                       # zero_grad
        _ = g_.norm(2)                   # grad-norm clip pass
        torch._foreach_mul_([m_], 0.9)
        torch._foreach_add_([m_], [g_], alpha=0.1)
        torch._foreach_mul_([v_], 0.95)
        torch._foreach_addcmul_([v_], [g_], [g_], value=0.05)
        denom = v_.sqrt().add_(1e-8)
        master.addcdiv_(m_, denom, value=-1e-4)
        pb.copy_(master)

    t = _timeit(adam, iters=5)
    traffic = numel * OPTIMIZER_TRAFFIC_BYTES_PER_PARAM
    out["optimizer_eff"] = traffic / (t / 1e3) / (HBM_PEAK_GBPS * GiB)
    out["optimizer_gbps"] = traffic / (t / 1e3) / GiB

    _save(path, out)
    print(json.dumps(out, indent=1), flush=True)
    return out


# --------------------------------------------------------------------------
def enumerate_shapes(cases):
    """Run PerfLLM on (model, strategy-overrides) pairs, CPU-only, collecting
    op shape keys."""
    from simumax_amd import (ModelConfig, PerfLLM, StrategyConfig,
                             SystemConfig, get_simu_model_config,
                             get_simu_system_config)

    gemm, sdp, group = set(), set(), set()
    sysc = SystemConfig.init_from_config_file(get_simu_system_config("mi355x"))
    for model_name, overrides in cases:
        mc = ModelConfig.init_from_config_file(get_simu_model_config(model_name))
        st = StrategyConfig(seq_len=4096, micro_batch_size=1, micro_batch_num=1,
                            world_size=8, attention_sparse_ratio=0.5,
                            enable_recompute=False)
        for k, v in overrides.items():
            setattr(st, k, v)
        p = PerfLLM()
        try:
            p.configure(st, mc, sysc)
            p.run_estimate()
        except AssertionError as e:
            print(f"[enum] skip {model_name} {overrides}: {e}", flush=True)
            continue
        ops = p.analysis_op_info()
        gemm.update(ops.get("matmul", {}))
        sdp.update(ops.get("sdp_fwd", {}))
        sdp.update(ops.get("sdp_bwd", {}))
        group.update(ops.get("group_matmul", {}))
    return sorted(gemm), sorted(sdp), sorted(group)


DEFAULT_CASES = [
    ("llama3-8b", dict(tp_size=1, pp_size=1, enable_sequence_parallel=False,
                       micro_batch_num=4, zero_state=0)),
    # single-GPU validation-sweep shapes
    ("llama3-8b", dict(world_size=1, tp_size=1, pp_size=1, seq_len=2048,
                       micro_batch_size=2, enable_sequence_parallel=False,
                       zero_state=0)),
    ("llama3-8b", dict(world_size=1, tp_size=1, pp_size=1, seq_len=8192,
                       enable_sequence_parallel=False, zero_state=0)),
    ("llama3-70b-l12", dict(world_size=1, tp_size=1, pp_size=1,
                            enable_sequence_parallel=False, zero_state=0)),
    ("llama3-8b", dict(tp_size=2)),
    ("llama3-8b", dict(tp_size=4)),
    ("llama3-8b", dict(tp_size=8)),
    ("llama3-8b", dict(tp_size=1, pp_size=2)),
    ("llama3-70b-l12", dict(tp_size=2, pp_size=2, world_size=8)),
    ("llama3-70b-l12", dict(tp_size=8)),
    ("deepseekv2-l4", dict(ep_size=8, enable_sequence_parallel=False)),
    ("mixtral-8x7b-l8", dict(world_size=1, tp_size=1, pp_size=1, ep_size=1,
                             enable_sequence_parallel=False, zero_state=0)),
    ("deepseekv2-l4", dict(ep_size=4, pp_size=2,
                           enable_sequence_parallel=False)),
    # broader family coverage: every registered model's headline shapes
    ("llama3-8b", dict(world_size=1, tp_size=1, pp_size=1, seq_len=16384,
                       enable_sequence_parallel=False, zero_state=0)),
    ("llama3-8b", dict(world_size=1, tp_size=1, pp_size=1, seq_len=32768,
                       enable_sequence_parallel=False, zero_state=0,
                       enable_recompute=True,
                       recompute_granularity="full_block")),
    ("qwen3-32b-l12", dict(world_size=1, tp_size=1, pp_size=1,
                           enable_sequence_parallel=False, zero_state=0)),
    ("qwen3-32b", dict(tp_size=4)),
    ("qwen3-32b", dict(tp_size=8)),
    ("llama3-405b", dict(tp_size=8)),
    ("gpt3-175b", dict(tp_size=8)),
    ("llama2-7b", dict(tp_size=1, pp_size=1,
                       enable_sequence_parallel=False)),
    ("llama2-70b", dict(tp_size=4)),
    ("deepseekv3", dict(ep_size=8, world_size=8,
                        enable_sequence_parallel=False)),
    # long-context CP (Ulysses a2a) shapes: per-rank seq 8k of a 32k
    # context, heads scattered over cp
    ("llama3-70b-l12", dict(world_size=8, tp_size=1, pp_size=1, cp_size=4,
                            seq_len=32768, cp_comm_type="a2a",
                            enable_sequence_parallel=False)),
    ("llama3-8b", dict(world_size=8, tp_size=1, pp_size=1, cp_size=8,
                       seq_len=65536, cp_comm_type="a2a",
                       enable_sequence_parallel=False)),
]


def main():
    sys.path.insert(0, os.getcwd())
    which = sys.argv[1] if len(sys.argv) > 1 else "all"
    gemm_keys, sdp_keys, group_keys = enumerate_shapes(DEFAULT_CASES)
    print(f"[enum] {len(gemm_keys)} gemm, {len(sdp_keys)} sdp, "
          f"{len(group_keys)} grouped shapes", flush=True)
    if which in ("all", "bw"):
        sweep_bandwidth(os.path.join(OUT_DIR, "bandwidth.json"))
    if which in ("all", "gemm"):
        sweep_gemms(gemm_keys, os.path.join(OUT_DIR, "matmul.json"))
    if which in ("all", "sdp"):
        sweep_sdp(sdp_keys, os.path.join(OUT_DIR, "sdp_fwd.json"),
                  os.path.join(OUT_DIR, "sdp_bwd.json"))
    if which in ("all", "group"):
        sweep_grouped(group_keys, os.path.join(OUT_DIR, "group_matmul.json"))
    if which in ("all", "fp8"):
        sweep_fp8_gemms(gemm_keys, os.path.join(OUT_DIR, "fp8_matmul.json"))
    if which in ("all", "fp8group"):
        sweep_fp8_grouped(group_keys,
                          os.path.join(OUT_DIR, "fp8_group_matmul.json"))
    print("[calib] done", flush=True)


if __name__ == "__main__":
    main()
