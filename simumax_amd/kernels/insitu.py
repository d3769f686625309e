"""In-situ operator calibration: hipEvent timing of the trainer's real
op calls, keyed by the simulator's exact shape-key strings.

Microbenchmarks (calib/sweeps.py) time 10 identical iterations with hot
L2 and burst clocks; inside a training step the same op runs in a mixed
kernel stream. Enabling this registry during real steps produces
per-shape-key efficiency tables measured UNDER TRAINING CONDITIONS,
which calib/merge.py merges with precedence over the microbench values
(reference analog: SimuMax's benchmark suite measuring operator
efficiency on the target model's own shapes).

Usage (see scripts/insitu_calib.py):
    from simumax_amd.kernels import insitu
    insitu.enable()
    ... run training steps ...
    insitu.dump("gpurun_out/calib")        # *_insitu.json sweep files
"""

from __future__ import annotations

import json
import os
from collections import defaultdict

import torch

ENABLED = False
_RECORDS = defaultdict(list)  # (table, key) -> [(start_evt, end_evt), ...]

PEAK_BF16_TFLOPS = 2500.0


def enable():
    global ENABLED
    ENABLED = True
    _RECORDS.clear()


def disable():
    global ENABLED
    ENABLED = False


def start(table: str, key: str):
    """Record an event pair around the op the caller is about to launch.
    Returns a closure to call right after the launch (same stream).

    Each instance is tagged with whether it ran inside a backward graph
    task: activation-recompute reruns (torch.utils.checkpoint) execute
    the SAME fwd-shaped ops during .backward(), and summarize() must not
    let those duplicates shift the fwd keys (the simulator prices the
    rerun separately via recompute_factor)."""
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    in_bwd = torch._C._current_graph_task_id() != -1
    s.record()

    def stop():
        e.record()
        _RECORDS[(table, key)].append((s, e, in_bwd))

    return stop


def gemm_key(b, m, k, n, layout, accumulate, out_dtype):
    return (f"b={b}, m={m}, k={k}, n={n}, layout={layout}, "
            f"accumulate={accumulate}, out_dtype={out_dtype}")


def sdp_key(b, s, hq, hkv, dqk, dv, contiguous=True):
    return (f"batch={b}, seq_len={s}, head_num={hq}, kv_head_num={hkv}, "
            f"qk_head_dim={dqk}, v_head_dim={dv}, "
            f"qkv_contiguous={contiguous}")


def _gemm_flops(key):
    import re

    m = re.match(r"b=(\d+), m=(\d+), k=(\d+), n=(\d+)", key)
    b, mm, k, n = (int(m.group(i)) for i in range(1, 5))
    return 2 * b * mm * k * n


def _sdp_flops(key, stage):
    import re

    m = re.match(
        r"batch=(\d+), seq_len=(\d+), head_num=(\d+), kv_head_num=(\d+), "
        r"qk_head_dim=(\d+), v_head_dim=(\d+)", key)
    b, s, hq, _, dqk, dv = (int(m.group(i)) for i in range(1, 7))
    qk = 2 * b * hq * s * s * dqk
    pv = 2 * b * hq * s * s * dv
    causal = 0.5
    if stage == "fwd":
        return (qk + pv) * causal
    return (3 * qk + 2 * pv) * causal  # bwd recomputes QK^T


HBM_PEAK_GBPS = 8000.0


def summarize():
    """Sync and reduce recorded event pairs -> {table: {key: stats}}.
    `bw_<op>` tables carry modeled byte counts as the key; their summary
    is one stream efficiency per op (total bytes / total time / peak)."""
    torch.cuda.synchronize()
    out = defaultdict(dict)
    bw_acc = defaultdict(lambda: [0.0, 0.0, 0])  # op -> [bytes, ms, n]
    # measured checkpoint-rerun cost ratio: for every key seen both
    # outside and inside backward, the in-backward instances are the
    # recompute reruns — their per-instance time over the fwd instance
    # time, weighted by rerun count, IS the accelerator recompute_factor
    rc_wsum = fwd_wsum = 0.0
    saw_compute_rerun = False  # bw_ ops reuse keys across fwd/bwd (e.g.
    # rope), so only compute-table mixed keys prove a recompute ran
    for (table, key), pairs in _RECORDS.items():
        if table.startswith("bw_"):
            acc = bw_acc[table[3:]]
            f_ms = f_n = r_ms = r_n = 0
            for s, e, inb in pairs:
                t = s.elapsed_time(e)
                if inb:
                    r_ms += t
                    r_n += 1
                else:
                    f_ms += t
                    f_n += 1
            if f_n and r_n:
                # mixed context = fwd op rerun under recompute: price the
                # bandwidth from true-fwd instances only, feed the ratio
                # into the measured recompute factor
                fwd_wsum += (f_ms / f_n) * r_n
                rc_wsum += r_ms
                acc[0] += float(key) * f_n
                acc[1] += f_ms
                acc[2] += f_n
            else:
                acc[0] += float(key) * len(pairs)
                acc[1] += f_ms + r_ms
                acc[2] += len(pairs)
            continue
        # a key seen both outside and inside backward is a fwd op whose
        # checkpoint-recompute rerun shares the shape key: keep only the
        # true-fwd instances (pure-bwd keys keep everything)
        fwd_p = [p for p in pairs if not p[2]]
        rc_p = [p for p in pairs if p[2]]
        if fwd_p and rc_p:
            m_f = sorted(s.elapsed_time(e) for s, e, _ in fwd_p)
            m_r = sorted(s.elapsed_time(e) for s, e, _ in rc_p)
            fwd_wsum += m_f[len(m_f) // 2] * len(rc_p)
            rc_wsum += m_r[len(m_r) // 2] * len(rc_p)
            saw_compute_rerun = True
            pairs = fwd_p
        ts = sorted(s.elapsed_time(e) for s, e, _ in pairs)
        t = ts[len(ts) // 2]  # median instance
        row = dict(t_ms=round(t, 5), n=len(ts))
        if table == "matmul":
            row["eff"] = _gemm_flops(key) / (t / 1e3) / (PEAK_BF16_TFLOPS * 1e12)
        elif table == "fp8_matmul":
            row["eff"] = _gemm_flops(key) / (t / 1e3) / (2 * PEAK_BF16_TFLOPS * 1e12)
        elif table in ("sdp_fwd", "sdp_bwd"):
            stage = "fwd" if table == "sdp_fwd" else "bwd"
            row["eff"] = _sdp_flops(key, stage) / (t / 1e3) / (PEAK_BF16_TFLOPS * 1e12)
        elif table == "group_matmul":
            import re

            m = re.match(r"ng=(\d+), M=(\d+), N=(\d+), K=(\d+)", key)
            ng, mm, n, k = (int(m.group(i)) for i in range(1, 5))
            row["eff"] = (2 * ng * mm * n * k) / (t / 1e3) / (PEAK_BF16_TFLOPS * 1e12)
        out[table][key] = row
    for op, (byt, ms, n) in bw_acc.items():
        # subtract launch latency so the efficiency composes with the
        # model's additive latency_us term
        ms = max(ms - n * 0.004, 1e-6)
        out["bandwidth"][f"{op}_eff"] = round(
            byt / (ms / 1e3) / (HBM_PEAK_GBPS * 1024**3), 4)
    if saw_compute_rerun and fwd_wsum > 0 and rc_wsum > 0:
        out["meta"]["recompute_factor"] = round(rc_wsum / fwd_wsum, 4)
    return out


def dump(outdir):
    os.makedirs(outdir, exist_ok=True)
    summary = summarize()
    # matmul_insitu.json matches the sweep-file schema {desc: {eff, t_ms}}
    for table, rows in summary.items():
        path = os.path.join(outdir, f"{table}_insitu.json")
        existing = {}
        if os.path.exists(path):
            with open(path) as f:
                existing = json.load(f)
        existing.update(rows)
        with open(path, "w") as f:
            json.dump(existing, f, indent=1, sort_keys=True)
        print(f"[insitu] {path}: {len(rows)} keys")
    _RECORDS.clear()
    return summary
