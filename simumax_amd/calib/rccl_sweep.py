"""RCCL collective sweeps over the xGMI mesh + alpha-beta fits.

Parity target: simu_tools/efficency_test/nccl_test.sh + nccl_fit.py +
one_click_common.py:fit_bw_latency (245-260) and
measure_comm_burst_window_worker.py — rebuilt on torch.distributed
(backend "nccl" IS RCCL on ROCm).

Launch on an N-GPU node (one rank per GPU):
    python -m torch.distributed.run --standalone --nproc-per-node N \
        -m simumax_amd.calib.rccl_sweep
Writes gpurun_out/calib/rccl_ws{N}.json on rank 0. Run at N in {2,4,8}
to populate the per-comm_num efficiency/latency tables, then
`python -m simumax_amd.calib.merge_rccl` folds the fits into
configs/system/mi355x.json.

The alpha-beta fit convention matches the cost model exactly:
    time = actual/(bw*eff) + latency,
    actual = size*scale + size*scale/n*offset
(ring all_reduce scale=2 offset=-1; ag/rs/a2a 1/-1; p2p sendrecv 1/0), so
eff is fitted against the FC8 xGMI bandwidth (n-1)/7 * 7*153 GB/s.
"""

from __future__ import annotations

import json
import os
import time

XGMI_LINK_GBPS = 153.0
NUM_LINKS = 7

SIZES = [2**i for i in range(20, 34)]  # 1 MiB .. 8 GiB payload bytes
OPS = ("all_reduce", "all_gather", "reduce_scatter", "all2all", "p2p")


def _time_collective(dist, torch, op, size_bytes, group=None, iters=10):
    n = dist.get_world_size(group)
    rank = dist.get_rank(group)
    dev = torch.cuda.current_device()
    elem = size_bytes // 2
    x = torch.empty(elem, dtype=torch.bfloat16, device=dev)

    if op == "all_reduce":
        fn = lambda: dist.all_reduce(x, group=group)
    elif op == "all_gather":
        out = torch.empty(elem, dtype=torch.bfloat16, device=dev)
        shard = x[: elem // n]
        outs = list(out.chunk(n))
        fn = lambda: dist.all_gather(outs, shard, group=group)
    elif op == "reduce_scatter":
        out = torch.empty(elem // n, dtype=torch.bfloat16, device=dev)
        fn = lambda: dist.reduce_scatter_tensor(out, x, group=group)
    elif op == "all2all":
        out = torch.empty_like(x)
        fn = lambda: dist.all_to_all_single(out, x, group=group)
    elif op == "p2p":
        peer = rank ^ 1
        if peer >= n:
            return None

        def fn():
            if rank % 2 == 0:
                dist.send(x, peer, group=group)
                dist.recv(x, peer, group=group)
            else:
                dist.recv(x, peer, group=group)
                dist.send(x, peer, group=group)
    else:
        raise ValueError(op)

    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    dist.barrier(group)
    t0 = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    dist.barrier(group)
    t = (time.time() - t0) / iters * 1e3  # ms
    if op == "p2p":
        t /= 2  # one direction of the ping-pong
    return t


def fit_bw_latency(rows, scale, offset, n):
    """Least-squares fit time = actual/bw + lat over (size, ms) rows;
    returns (bw GiB/s, latency ms). Reference parity:
    one_click_common.py:fit_bw_latency."""
    import numpy as np

    xs = np.array([r[0] * scale + r[0] * scale / n * offset for r in rows],
                  dtype=float) / 1024**3
    ys = np.array([r[1] / 1e3 for r in rows], dtype=float)  # seconds
    A = np.vstack([xs, np.ones_like(xs)]).T
    slope, intercept = np.linalg.lstsq(A, ys, rcond=None)[0]
    bw = 1.0 / slope if slope > 0 else float("inf")
    return bw / 1024**3 * 1024**3, max(intercept, 0.0) * 1e3  # GiB/s, ms


SCALE_OFFSET = {
    "all_reduce": (2, -1),
    "all_gather": (1, -1),
    "reduce_scatter": (1, -1),
    "all2all": (1, -1),
    "p2p": (1, 0),
}


def main():
    import torch
    import torch.distributed as dist

    dist.init_process_group("nccl")
    rank = dist.get_rank()
    n = dist.get_world_size()
    torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)))
    out = {"world_size": n, "ops": {}}
    for op in OPS:
        rows = []
        for size in SIZES:
            t = _time_collective(dist, torch, op, size)
            if t is None:
                break
            if rank == 0:
                rows.append((size, t))
                print(f"[rccl] ws{n} {op} {size>>20} MiB: {t:.3f} ms",
                      flush=True)
        if rank == 0 and rows:
            scale, offset = SCALE_OFFSET[op]
            bw, lat = fit_bw_latency(rows[3:], scale, offset, n)
            fc8_bw = (n - 1) / NUM_LINKS * NUM_LINKS * XGMI_LINK_GBPS
            out["ops"][op] = {
                "rows": rows, "fit_bw_gibps": bw, "fit_latency_ms": lat,
                "efficient_factor": bw / fc8_bw if op != "p2p"
                else bw / XGMI_LINK_GBPS,
            }
    if rank == 0:
        os.makedirs("gpurun_out/calib", exist_ok=True)
        with open(f"gpurun_out/calib/rccl_ws{n}.json", "w") as f:
            json.dump(out, f, indent=1)
        print(json.dumps({k: {kk: vv for kk, vv in v.items() if kk != "rows"}
                          for k, v in out["ops"].items()}, indent=1))
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
