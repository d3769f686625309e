"""Global constants and environment flags for the MI355X simulator core.

Reference parity: env flags mirror simumax/core/config.py:16-25 in the
upstream SimuMax (SIMU_CHECK / SIMU_DEBUG / SIMUMAX_TMP_PATH /
ENABLE_SIMU_GRAPH); numbers here are CDNA4 (gfx950) facts, not CUDA ones.
"""

import os

SIMU_DEBUG = os.getenv("SIMU_DEBUG", "0") == "1"
SIMU_CHECK = os.getenv("SIMU_CHECK", "0") == "1"
ENABLE_SIMU_GRAPH = os.getenv("ENABLE_SIMU_GRAPH", "0") == "1"
TMP_PATH = os.getenv("SIMUMAX_TMP_PATH", "tmp_check" if SIMU_CHECK else "tmp")

# bytes per element by dtype tag
DTYPE_BYTES = {
    "fp64": 8,
    "fp32": 4,
    "float32": 4,
    "tf32": 4,
    "bf16": 2,
    "fp16": 2,
    "half": 2,
    "fp8": 1,
    "e4m3": 1,
    "e5m2": 1,
    "int8": 1,
    "uint8": 1,
    "int32": 4,
    "int64": 8,
    "bool": 1,
}

# Collective op vocabulary priced by the network model.
NET_OPS = ("all_reduce", "all_gather", "reduce_scatter", "all2all", "p2p")

# MI355X (gfx950) hardware facts used for starter configs and sanity checks.
# Dense MFMA peaks (AMD spec figures with 2:1 sparsity removed); measured
# microbenchmark ceilings in parentheses per /opt/skills guides.
MI355X = {
    "arch": "gfx950",
    "num_cus": 256,
    "num_xcds": 8,
    "hbm_gb": 288,
    "hbm_peak_gbps": 8000.0,       # spec; ~6300 GB/s achievable (79%)
    "bf16_dense_tflops": 2500.0,   # measured ceiling ~2495
    "fp8_dense_tflops": 5000.0,    # MX-scaled path; measured ~4647
    "fp32_tflops": 157.3,
    "xgmi_links": 7,
    "xgmi_link_gbps": 153.0,       # per link, per direction
    "lds_kib_per_cu": 160,
}

# Mixed-precision Adam HBM traffic per parameter (bytes), matching the
# trainer's exact step sequence (train/trainer.py MixedPrecisionAdam):
# zero_grad(w4) + grad-norm(r4) + m update(r4 rw8) + v update(r4 rw8)
# + denom sqrt/add (r4 w4 rw8) + addcdiv (r4 r4 rw8) + bf16 copy (r4 w2).
# The calibration sweep times the same sequence, so the efficiency factor
# is consistent by construction.
# measured (torch.profiler, mixtral-8x7b-l8 11.9B params, 2026-09): the
# flat mixed-precision Adam chain (zero_grad, chunked grad-norm dot, clip,
# m/v updates, chunked denom, addcdiv, param copy) moves ~76 B/param at the
# calibrated optimizer-stream efficiency
OPTIMIZER_TRAFFIC_BYTES_PER_PARAM = 76
