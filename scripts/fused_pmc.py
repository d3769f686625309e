"""Few iterations of the fused memory-bound kernels (rmsnorm fwd/bwd,
rope, swiglu, cross-entropy) for rocprofv3 PMC collection — the round-3
starting point for the rmsnorm_bwd eff-0.04 anomaly (ROADMAP §4).

  cd /tmp && export TMPDIR=/tmp && rocprofv3 --pmc SQ_WAVE_CYCLES \
      SQ_WAIT_ANY SQ_ACTIVE_INST_ANY SQ_LDS_BANK_CONFLICT \
      --output-format csv -d $GRAFT_REPO_ROOT/gpurun_out/prof -- \
      python $GRAFT_REPO_ROOT/scripts/fused_pmc.py
"""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from simumax_amd.kernels.ops import build_rope_cache, ext

E = ext()
torch.manual_seed(0)
x = torch.randn(4096, 4096, device="cuda", dtype=torch.bfloat16)
w = torch.randn(4096, device="cuda", dtype=torch.bfloat16)
y, rstd = E.rmsnorm_fwd(x, w, 1e-5)
dy = torch.randn_like(x)
q = torch.randn(4096, 32, 128, device="cuda", dtype=torch.bfloat16)
cs = build_rope_cache(8192, 128, device="cuda")
pos = torch.arange(4096, dtype=torch.int32, device="cuda")
g = torch.randn(4096, 28672, device="cuda", dtype=torch.bfloat16)
logits = torch.randn(4096, 128256, device="cuda", dtype=torch.bfloat16)
tgt = torch.randint(0, 128256, (4096,), device="cuda", dtype=torch.int64)
torch.cuda.synchronize()
for _ in range(3):
    E.rmsnorm_fwd(x, w, 1e-5)
    E.rmsnorm_bwd(dy, x, w, rstd)
    E.rope(q, cs, pos, 1.0)
    E.swiglu_fwd(g)
    E.ce_fwd(logits, tgt)
torch.cuda.synchronize()
print("fused kernels profiled")
