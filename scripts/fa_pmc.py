"""Few iterations of the FA kernels for rocprofv3 PMC collection."""
import sys, os, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from simumax_amd.kernels.ops import ext
E = ext()
torch.manual_seed(0)
B, S, Hq, Hkv = 1, 4096, 32, 8
q = torch.randn(B, S, Hq, 128, device="cuda", dtype=torch.bfloat16)
k = torch.randn(B, S, Hkv, 128, device="cuda", dtype=torch.bfloat16)
v = torch.randn(B, S, Hkv, 128, device="cuda", dtype=torch.bfloat16)
o, lse = E.fa_fwd(q, k, v, True)
do = torch.randn_like(o)
torch.cuda.synchronize()
for _ in range(3):
    E.fa_fwd(q, k, v, True)
    E.fa_bwd(do, q, k, v, o, lse, True)
torch.cuda.synchronize()
