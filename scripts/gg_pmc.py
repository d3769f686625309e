import sys
sys.path.insert(0, "/root/repo")
import torch
from simumax_amd.kernels.ops import ext
E_ = ext()
E, M, N, K = 8, 1024, 28672, 4096
x = torch.randn(E, M, K, device="cuda", dtype=torch.bfloat16) / 8
w = torch.randn(E, N, K, device="cuda", dtype=torch.bfloat16) / 8
dout = torch.randn(E, M, N, device="cuda", dtype=torch.bfloat16) / 8
for _ in range(3):
    E_.grouped_fwd(x, w)
    E_.grouped_dgrad(dout, w)
torch.cuda.synchronize()
