"""Time the FA kernels at the llama3-8b shape and print TF/s."""
import sys, os, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from simumax_amd.kernels.ops import ext
from simumax_amd.calib.sweeps import _timeit
E = ext()
torch.manual_seed(0)
for (B,S,Hq,Hkv) in [(1,4096,32,8),(4,4096,32,8),(1,8192,32,8)]:
    q = torch.randn(B,S,Hq,128, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B,S,Hkv,128, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B,S,Hkv,128, device="cuda", dtype=torch.bfloat16)
    o, lse = E.fa_fwd(q,k,v,True)
    do = torch.randn_like(o)
    tf = _timeit(lambda: E.fa_fwd(q,k,v,True), iters=10)
    tb = _timeit(lambda: E.fa_bwd(do,q,k,v,o,lse,True), iters=5)
    fl = 2*2*B*Hq*S*S*128*0.5
    flb = fl*2.5
    print(f"B{B} S{S} Hq{Hq}: fwd {tf:.2f} ms = {fl/tf/1e9:.0f} TF/s | bwd {tb:.2f} ms = {flb/tb/1e9:.0f} TF/s")
