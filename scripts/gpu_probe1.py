"""First-GPU-call probe: MFMA layout check + kernel micro-verification.
Writes results to gpurun_out/probe1.log (stdout captured by wrapper)."""
import sys, os, json
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from simumax_amd.kernels.ops import ext

torch.manual_seed(0)
E = ext()
dev = "cuda:0"

# --- MFMA layout probe ---
A = (torch.randint(-3, 4, (16, 32), device=dev).to(torch.bfloat16))
B = (torch.randint(-3, 4, (32, 16), device=dev).to(torch.bfloat16))
cm, cr = E.mfma_probe(A.contiguous(), B.contiguous(), 0)
ref = (A.float() @ B.float())
err = (cm - ref).abs().max().item()
print("probe mode0 (assumed layouts) max err:", err)
if err > 1e-3:
    print("LAYOUT MISMATCH — dumping diagnostics")
    cm1, cr1 = E.mfma_probe(A, B, 1)
    cm2, cr2 = E.mfma_probe(A, B, 2)
    print("mode1 raw (A-id, B-ones):"); print(cr1.cpu().numpy().tolist())
    print("mode2 raw (A-ones, B-id):"); print(cr2.cpu().numpy().tolist())
    print("mode0 mapped:"); print(cm.cpu().numpy().tolist())
    print("mode0 ref:"); print(ref.cpu().numpy().tolist())
else:
    print("MFMA 16x16x32 bf16 layout CONFIRMED")
