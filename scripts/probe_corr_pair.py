"""A/B the adjacent-pair raw corr kernel (BRAINIAK_CORR_VPT=3).

Correctness: raw Z vs a torch einsum reference.  Timing: raw corr
launch alone at the flagship shape.
"""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from brainiak_amd import ops
ext = ops.load_extension()

torch.manual_seed(0)
dev = "cuda:0"
L, S, P, VB, count = 12, 16, 4, 34480, 512
E = S * P
A = torch.randn(E, L, VB, device=dev, dtype=torch.bfloat16)
vpt = os.environ.get("BRAINIAK_CORR_VPT", "1")
Z = ext.fcma_corr_norm_z(A, A, 0, count, P, E, None, True)
torch.cuda.synchronize()
ref = torch.einsum('elc,elv->cev', A[:, :, :count].float(), A.float())
err = (Z.float() - ref).abs().max().item()
print(f"vpt={vpt} max|err|={err:.4f}")
assert err < 0.25, err

n = 30
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(n):
    ext.fcma_corr_norm_z(A, A, 0, count, P, E, None, True)
torch.cuda.synchronize()
print(f"vpt={vpt} raw-corr {1e3*(time.perf_counter()-t0)/n:.3f} ms")
