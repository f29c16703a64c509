"""A/B the adjacent-pair raw corr kernel (BRAINIAK_CORR_VPT=3).

Correctness: raw Z vs a torch einsum reference.  Timing: raw corr
kernel alone at the flagship shape.
"""
import os, sys, time, torch
from brainiak_amd.ops import hip_ops as ops

torch.manual_seed(0)
dev = "cuda:0"
L, S, P, VB, C = 12, 16, 4, 34480, 512
E = S * P
A = torch.randn(C, E, L, device=dev, dtype=torch.bfloat16)
B = torch.randn(S * P, L, VB, device=dev, dtype=torch.bfloat16)
Zout = torch.empty(C, E, VB, device=dev, dtype=torch.bfloat16)

vpt = os.environ.get("BRAINIAK_CORR_VPT", "1")
ops.fcma_corr_norm_z(A, B, Zout, True)   # raw=True
torch.cuda.synchronize()
# reference: raw dot products per (c, s, p, v)
Bf = B.float().view(S, P, L, VB)
Af = A.float().view(C, S, P, L)
ref = torch.einsum('cspl,splv->cspv', Af, Bf).view(C, E, VB)
err = (Zout.float() - ref).abs().max().item()
print(f"vpt={vpt} max|err|={err:.4f}")
assert err < 0.25, err

n = 30
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(n):
    ops.fcma_corr_norm_z(A, B, Zout, True)
torch.cuda.synchronize()
print(f"vpt={vpt} raw-corr {1e3*(time.perf_counter()-t0)/n:.3f} ms")
