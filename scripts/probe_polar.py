"""A/B: ops.polar_invsqrt (one-wave Jacobi) vs torch.linalg.eigh for
the SRM Procrustes G^{-1/2} at [S, K, K]."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from brainiak_amd import ops

torch.manual_seed(0)
dev = "cuda"
for S, K in ((16, 50), (64, 50), (16, 64)):
    A = torch.randn(S, K, K, device=dev)
    G = A @ A.transpose(1, 2) + 0.01 * torch.eye(K, device=dev)
    r1 = ops.polar_invsqrt(G.contiguous())
    w, v = torch.linalg.eigh(G)
    r2 = (v * w.clamp_min(1e-12).rsqrt()[:, None, :]) @ v.transpose(1, 2)
    err = (r1 - r2).abs().max().item()
    for name, fn in (
        ("jacobi", lambda: ops.polar_invsqrt(G.contiguous())),
        ("eigh  ", lambda: torch.linalg.eigh(G)),
    ):
        torch.cuda.synchronize(); t0 = time.perf_counter()
        for _ in range(20):
            fn()
        torch.cuda.synchronize()
        print(f"S={S} K={K} {name} {1e3*(time.perf_counter()-t0)/20:.3f} ms", 
              f"(maxdiff {err:.2e})" if name == "jacobi" else "")
