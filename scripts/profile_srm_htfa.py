#!/usr/bin/env python3
"""GPU-scale SRM and HTFA profiling.

Run under rocprofv3 for kernel stats, or bare for wall-clock splits:

    python scripts/profile_srm_htfa.py srm    # 16 subj x 50k vox, K=50
    python scripts/profile_srm_htfa.py htfa   # 100k vox x 2 subj local

Emits one JSON line per phase with wall-clock decomposition; the HTFA
run instruments the scipy-NLSS (CPU) vs kernel (GPU) split.
"""

import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def profile_srm():
    from brainiak_amd.funcalign.srm import SRM
    subjects, V, T, K = 16, 50000, 200, 50
    rng = np.random.RandomState(0)
    S = rng.randn(K, T)
    data = []
    for _ in range(subjects):
        W = np.linalg.qr(rng.randn(V, K))[0]
        data.append((W @ S + 0.1 * rng.randn(V, T)).astype(np.float64))
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    m = SRM(n_iter=10, features=K, rand_seed=0, device="cuda")
    m.fit(data)
    torch.cuda.synchronize()
    total = time.perf_counter() - t0
    print(json.dumps({
        "phase": "srm_gpu_scale",
        "subjects": subjects, "voxels": V, "trs": T, "features": K,
        "n_iter": 10, "total_s": total, "s_per_iter": total / 10,
    }), flush=True)


def profile_htfa():
    """HTFA at benchmark scale, with the TFA inner solve instrumented:
    how much of an iteration is scipy trf (CPU) vs the N8/N9 kernels."""
    from brainiak_amd.factoranalysis import tfa as tfa_mod
    from brainiak_amd.factoranalysis.htfa import HTFA
    from brainiak_amd.parallel import DistContext

    V, T, K, subjects = 100000, 100, 20, 2   # 2 local subjects, 1 GPU
    rng = np.random.RandomState(0)
    X, R = [], []
    for _ in range(subjects):
        coords = rng.rand(V, 3) * 40
        centers = rng.rand(K, 3) * 40
        d2 = ((coords[:, None, :] - centers[None, :, :]) ** 2).sum(-1)
        F = np.exp(-d2 / 50.0)
        W = rng.randn(K, T)
        X.append((F @ W + 0.1 * rng.randn(V, T)))
        R.append(coords)

    # monkeypatch-time the NLSS solve and the kernel-facing residual
    from scipy.optimize import least_squares as _ls
    acc = {"nlss_s": 0.0, "nlss_calls": 0, "resid_s": 0.0,
           "resid_calls": 0}
    orig_resid = tfa_mod.TFA._residual_multivariate
    orig_jac = tfa_mod.TFA._jacobian_multivariate

    def timed_ls(fun, x0, **kw):
        t0 = time.perf_counter()
        out = _ls(fun, x0, **kw)
        acc["nlss_s"] += time.perf_counter() - t0
        acc["nlss_calls"] += 1
        return out

    def timed_resid(self, *a, **k):
        t0 = time.perf_counter()
        out = orig_resid(self, *a, **k)
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        acc["resid_s"] += time.perf_counter() - t0
        acc["resid_calls"] += 1
        return out

    tfa_mod.least_squares = timed_ls
    tfa_mod.TFA._residual_multivariate = timed_resid

    ctx = DistContext(device="cuda")
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    htfa = HTFA(K=K, n_subj=subjects, max_global_iter=2,
                max_local_iter=2, comm=ctx, device="cuda")
    htfa.fit(X, R)
    torch.cuda.synchronize()
    total = time.perf_counter() - t0
    tfa_mod.least_squares = _ls
    tfa_mod.TFA._residual_multivariate = orig_resid
    del orig_jac
    print(json.dumps({
        "phase": "htfa_profile",
        "voxels": V, "trs": T, "K": K, "local_subjects": subjects,
        "global_iters": 2, "local_iters": 2,
        "total_s": total,
        "nlss_s": acc["nlss_s"], "nlss_calls": acc["nlss_calls"],
        "resid_in_nlss_s": acc["resid_s"],
        "resid_calls": acc["resid_calls"],
        "nlss_frac": acc["nlss_s"] / total,
    }), flush=True)


if __name__ == "__main__":
    which = sys.argv[1] if len(sys.argv) > 1 else "srm"
    if which == "srm":
        profile_srm()
    else:
        profile_htfa()
