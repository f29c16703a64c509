#!/usr/bin/env python3
"""Probe: why is SVM CV slow on fp8-path Gram matrices?

Builds bench-shaped grams via both Z dtypes, then times
cross_validate_voxels at several (tol, max_iter) settings and compares
accuracies.  Run on a GPU box; prints JSON lines.
"""

import json
import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from brainiak_amd import ops
from brainiak_amd.fcma.core import CorrelationPipeline
from brainiak_amd.fcma.svm import stratified_folds


def make_epochs(E, L, V):
    g = torch.Generator(device="cpu").manual_seed(1234)
    out = []
    for _ in range(E):
        m = torch.randn((L, V), generator=g, dtype=torch.float32)
        m = (m - m.mean(0)) / m.std(0, unbiased=False).clamp_min(1e-12)
        out.append(m / math.sqrt(L))
    return out


def time_cv(kernels, labels, tol, max_iter):
    y = np.where(labels == 1, 1.0, -1.0).astype(np.float32)
    folds = stratified_folds(labels, 4)
    E = kernels.shape[1]
    F = len(folds)
    tr = np.zeros((F, E), dtype=np.int32)
    te = np.zeros((F, E), dtype=np.int32)
    ntr = np.zeros(F, dtype=np.int32)
    nte = np.zeros(F, dtype=np.int32)
    for f, (a, b) in enumerate(folds):
        tr[f, :len(a)] = a
        te[f, :len(b)] = b
        ntr[f] = len(a)
        nte[f] = len(b)
    dev = kernels.device
    args = [kernels.float().contiguous(),
            torch.as_tensor(y, device=dev),
            torch.as_tensor(tr, device=dev),
            torch.as_tensor(te, device=dev),
            torch.as_tensor(ntr, device=dev),
            torch.as_tensor(nte, device=dev)]
    ext = ops.load_extension()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    correct = ext.svm_cv(*args, 1.0, tol, max_iter)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    acc = (correct.float() / torch.as_tensor(
        nte, device=dev, dtype=torch.float32)).mean(dim=1).cpu().numpy()
    return dt, acc


def main():
    E, L, V, P = 64, 12, 34470, 4
    raw = make_epochs(E, L, V)
    labels = np.array([e % 2 for e in range(E)])
    grams = {}
    for tag, fp8 in (("bf16", False), ("fp8", True)):
        pipe = CorrelationPipeline(raw, None, P, device="cuda",
                                   z_fp8=fp8)
        grams[tag] = pipe.pipelined_kernel_matrices(
            [(i * 1024, 1024) for i in range(4)]).float()
        del pipe
        torch.cuda.empty_cache()
    d = (grams["bf16"] - grams["fp8"]).abs()
    scale = grams["bf16"].diagonal(dim1=1, dim2=2).mean()
    print(json.dumps({"gram_max_diff": float(d.max()),
                      "gram_mean_diff": float(d.mean()),
                      "diag_scale": float(scale)}), flush=True)
    accs = {}
    for tag in ("bf16", "fp8"):
        for tol, mi in ((1e-3, 10000), (1e-3, 2000), (1e-3, 500),
                        (1e-2, 10000)):
            dt, acc = time_cv(grams[tag], labels, tol, mi)
            accs[(tag, tol, mi)] = acc
            print(json.dumps({"z": tag, "tol": tol, "max_iter": mi,
                              "cv_ms": dt * 1e3,
                              "mean_acc": float(acc.mean())}), flush=True)
    base = accs[("bf16", 1e-3, 10000)]
    for k, v in accs.items():
        agree = float((np.abs(v - base) < 1e-6).mean())
        print(json.dumps({"vs_base": str(k), "frac_identical": agree,
                          "max_acc_diff": float(np.abs(v - base).max())}),
              flush=True)


if __name__ == "__main__":
    main()
