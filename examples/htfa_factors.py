#!/usr/bin/env python3
"""(H)TFA: topographic factor analysis on synthetic RBF data
(the reference's htfa example).  HTFA distributes subjects over ranks:
    torchrun --nnodes=1 --nproc-per-node N --master-addr 127.0.0.1 \
        examples/htfa_factors.py
"""

import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from brainiak_amd.factoranalysis import TFA


def main():
    rng = np.random.RandomState(0)
    K, n_vox, trs = 4, 500, 60
    coords = rng.rand(n_vox, 3) * 30
    centers = rng.rand(K, 3) * 30
    widths = np.full((K, 1), 20.0)
    d2 = ((coords[:, None, :] - centers[None, :, :]) ** 2).sum(-1)
    F = np.exp(-d2 / widths.ravel()[None, :])
    X = F @ rng.randn(K, trs) + 0.1 * rng.randn(n_vox, trs)

    tfa = TFA(K=K, max_iter=8, max_num_voxel=n_vox, max_num_tr=trs,
              verbose=False)
    tfa.fit(X, coords)
    est = tfa.get_centers(tfa.local_posterior_)
    from scipy.spatial.distance import cdist
    match = cdist(centers, est).min(axis=1)
    print("true-to-recovered center distances:",
          [round(d, 1) for d in match])
    recon = tfa.F_ @ tfa.W_
    print("reconstruction corr:",
          round(np.corrcoef(recon.ravel(), X.ravel())[0, 1], 3))


if __name__ == "__main__":
    main()
