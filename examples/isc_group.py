#!/usr/bin/env python3
"""Group ISC/ISFC with bootstrap significance on synthetic data
(the reference's isc example).  `isfc_distributed` runs the same
computation subject-sharded over RCCL when launched with torchrun."""

import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from brainiak_amd.isc import bootstrap_isc, isc, isfc


def main():
    rng = np.random.RandomState(0)
    trs, voxels, n_subj = 100, 50, 10
    shared = rng.randn(trs, voxels)
    data = np.dstack([0.6 * shared + 0.4 * rng.randn(trs, voxels)
                      for _ in range(n_subj)])

    iscs = isc(data, pairwise=False, summary_statistic=None)
    print(f"leave-one-out ISC: mean={np.nanmean(iscs):.3f}")

    observed, ci, p, distribution = bootstrap_isc(
        iscs, pairwise=False, summary_statistic='median',
        n_bootstraps=200, ci_percentile=95)
    print(f"median ISC={np.nanmean(observed):.3f}, "
          f"p<0.05 voxels: {(p < 0.05).sum()}/{voxels}")

    isfcs, iscs_diag = isfc(data, pairwise=False,
                            summary_statistic='mean')
    print(f"ISFC condensed shape: {isfcs.shape}")


if __name__ == "__main__":
    main()
