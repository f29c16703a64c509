#!/usr/bin/env python3
"""Distributed searchlight with an ISC block function on synthetic data
(the reference's searchlight example family).

Run on N GPUs/ranks:
    torchrun --nnodes=1 --nproc-per-node N --master-addr 127.0.0.1 \
        examples/searchlight_isc.py
"""

import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from brainiak_amd.searchlight import Ball, Searchlight


def isc_voxel_fn(subjects, mask, sl_rad, bcast):
    """Mean pairwise correlation of the center-sphere mean time course."""
    courses = [s[mask].mean(axis=0) for s in subjects]
    n = len(courses)
    r = [np.corrcoef(courses[i], courses[j])[0, 1]
         for i in range(n) for j in range(i + 1, n)]
    return float(np.mean(r))


def main():
    rng = np.random.RandomState(0)
    dim, trs, n_subj = (15, 15, 15), 40, 4
    shared = rng.randn(*dim, trs)
    subjects = [0.7 * shared + 0.3 * rng.randn(*dim, trs)
                for _ in range(n_subj)]
    mask = np.zeros(dim, dtype=bool)
    mask[3:12, 3:12, 3:12] = True

    sl = Searchlight(sl_rad=2, max_blk_edge=5, shape=Ball)
    sl.distribute(subjects, mask)
    sl.broadcast(None)
    result = sl.run_searchlight(isc_voxel_fn, pool_size=1)
    if result is not None and result[7, 7, 7] is not None:
        vals = [v for v in result[mask] if v is not None]
        print(f"searchlight ISC over {len(vals)} centers: "
              f"mean={np.mean(vals):.3f} (shared signal => high ISC)")


if __name__ == "__main__":
    main()
