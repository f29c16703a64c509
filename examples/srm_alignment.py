#!/usr/bin/env python3
"""Probabilistic SRM functional alignment on synthetic data.

Distributed across GPUs with rank-cyclic subject ownership:
    torchrun --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 \
        examples/srm_alignment.py
"""

import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from brainiak_amd.funcalign import SRM
from brainiak_amd.parallel import DistContext


def main():
    ctx = DistContext()
    rng = np.random.RandomState(42)
    subjects, voxels, trs, features = 8, 1000, 200, 50
    S = rng.randn(features, trs)
    data = []
    for i in range(subjects):
        q, _ = np.linalg.qr(rng.randn(voxels, features))
        subj = q @ S + 0.1 * rng.randn(voxels, trs)
        data.append(subj if i % ctx.world_size == ctx.rank else None)

    model = SRM(n_iter=10, features=features, comm=ctx).fit(data)
    if ctx.is_root:
        shared = model.transform(data)
        owned = [s for s in shared if s is not None]
        print("shared response:", model.s_.shape,
              "first owned projection:", owned[0].shape)


if __name__ == "__main__":
    main()
