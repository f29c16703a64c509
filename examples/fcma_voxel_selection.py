#!/usr/bin/env python3
"""Whole-brain FCMA voxel selection on synthetic data.

Single process uses one GPU (or CPU); scale to 8 GPUs with:
    torchrun --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 \
        examples/fcma_voxel_selection.py
(the reference's equivalent is an mpirun script,
ref docs/examples/fcma/FCMA_script/fcma_voxel_selection_cv.py).
"""

import sys
from pathlib import Path

import numpy as np
from sklearn import svm

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from brainiak_amd.fcma import VoxelSelector
from brainiak_amd.fcma.preprocessing import _separate_epochs
from brainiak_amd.parallel import DistContext


def main():
    ctx = DistContext()
    rng = np.random.RandomState(0)
    n_subjects, epochs_per_subj, voxels, trs = 4, 4, 2000, 64
    epoch_len = trs // epochs_per_subj

    activity, epoch_list = [], []
    for _ in range(n_subjects):
        data = rng.randn(voxels, trs).astype(np.float32)
        spec = np.zeros((2, epochs_per_subj, trs), dtype=np.int32)
        for e in range(epochs_per_subj):
            sl = slice(e * epoch_len, (e + 1) * epoch_len)
            spec[e % 2, e, sl] = 1
            if e % 2 == 0:  # condition-dependent correlation in voxels 0-9
                sig = rng.randn(epoch_len)
                data[:10, sl] += sig * 3.0
        activity.append(data)
        epoch_list.append(spec)

    raw_data, labels = _separate_epochs(activity, epoch_list)
    vs = VoxelSelector(labels, epochs_per_subj=epochs_per_subj,
                       num_folds=4, raw_data=raw_data, comm=ctx)
    clf = svm.SVC(kernel='precomputed', shrinking=False, C=1.0)
    results = vs.run(clf)
    if ctx.is_root:
        print("top 10 voxels:", results[:10])


if __name__ == "__main__":
    main()
