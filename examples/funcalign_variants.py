#!/usr/bin/env python3
"""The non-probabilistic functional-alignment family on one synthetic
dataset: deterministic SRM, robust SRM (RSRM), semi-supervised SRM
(SSSRM) and atlas-projected FastSRM.

    python examples/funcalign_variants.py
"""

import sys
import tempfile
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from brainiak_amd.funcalign.fastsrm import FastSRM
from brainiak_amd.funcalign.rsrm import RSRM
from brainiak_amd.funcalign.srm import DetSRM
from brainiak_amd.funcalign.sssrm import SSSRM


def make_subjects(rng, subjects=4, voxels=400, trs=120, features=20):
    S = rng.randn(features, trs)
    data = []
    for _ in range(subjects):
        q, _ = np.linalg.qr(rng.randn(voxels, features))
        data.append(q @ S + 0.1 * rng.randn(voxels, trs))
    return data


def main():
    rng = np.random.RandomState(0)
    data = make_subjects(rng)
    trs = data[0].shape[1]

    det = DetSRM(n_iter=8, features=20).fit(data)
    print("DetSRM shared response:", det.s_.shape)

    rob = RSRM(n_iter=8, features=20).fit(data)
    print("RSRM shared response:", rob.r_.shape,
          "sparse residual nnz:", int((np.abs(rob.s_[0]) > 0).sum()))

    # SSSRM: half the TRs labeled into 3 conditions
    labels = [np.tile(np.arange(3), trs // 6)[:trs // 2]
              for _ in data]
    labeled = [d[:, :trs // 2] for d in data]
    semi = SSSRM(n_iter=4, features=20, gamma=1.0, alpha=0.5)
    semi.fit(data, labels, labeled)
    preds = semi.predict(labeled)
    acc = np.mean([np.mean(p == y) for p, y in zip(preds, labels)])
    print("SSSRM classifier accuracy on train:", round(float(acc), 3))

    with tempfile.TemporaryDirectory() as tmp:
        fast = FastSRM(n_components=20, n_iter=8, temp_dir=tmp,
                       aggregate="mean")
        fast.fit([[d] for d in data])
        shared = fast.transform([[d] for d in data])
        shared = np.asarray(shared)
        print("FastSRM aggregated shared response:", shared.shape)


if __name__ == "__main__":
    main()
