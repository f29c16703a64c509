#!/usr/bin/env python3
"""FCMA stage 2: train and test the correlation-space Classifier
(precomputed-kernel SVM) on synthetic epochs — the counterpart of the
reference's fcma_classification example.

    python examples/fcma_classification.py
"""

import sys
from pathlib import Path

import numpy as np
from sklearn import svm

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from brainiak_amd.fcma.classifier import Classifier


def make_epochs(rng, n_epochs, trs, voxels, cond):
    """Two conditions distinguished by a planted correlation block."""
    out = []
    for c in cond:
        e = rng.randn(trs, voxels).astype(np.float32)
        if c == 1:
            driver = rng.randn(trs, 1).astype(np.float32)
            e[:, : voxels // 4] += 1.5 * driver
        e = (e - e.mean(0)) / e.std(0)
        out.append(e / np.sqrt(trs))
    return out


def main():
    rng = np.random.RandomState(7)
    trs, voxels = 24, 60
    labels = np.tile([0, 1], 12)
    epochs = make_epochs(rng, len(labels), trs, voxels, labels)
    samples = [(e, e) for e in epochs]     # self-correlation features

    clf = Classifier(svm.SVC(kernel='precomputed', shrinking=False,
                             C=1.0), epochs_per_subj=4)
    train_n = 16
    clf.fit(samples, labels, num_training_samples=train_n)
    acc = clf.score(None, labels[train_n:])
    print("held-out accuracy (%d test epochs): %.3f"
          % (len(labels) - train_n, acc))


if __name__ == "__main__":
    main()
