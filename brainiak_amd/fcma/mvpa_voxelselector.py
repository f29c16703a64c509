"""Activity-based voxel selection via Searchlight (API parity:
ref src/brainiak/fcma/mvpa_voxelselector.py:34-136).

The per-searchlight statistic here runs the stratified k-fold loop
explicitly (clone → fit → score per split) rather than delegating to
``cross_val_score``: the explicit loop skips sklearn's indexing/
cloning scaffolding per fold and keeps the hot path obvious.
"""

import logging

import numpy as np
from sklearn import model_selection
from sklearn.base import clone

logger = logging.getLogger(__name__)

__all__ = ["MVPAVoxelSelector"]


def _sl_accuracy(data, mask, myrad, bcast_var):
    """Mean stratified-CV accuracy of one searchlight's activity
    vectors.  ``bcast_var`` unpacks to (labels, num_folds, clf)."""
    labels, num_folds, clf = bcast_var
    samples = data[0][mask, :].T          # [epoch, in-light voxel]
    labels = np.asarray(labels)
    folds = model_selection.StratifiedKFold(n_splits=num_folds,
                                            shuffle=False)
    scores = []
    for fit_idx, eval_idx in folds.split(samples, labels):
        member = clone(clf)
        member.fit(samples[fit_idx], labels[fit_idx])
        scores.append(member.score(samples[eval_idx],
                                   labels[eval_idx]))
    return float(np.mean(scores))


class MVPAVoxelSelector:
    """Searchlight-driven activity-based voxel selection.

    Parameters: data [x, y, z, epoch] (from
    prepare_searchlight_mvpa_data), mask 3-D, labels per epoch,
    num_folds, and a Searchlight instance.
    """

    def __init__(self, data, mask, labels, num_folds, sl):
        self.data = data
        self.mask = np.asarray(mask).astype(bool)
        if not self.mask.any():
            raise ValueError('Zero processed voxels')
        self.labels = labels
        self.num_folds = num_folds
        self.sl = sl

    def run(self, clf):
        """Returns (result_volume, [(voxel_id, accuracy)] sorted desc)."""
        ctx = self.sl.comm
        if ctx.is_root:
            logger.info('activity-based selection: searchlight sweep '
                        'starting')
        self.sl.distribute([self.data], self.mask)
        self.sl.broadcast((self.labels, self.num_folds, clf))
        volume = self.sl.run_searchlight(_sl_accuracy)

        ranked = []
        if ctx.is_root:
            in_mask = volume[self.mask]
            ranked = [(vid, acc if acc is not None else 0)
                      for vid, acc in enumerate(in_mask)]
            ranked.sort(key=lambda pair: pair[1], reverse=True)
            logger.info('activity-based selection: sweep finished '
                        '(%d voxels scored)', len(ranked))
        return volume, ranked
