"""Activity-based voxel selection via Searchlight (API parity:
ref src/brainiak/fcma/mvpa_voxelselector.py:34-136)."""

import logging

import numpy as np
from sklearn import model_selection

logger = logging.getLogger(__name__)

__all__ = ["MVPAVoxelSelector"]


def _sfn(data, mask, myrad, bcast_var):
    """Cross-validated accuracy of the activity vectors inside one
    searchlight; bcast_var = (labels, num_folds, clf)."""
    clf = bcast_var[2]
    masked_data = data[0][mask, :].T
    skf = model_selection.StratifiedKFold(n_splits=bcast_var[1],
                                          shuffle=False)
    return np.mean(model_selection.cross_val_score(
        clf, masked_data, y=bcast_var[0], cv=skf, n_jobs=1))


class MVPAVoxelSelector:
    """Searchlight-driven activity-based voxel selection.

    Parameters: data [x, y, z, epoch] (from
    prepare_searchlight_mvpa_data), mask 3-D, labels per epoch,
    num_folds, and a Searchlight instance.
    """

    def __init__(self, data, mask, labels, num_folds, sl):
        self.data = data
        self.mask = mask.astype(bool)
        self.labels = labels
        self.num_folds = num_folds
        self.sl = sl
        if np.sum(self.mask) == 0:
            raise ValueError('Zero processed voxels')

    def run(self, clf):
        """Returns (result_volume, [(voxel_id, accuracy)] sorted desc)."""
        ctx = self.sl.comm
        if ctx.is_root:
            logger.info(
                'running activity-based voxel selection via Searchlight')
        self.sl.distribute([self.data], self.mask)
        self.sl.broadcast((self.labels, self.num_folds, clf))

        result_volume = self.sl.run_searchlight(_sfn)
        result_list = result_volume[self.mask]
        results = []
        if ctx.is_root:
            for idx, value in enumerate(result_list):
                if value is None:
                    value = 0
                results.append((idx, value))
            results.sort(key=lambda tup: tup[1], reverse=True)
            logger.info(
                'activity-based voxel selection via Searchlight is done')
        return result_volume, results
