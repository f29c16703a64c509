"""End-to-end MVPAVoxelSelector (ref tests/fcma/
test_mvpa_voxel_selection.py)."""

import numpy as np
from sklearn import svm

from brainiak_amd.fcma.mvpa_voxelselector import MVPAVoxelSelector
from brainiak_amd.searchlight import Searchlight


def test_mvpa_voxel_selection(seeded_rng):
    dims = (5, 5, 5)
    n_epochs = 12
    mask = np.zeros(dims, dtype=bool)
    mask[1:4, 1:4, 1:4] = True
    labels = np.array([e % 2 for e in range(n_epochs)])
    # epoch-mean activity [x, y, z, epochs]; an informative center
    data = seeded_rng.randn(*dims, n_epochs).astype(np.float32)
    data[2, 2, 2, :] += labels * 4.0      # voxel carries the condition
    data[2, 2, 1, :] += labels * 4.0
    sl = Searchlight(sl_rad=1, max_blk_edge=3)
    mvs = MVPAVoxelSelector(data, mask, labels, num_folds=4, sl=sl)
    result_volume, results = mvs.run(
        svm.SVC(kernel='rbf', C=10, gamma='auto'))
    assert len(results) == int(mask.sum())
    assert all(0.0 <= acc <= 1.0 for _, acc in results)
    # searchlights that include the informative voxels should dominate:
    # every top-5 center must lie within radius 2 of (2, 2, 2)
    coords = np.array(np.where(mask)).T
    best = results[0][1]
    assert best > 0.8
    for vid, acc in results[:5]:
        assert np.abs(coords[vid] - np.array([2, 2, 2])).max() <= 2
