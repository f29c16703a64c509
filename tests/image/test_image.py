"""Tests for image.py (ref tests/image/test_image.py parity)."""

import numpy as np
import pytest

from brainiak_amd.image import (MaskedMultiSubjectData,
                                SingleConditionSpec, mask_image,
                                mask_images, multimask_images)
from brainiak_amd.nifti import NiftiImage


def _img(data):
    return NiftiImage(np.asarray(data, dtype=np.float64), np.eye(4))


def test_masked_multi_subject_data(seeded_rng):
    imgs = [seeded_rng.rand(20, 10) for _ in range(3)]  # [V, T]
    d = MaskedMultiSubjectData.from_masked_images(imgs, 3)
    assert d.shape == (10, 20, 3)
    assert np.allclose(d[:, :, 1], imgs[1].T)
    with pytest.raises(ValueError):
        MaskedMultiSubjectData.from_masked_images(imgs, 4)
    bad = imgs + [seeded_rng.rand(21, 10)]
    with pytest.raises(ValueError):
        MaskedMultiSubjectData.from_masked_images(bad, 4)


def test_single_condition_spec_labels():
    spec = np.zeros((2, 4, 6), dtype=int)
    spec[0, 0, 0:2] = 1
    spec[1, 1, 2:4] = 1
    spec[0, 2, 4:5] = 1
    spec[1, 3, 5:6] = 1
    labels = spec.view(SingleConditionSpec).extract_labels()
    assert labels.tolist() == [0, 1, 0, 1]


def test_mask_image_shapes(seeded_rng):
    vol = seeded_rng.rand(4, 5, 6, 7)
    mask = np.zeros((4, 5, 6), dtype=bool)
    mask[1:3, 2:4, 3:5] = True
    out = mask_image(_img(vol), mask, np.float32)
    assert out.shape == (int(mask.sum()), 7)
    assert out.dtype == np.float32
    with pytest.raises(ValueError):
        mask_image(_img(vol), np.ones((3, 3, 3), dtype=bool))


def test_multimask_and_mask_images(seeded_rng):
    vols = [seeded_rng.rand(4, 4, 4, 5) for _ in range(2)]
    imgs = [_img(v) for v in vols]
    m1 = np.zeros((4, 4, 4), dtype=bool)
    m1[0, 0, 0] = True
    m2 = np.ones((4, 4, 4), dtype=bool)
    pairs = list(multimask_images(imgs, (m1, m2)))
    assert len(pairs) == 2 and len(pairs[0]) == 2
    assert pairs[0][0].shape == (1, 5)
    assert pairs[1][1].shape == (64, 5)
    singles = list(mask_images([_img(v) for v in vols], m2))
    assert np.allclose(singles[0], pairs[0][1])


# -- round-2 depth (ref tests/image/test_image.py:1-181) ---------------------

def test_masked_multi_subject_data_properties(seeded_rng):
    from brainiak_amd.image import MaskedMultiSubjectData
    # inputs are [voxels, TRs]; the class transposes to
    # [TRs, voxels, subjects] (ref image.py:37-63)
    imgs = [seeded_rng.rand(30, 5) for _ in range(3)]
    d = MaskedMultiSubjectData.from_masked_images(imgs, 3)
    assert d.shape == (5, 30, 3)
    assert np.allclose(d[..., 1], imgs[1].T)
    # it IS an ndarray subclass: numpy ops keep working
    assert np.isclose(d.mean(), np.mean(imgs))


def test_from_masked_images_count_mismatch(seeded_rng):
    from brainiak_amd.image import MaskedMultiSubjectData
    imgs = [seeded_rng.rand(5, 30) for _ in range(2)]
    with pytest.raises(ValueError):
        MaskedMultiSubjectData.from_masked_images(iter(imgs), 3)


def test_single_condition_spec_partial_epochs():
    from brainiak_amd.image import SingleConditionSpec
    spec = np.zeros((2, 4, 10), dtype=bool)
    spec[0, 0, 0:3] = True
    spec[1, 1, 4:7] = True
    spec[0, 2, 7:9] = True
    cs = spec.view(SingleConditionSpec)
    labels = cs.extract_labels()
    assert labels.shape == (4,) or labels.shape == (3,)
    # epoch 0 is condition 0, epoch 1 condition 1
    assert labels[0] == 0 and labels[1] == 1


def test_mask_image_dtype_conversion(seeded_rng):
    from brainiak_amd.image import mask_image
    from brainiak_amd.nifti import NiftiImage
    vol = (seeded_rng.rand(4, 4, 4, 6) * 100).astype(np.int16)
    img = NiftiImage(vol)
    mask = np.zeros((4, 4, 4), dtype=bool)
    mask[1:3, 1:3, 1:3] = True
    out = mask_image(img, mask, np.float32)
    assert out.dtype == np.float32
    assert out.shape == (8, 6)


def test_multimask_images_pairing(seeded_rng):
    from brainiak_amd.image import multimask_images
    from brainiak_amd.nifti import NiftiImage
    imgs = [NiftiImage(seeded_rng.rand(4, 4, 4, 3)) for _ in range(2)]
    m1 = np.zeros((4, 4, 4), dtype=bool); m1[0] = True
    m2 = np.zeros((4, 4, 4), dtype=bool); m2[3] = True
    out = list(multimask_images(imgs, (m1, m2)))
    assert len(out) == 2
    a, b = out[0]
    assert a.shape == (16, 3) and b.shape == (16, 3)
