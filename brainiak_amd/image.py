"""Volume masking and condition specifications.

API parity with the reference image module (ref src/brainiak/image.py:37-187);
volumes are ``brainiak_amd.nifti.NiftiImage`` (or anything exposing
``get_fdata()``), not nibabel SpatialImage.
"""

import itertools
from typing import Iterable, Optional, Sequence, Type, TypeVar

import numpy as np

__all__ = [
    "ConditionSpec",
    "MaskedMultiSubjectData",
    "SingleConditionSpec",
    "mask_image",
    "mask_images",
    "multimask_images",
]

T = TypeVar("T", bound="MaskedMultiSubjectData")


class MaskedMultiSubjectData(np.ndarray):
    """Array with shape [n_TRs, n_voxels, n_subjects]."""

    @classmethod
    def from_masked_images(cls: Type[T], masked_images: Iterable[np.ndarray],
                           n_subjects: int) -> T:
        """Stack per-subject [n_voxels, n_TRs] masked images into a
        [n_TRs, n_voxels, n_subjects] array (each image is transposed)."""
        it = iter(masked_images)
        first = next(it)
        shape = first.T.shape
        result = np.empty((shape[0], shape[1], n_subjects))
        n_images = 0
        for n_images, image in enumerate(itertools.chain([first], it)):
            image = image.T
            if image.shape != shape:
                raise ValueError(
                    "Image {} has different shape from first image: "
                    "{} != {}".format(n_images, image.shape, shape))
            result[:, :, n_images] = image
        n_images += 1
        if n_images != n_subjects:
            raise ValueError("n_subjects != number of images: {} != {}"
                             .format(n_subjects, n_images))
        return result.view(cls)


class ConditionSpec(np.ndarray):
    """One-hot [n_conditions, n_epochs, n_TRs] condition representation."""


class SingleConditionSpec(ConditionSpec):
    """ConditionSpec where each epoch belongs to exactly one condition."""

    def extract_labels(self) -> np.ndarray:
        """Condition label of each epoch."""
        condition_idxs, epoch_idxs, _ = np.where(self)
        _, unique_epoch_idxs = np.unique(epoch_idxs, return_index=True)
        return condition_idxs[unique_epoch_idxs]


def mask_image(image, mask: np.ndarray,
               data_type: Optional[type] = None) -> np.ndarray:
    """Apply a boolean volume mask (optionally casting first).

    ``image`` may include time as the last dimension; the mask applies
    to the first three spatial dims.
    """
    image_data = image.get_fdata()
    if image_data.shape[:3] != mask.shape:
        raise ValueError("Image data and mask have different shapes.")
    if data_type is not None:
        image_data = image_data.astype(data_type)
    return image_data[mask]


def multimask_images(images, masks: Sequence[np.ndarray],
                     image_type: Optional[type] = None):
    """For each image, yield the list of maskings by each mask."""
    for image in images:
        yield [mask_image(image, mask, image_type) for mask in masks]


def mask_images(images, mask: np.ndarray,
                image_type: Optional[type] = None):
    """Yield each image masked by ``mask``."""
    for imgs in multimask_images(images, (mask,), image_type):
        yield imgs[0]
