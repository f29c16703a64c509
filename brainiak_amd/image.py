"""Volume masking and condition specifications.

API parity with the reference image module (ref src/brainiak/image.py:37-187);
volumes are ``brainiak_amd.nifti.NiftiImage`` (or anything exposing
``get_fdata()``), not nibabel SpatialImage.
"""

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
    def from_masked_images(cls: Type[T],
                           masked_images: Iterable[np.ndarray],
                           n_subjects: int) -> T:
        """Stack per-subject [n_voxels, n_TRs] masked images into one
        [n_TRs, n_voxels, n_subjects] array (each image transposed)."""
        collected = []
        for pos, img in enumerate(masked_images):
            tr_major = np.asarray(img).T
            if collected and tr_major.shape != collected[0].shape:
                raise ValueError(
                    "Image %d has different shape from first image: "
                    "%s != %s"
                    % (pos, tr_major.shape, collected[0].shape))
            collected.append(tr_major)
        if len(collected) != n_subjects:
            raise ValueError("n_subjects != number of images: %d != %d"
                             % (n_subjects, len(collected)))
        return np.stack(collected, axis=-1).view(cls)


class ConditionSpec(np.ndarray):
    """One-hot [n_conditions, n_epochs, n_TRs] condition representation."""


class SingleConditionSpec(ConditionSpec):
    """ConditionSpec where each epoch belongs to exactly one condition."""

    def extract_labels(self) -> np.ndarray:
        """Condition label of each epoch: the condition whose one-hot
        row is active anywhere in that epoch's TR range."""
        active = np.asarray(self).any(axis=2)       # [condition, epoch]
        return active.argmax(axis=0)


def mask_image(image, mask: np.ndarray,
               data_type: Optional[type] = None) -> np.ndarray:
    """Apply a boolean volume mask (optionally casting first).

    ``image`` may carry time as a trailing dimension; the mask covers
    the three spatial dims.
    """
    voxels = image.get_fdata()
    if voxels.shape[:3] != mask.shape:
        raise ValueError("Image data and mask have different shapes.")
    if data_type is not None:
        voxels = voxels.astype(data_type)
    return voxels[mask]


def mask_images(images, mask: np.ndarray,
                image_type: Optional[type] = None):
    """Yield each image masked by ``mask``."""
    return (mask_image(img, mask, image_type) for img in images)


def multimask_images(images, masks: Sequence[np.ndarray],
                     image_type: Optional[type] = None):
    """For each image, yield its masking under every mask in turn."""
    for img in images:
        yield [mask_image(img, m, image_type) for m in masks]
