"""Volume / label IO (API parity: ref src/brainiak/io.py:39-168).

Backed by the self-contained NIfTI-1 implementation in
``brainiak_amd.nifti`` instead of nibabel.
"""

from pathlib import Path
from typing import Callable, Iterable, List, Union

import numpy as np

from . import nifti
from .image import SingleConditionSpec

__all__ = [
    "load_boolean_mask",
    "load_images",
    "load_images_from_dir",
    "load_labels",
    "save_as_nifti_file",
]


def load_images_from_dir(in_dir: Union[str, Path], suffix: str = "nii.gz"):
    """Lazily load all images in ``in_dir`` with ``suffix``, sorted by name."""
    in_dir = Path(in_dir)
    files = sorted(in_dir.glob("*" + suffix))
    for f in files:
        yield nifti.load(str(f))


def load_images(image_paths: Iterable[Union[str, Path]]):
    """Lazily load images from an iterable of paths."""
    for path in image_paths:
        yield nifti.load(str(path))


def load_boolean_mask(path: Union[str, Path],
                      predicate: Callable[[np.ndarray], np.ndarray] = None
                      ) -> np.ndarray:
    """Load a boolean mask volume; default predicate is truthiness."""
    img = nifti.load(str(path))
    data = img.get_fdata()
    if predicate is not None:
        mask = predicate(data)
    else:
        mask = data.astype(bool)
    return mask


def load_labels(path: Union[str, Path]) -> List[SingleConditionSpec]:
    """Load condition labels from an .npy file of one-hot epoch arrays."""
    condition_specs = np.load(str(path))
    return [c.view(SingleConditionSpec) for c in condition_specs]


def save_as_nifti_file(data: np.ndarray, affine: np.ndarray,
                       path: Union[str, Path]) -> None:
    """Save an array + affine as a NIfTI-1 file."""
    image = nifti.NiftiImage(data, affine)
    nifti.save(image, str(path))
