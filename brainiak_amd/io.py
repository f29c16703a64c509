"""Volume / label IO (API parity: ref src/brainiak/io.py:39-168).

Backed by the self-contained NIfTI reader/writer in
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

PathLike = Union[str, Path]


def _read_volume(source: PathLike):
    return nifti.load(str(source))


def load_images_from_dir(in_dir: PathLike, suffix: str = "nii.gz"):
    """Lazily load every ``*suffix`` volume in ``in_dir`` (name order)."""
    matches = sorted(p for p in Path(in_dir).iterdir()
                     if p.name.endswith(suffix))
    return load_images(matches)


def load_images(image_paths: Iterable[PathLike]):
    """Lazily load volumes from an iterable of paths."""
    return map(_read_volume, image_paths)


def load_boolean_mask(path: PathLike,
                      predicate: Callable[[np.ndarray], np.ndarray] = None
                      ) -> np.ndarray:
    """Load a volume as a boolean mask; ``predicate`` (default:
    truthiness) maps the voxel data to booleans."""
    voxels = _read_volume(path).get_fdata()
    keep = predicate(voxels) if predicate is not None else voxels != 0
    return np.asarray(keep, dtype=bool)


def load_labels(path: PathLike) -> List[SingleConditionSpec]:
    """Condition labels: an .npy stack of one-hot epoch arrays."""
    stack = np.load(str(path))
    return [spec.view(SingleConditionSpec) for spec in stack]


def save_as_nifti_file(data: np.ndarray, affine: np.ndarray,
                       path: PathLike) -> None:
    """Write ``data`` with ``affine`` as a NIfTI-1 volume."""
    nifti.save(nifti.NiftiImage(data, affine), str(path))
