"""Full Correlation Matrix Analysis (FCMA).

The headline workload of this toolkit: voxel-by-voxel whole-brain
correlation, Fisher-z within-subject normalization, per-voxel SVM kernel
matrices and cross-validated voxel selection — fused into a HIP/CDNA4
pipeline on MI355X (see brainiak_amd.ops) with RCCL voxel sharding in
place of the reference's MPI master-worker farm.

Method citations (same algorithms as the reference implements):
[Wang2015-1] "Full correlation matrix analysis (FCMA): An unbiased method
for task-related functional connectivity", J. Neurosci. Methods 2015.
[Wang2015-2] "Full correlation matrix analysis of fMRI data on Intel Xeon
Phi coprocessors", SC'15.
"""

from .classifier import Classifier  # noqa: F401
from .preprocessing import (  # noqa: F401
    RandomType,
    generate_epochs_info,
    prepare_fcma_data,
    prepare_mvpa_data,
    prepare_searchlight_mvpa_data,
)
from .util import compute_correlation  # noqa: F401
from .voxelselector import VoxelSelector  # noqa: F401

__all__ = [
    "Classifier",
    "RandomType",
    "VoxelSelector",
    "compute_correlation",
    "generate_epochs_info",
    "prepare_fcma_data",
    "prepare_mvpa_data",
    "prepare_searchlight_mvpa_data",
]
