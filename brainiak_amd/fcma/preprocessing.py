"""FCMA data preparation (API parity: ref src/brainiak/fcma/preprocessing.py).

Epoch separation + z-scoring, voxel-shuffled null data, and the
distributed data distribution.  Where the reference broadcasts each
epoch matrix one-by-one over mpi4py (preprocessing.py:211-223), here the
epochs ride a single ``broadcast_object`` through the DistContext (gloo
on CPU, RCCL device broadcast on GPU).

Round-2 redesign: the reference walks the one-hot epoch specs with a
triple Python loop per call site (ref preprocessing.py:41-153); here a
single vectorized span-table builder (`_epoch_spans` — argmax/sum over
the one-hot axis) feeds every public function, and the per-epoch
normalization is one fused numpy expression.  Epochs are taken as
contiguous TR runs (the reference's own `generate_epochs_info` already
assumes this when it emits ``(start, start+r)`` spans).
"""

import logging
import math
from enum import Enum

import numpy as np

from ..image import mask_images, multimask_images
from ..parallel import DistContext
from ..utils.timing import stage_timer

logger = logging.getLogger(__name__)

__all__ = [
    "RandomType",
    "generate_epochs_info",
    "prepare_fcma_data",
    "prepare_mvpa_data",
    "prepare_searchlight_mvpa_data",
]


class RandomType(Enum):
    """NORANDOM = leave data; REPRODUCIBLE = voxel-shuffle with fixed seed;
    UNREPRODUCIBLE = voxel-shuffle with fresh entropy."""
    NORANDOM = 0
    REPRODUCIBLE = 1
    UNREPRODUCIBLE = 2


def _epoch_spans(epoch_list):
    """Vectorized span table from one-hot epoch specs.

    Each subject's spec is ``[n_conditions, n_epochs, n_TRs]`` one-hot.
    Returns a list of ``(condition, subject, start_tr, end_tr)`` tuples
    in the reference's emission order (subject-major, then condition,
    then epoch), with empty epochs dropped.  Start/length come from one
    ``argmax``/``sum`` over the TR axis instead of per-epoch scans.
    """
    spans = []
    for sid, spec in enumerate(epoch_list):
        spec = np.asarray(spec)
        lengths = spec.sum(axis=2)          # [n_cond, n_epochs]
        starts = spec.argmax(axis=2)
        for cond, eid in np.argwhere(lengths > 0):
            a = int(starts[cond, eid])
            spans.append((int(cond), sid, a, a + int(lengths[cond, eid])))
    return spans


def _unit_epoch(segment):
    """``[L, V]`` epoch → per-voxel z-score (population std) scaled by
    1/sqrt(L), so a later matrix product of two epochs IS their Pearson
    correlation.  Constant voxels become all-zero columns (the
    reference's nan_to_num semantics)."""
    seg = np.asarray(segment, dtype=np.float64)
    mu = seg.mean(axis=0)
    sd = seg.std(axis=0)
    with np.errstate(divide="ignore", invalid="ignore"):
        z = (seg - mu) / sd
    z[~np.isfinite(z)] = 0.0
    return np.ascontiguousarray(
        (z / math.sqrt(seg.shape[0])).astype(np.float32))


def _separate_epochs(activity_data, epoch_list):
    """Cut per-subject ``[nVoxels, nTRs]`` data into per-epoch
    ``[epoch_len, nVoxels]`` unit-normalized matrices + labels."""
    spans = _epoch_spans(epoch_list)
    raw_data = [_unit_epoch(activity_data[sid][:, a:b].T)
                for _, sid, a, b in spans]
    labels = [cond for cond, _, _, _ in spans]
    return raw_data, labels


def _randomize_single_subject(data, seed=None):
    """Shuffle the voxel dimension of one [nVoxels, nTRs] subject
    in place."""
    if seed is not None:
        np.random.seed(seed)
    np.random.shuffle(data)


def _randomize_subject_list(data_list, random):
    if random == RandomType.REPRODUCIBLE:
        for i, d in enumerate(data_list):
            _randomize_single_subject(d, seed=i)
    elif random == RandomType.UNREPRODUCIBLE:
        for d in data_list:
            _randomize_single_subject(d)


def prepare_fcma_data(images, conditions, mask1, mask2=None,
                      random=RandomType.NORANDOM, comm=None):
    """Mask, epoch-separate and z-score the images on rank 0, then
    broadcast to every rank.

    Returns (raw_data1, raw_data2 or None, labels) exactly as the
    reference does.
    """
    ctx = comm if isinstance(comm, DistContext) else DistContext()
    results = ([], [], [])            # raw_data1, raw_data2, labels
    if ctx.is_root:
        timer = stage_timer("mask + epoch separation", logger)
        timer.__enter__()
        if mask2 is None:
            per_mask = [list(mask_images(images, mask1, np.float32))]
        else:
            stacked = [tuple(m) for m in
                       multimask_images(images, (mask1, mask2),
                                        np.float32)]
            per_mask = [[s[i] for s in stacked] for i in (0, 1)]
        # shuffle+separate mask2 FIRST, then mask1 — the reference's
        # RNG consumption order (its REPRODUCIBLE seeds reset per
        # subject, but UNREPRODUCIBLE draws stream through)
        epoched = {}
        for mi in reversed(range(len(per_mask))):
            _randomize_subject_list(per_mask[mi], random)
            epoched[mi] = _separate_epochs(per_mask[mi], conditions)
        results = (epoched[0][0],
                   epoched[1][0] if 1 in epoched else [],
                   epoched[0][1])
        timer.__exit__(None, None, None)

    if ctx.is_distributed:
        results = ctx.broadcast_object(results if ctx.is_root else None)
        logger.info('data broadcasting done')
    raw_data1, raw_data2, labels = results
    return raw_data1, (None if mask2 is None else raw_data2), labels


def generate_epochs_info(epoch_list):
    """Per-epoch (label, sid, start, end) tuples from one-hot epoch
    specs (the vectorized span table, see `_epoch_spans`)."""
    return _epoch_spans(epoch_list)


def _zscore_subject_groups(mat, subject_ids, axis):
    """Z-score (population std) the epoch axis within each subject's
    contiguous run of columns; subjects with a single epoch pass
    through untouched.  NaNs (constant rows) become zeros."""
    subject_ids = np.asarray(subject_ids)
    out = np.array(mat, copy=True)
    for sid in np.unique(subject_ids):
        sel = np.nonzero(subject_ids == sid)[0]
        if sel.size <= 1:
            continue
        block = np.take(out, sel, axis=axis)
        mu = block.mean(axis=axis, keepdims=True)
        sd = block.std(axis=axis, keepdims=True)
        with np.errstate(divide="ignore", invalid="ignore"):
            block = (block - mu) / sd
        ix = [slice(None)] * out.ndim
        ix[axis] = sel
        out[tuple(ix)] = block
    return np.nan_to_num(out)


def prepare_mvpa_data(images, conditions, mask):
    """Epoch-averaged, within-subject z-scored activity:
    returns ([num_voxels, num_epochs], labels)."""
    activity_data = list(mask_images(images, mask, np.float32))
    spans = _epoch_spans(conditions)
    labels = np.array([cond for cond, _, _, _ in spans], dtype=float)
    sids = [sid for _, sid, _, _ in spans]
    processed = np.column_stack(
        [activity_data[sid][:, a:b].mean(axis=1)
         for _, sid, a, b in spans])
    return _zscore_subject_groups(processed, sids, axis=1), labels


def prepare_searchlight_mvpa_data(images, conditions, data_type=np.float32,
                                  random=RandomType.NORANDOM):
    """Epoch-averaged, within-subject z-scored activity keeping the 3-D
    volume: returns ([x, y, z, num_epochs], labels).  Subjects are
    streamed one at a time so only one 4-D volume is resident."""
    spans = _epoch_spans(conditions)
    labels = np.array([cond for cond, _, _, _ in spans], dtype=float)
    sids = [sid for _, sid, _, _ in spans]
    processed = None

    for sid, f in enumerate(images):
        data = f.get_fdata().astype(data_type)
        if random != RandomType.NORANDOM:
            flat = data.reshape((-1, data.shape[3]))
            _randomize_single_subject(
                flat,
                seed=sid if random == RandomType.REPRODUCIBLE else None)
            data = flat.reshape(data.shape)
        if processed is None:
            processed = np.empty(data.shape[:3] + (len(spans),),
                                 dtype=data_type)
        for idx, (_, s, a, b) in enumerate(spans):
            if s == sid:
                processed[..., idx] = data[..., a:b].mean(axis=3)
    return _zscore_subject_groups(processed, sids, axis=3), labels
