"""FCMA data preparation (API parity: ref src/brainiak/fcma/preprocessing.py).

Epoch separation + z-scoring, voxel-shuffled null data, and the
distributed data distribution.  Where the reference broadcasts each
epoch matrix one-by-one over mpi4py (preprocessing.py:211-223), here the
epochs ride a single ``broadcast_object`` through the DistContext (gloo
on CPU, RCCL device broadcast on GPU).
"""

import logging
import math
from enum import Enum

import numpy as np
from scipy.stats import zscore

from ..image import mask_images, multimask_images
from ..parallel import DistContext

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


def _separate_epochs(activity_data, epoch_list):
    """Cut per-subject [nVoxels, nTRs] data into per-epoch
    [epoch_len, nVoxels] matrices, z-scored per voxel (ddof=0) and scaled
    by 1/sqrt(epoch_len) so correlation is a plain matrix product."""
    raw_data = []
    labels = []
    for sid in range(len(epoch_list)):
        epoch = epoch_list[sid]
        for cond in range(epoch.shape[0]):
            sub_epoch = epoch[cond, :, :]
            for eid in range(epoch.shape[1]):
                r = np.sum(sub_epoch[eid, :])
                if r > 0:
                    mat = activity_data[sid][:, sub_epoch[eid, :] == 1]
                    mat = np.ascontiguousarray(mat.T)
                    mat = zscore(mat, axis=0, ddof=0)
                    mat = np.nan_to_num(mat)
                    mat = mat / math.sqrt(r)
                    raw_data.append(mat.astype(np.float32))
                    labels.append(cond)
    return raw_data, labels


def _randomize_single_subject(data, seed=None):
    """Shuffle the voxel dimension of one [nVoxels, nTRs] subject
    in place."""
    if seed is not None:
        np.random.seed(seed)
    np.random.shuffle(data)


def _randomize_subject_list(data_list, random):
    if random == RandomType.REPRODUCIBLE:
        for i in range(len(data_list)):
            _randomize_single_subject(data_list[i], seed=i)
    elif random == RandomType.UNREPRODUCIBLE:
        for data in data_list:
            _randomize_single_subject(data)


def prepare_fcma_data(images, conditions, mask1, mask2=None,
                      random=RandomType.NORANDOM, comm=None):
    """Mask, epoch-separate and z-score the images on rank 0, then
    broadcast to every rank.

    Returns (raw_data1, raw_data2 or None, labels) exactly as the
    reference does.
    """
    ctx = comm if isinstance(comm, DistContext) else DistContext()
    labels = []
    raw_data1 = []
    raw_data2 = []
    if ctx.is_root:
        logger.info('start to apply masks and separate epochs')
        if mask2 is not None:
            masks = (mask1, mask2)
            pairs = [tuple(m) for m in multimask_images(images, masks,
                                                        np.float32)]
            activity_data1 = [p[0] for p in pairs]
            activity_data2 = [p[1] for p in pairs]
            _randomize_subject_list(activity_data2, random)
            raw_data2, _ = _separate_epochs(activity_data2, conditions)
            _randomize_subject_list(activity_data1, random)
            raw_data1, labels = _separate_epochs(activity_data1, conditions)
        else:
            activity_data1 = list(mask_images(images, mask1, np.float32))
            _randomize_subject_list(activity_data1, random)
            raw_data1, labels = _separate_epochs(activity_data1, conditions)

    if ctx.is_distributed:
        payload = ctx.broadcast_object(
            (raw_data1, raw_data2, labels) if ctx.is_root else None)
        raw_data1, raw_data2, labels = payload
        logger.info('data broadcasting done')
    if mask2 is None:
        raw_data2 = None
    return raw_data1, raw_data2, labels


def generate_epochs_info(epoch_list):
    """Per-epoch (label, sid, start, end) tuples from one-hot epoch specs."""
    epoch_info = []
    for sid, epoch in enumerate(epoch_list):
        for cond in range(epoch.shape[0]):
            sub_epoch = epoch[cond, :, :]
            for eid in range(epoch.shape[1]):
                r = np.sum(sub_epoch[eid, :])
                if r > 0:
                    start = np.nonzero(sub_epoch[eid, :])[0][0]
                    epoch_info.append((cond, sid, start, start + r))
    return epoch_info


def prepare_mvpa_data(images, conditions, mask):
    """Epoch-averaged, within-subject z-scored activity:
    returns ([num_voxels, num_epochs], labels)."""
    activity_data = list(mask_images(images, mask, np.float32))
    epoch_info = generate_epochs_info(conditions)
    num_epochs = len(epoch_info)
    d1 = activity_data[0].shape[0]
    processed_data = np.empty([d1, num_epochs])
    labels = np.empty(num_epochs)
    subject_count = [0]
    cur_sid = -1
    for idx, epoch in enumerate(epoch_info):
        labels[idx] = epoch[0]
        if cur_sid != epoch[1]:
            subject_count.append(0)
            cur_sid = epoch[1]
        subject_count[-1] += 1
        processed_data[:, idx] = np.mean(
            activity_data[cur_sid][:, epoch[2]:epoch[3]], axis=1)
    cur_epoch = 0
    for i in subject_count:
        if i > 1:
            processed_data[:, cur_epoch:cur_epoch + i] = zscore(
                processed_data[:, cur_epoch:cur_epoch + i], axis=1, ddof=0)
        cur_epoch += i
    processed_data = np.nan_to_num(processed_data)
    return processed_data, labels


def prepare_searchlight_mvpa_data(images, conditions, data_type=np.float32,
                                  random=RandomType.NORANDOM):
    """Epoch-averaged, within-subject z-scored activity keeping the 3-D
    volume: returns ([x, y, z, num_epochs], labels).  Subjects are
    streamed one at a time."""
    epoch_info = generate_epochs_info(conditions)
    num_epochs = len(epoch_info)
    processed_data = None
    labels = np.empty(num_epochs)
    for idx, epoch in enumerate(epoch_info):
        labels[idx] = epoch[0]
    subject_count = np.zeros(len(conditions), dtype=np.int32)

    for sid, f in enumerate(images):
        data = f.get_fdata().astype(data_type)
        d1, d2, d3, d4 = data.shape
        if random != RandomType.NORANDOM:
            flat = data.reshape((d1 * d2 * d3, d4))
            _randomize_single_subject(
                flat, seed=sid if random == RandomType.REPRODUCIBLE else None)
            data = flat.reshape((d1, d2, d3, d4))
        if processed_data is None:
            processed_data = np.empty([d1, d2, d3, num_epochs],
                                      dtype=data_type)
        for idx, epoch in enumerate(epoch_info):
            if sid == epoch[1]:
                subject_count[sid] += 1
                processed_data[:, :, :, idx] = np.mean(
                    data[:, :, :, epoch[2]:epoch[3]], axis=3)
    cur_epoch = 0
    for i in subject_count:
        if i > 1:
            processed_data[:, :, :, cur_epoch:cur_epoch + i] = zscore(
                processed_data[:, :, :, cur_epoch:cur_epoch + i],
                axis=3, ddof=0)
        cur_epoch += i
    processed_data = np.nan_to_num(processed_data)
    return processed_data, labels
