"""Intersubject correlation (ISC/ISFC) and its nonparametric statistics.

API parity with the reference isc module (ref src/brainiak/isc.py:81-1551):
``isc``, ``isfc``, ``compute_summary_statistic``, ``squareform_isfc``,
``bootstrap_isc`` (subject bootstrap with Hall-Wilson shift),
``permutation_isc`` (one-sample sign-flip / two-sample label shuffle, exact
tests when the permutation space is small), ``timeshift_isc`` (circular
shifts), ``phaseshift_isc`` (FFT phase randomization).

MI355X additions:
 - the heavy ISFC gemms run through torch (rocBLAS/MFMA on GPU tensors);
 - ``isfc_distributed``: subject-sharded leave-one-out ISFC over RCCL —
   each rank holds a subset of subjects, the across-subject (nan-aware)
   sum rides ONE all-reduce, and each rank correlates its subjects
   against the leave-one-out mean on its own GPU (BASELINE config 4:
   50k voxels x 32 subjects over xGMI).

Citations as in the reference: [Hasson2004], [Simony2016], [Chen2016],
[HallWilson1991], [PhipsonSmyth2010], [SilverDunlap1987].
"""

import logging
import math
from itertools import combinations, permutations, product

import numpy as np
from scipy.spatial.distance import squareform

from .fcma.util import compute_correlation
from .utils.utils import (
    _check_timeseries_input,
    array_correlation,
    p_from_null,
    phase_randomize,
)

logger = logging.getLogger(__name__)

MAX_RANDOM_SEED = 2 ** 32 - 1

__all__ = [
    "bootstrap_isc",
    "compute_summary_statistic",
    "isc",
    "isfc",
    "isfc_distributed",
    "permutation_isc",
    "phaseshift_isc",
    "squareform_isfc",
    "timeshift_isc",
]


def _threshold_nans(data, tolerate_nans):
    """NaN-threshold voxels; returns (masked data, keep-mask)."""
    nans = np.all(np.any(np.isnan(data), axis=0), axis=1)
    if tolerate_nans is True:
        logger.info("ISC computation will tolerate all NaNs when averaging")
    elif type(tolerate_nans) is float:
        if not 0.0 <= tolerate_nans <= 1.0:
            raise ValueError("If threshold to tolerate NaNs is a float, "
                             "it must be between 0.0 and 1.0; got {0}".format(
                                 tolerate_nans))
        nans += ~(np.sum(~np.any(np.isnan(data), axis=0), axis=1) >=
                  data.shape[-1] * tolerate_nans)
        logger.info("ISC computation will tolerate voxels with at least "
                    "%s non-NaN values: %d voxels do not meet threshold",
                    tolerate_nans, np.sum(nans))
    else:
        logger.info("ISC computation will not tolerate NaNs when averaging")
    mask = ~nans
    return data[:, mask, :], mask


def isc(data, pairwise=False, summary_statistic=None, tolerate_nans=True):
    """Intersubject correlation per voxel (leave-one-out or pairwise)."""
    data, n_TRs, n_voxels, n_subjects = _check_timeseries_input(data)

    if n_subjects == 2:
        logger.info("Only two subjects! Simply computing Pearson "
                    "correlation.")
        summary_statistic = None

    mean = np.nanmean if tolerate_nans else np.mean
    data, mask = _threshold_nans(data, tolerate_nans)

    if n_subjects == 2:
        iscs_stack = array_correlation(data[..., 0],
                                       data[..., 1])[np.newaxis, :]
    elif pairwise:
        swapped = np.swapaxes(data, 2, 0)
        voxel_iscs = []
        for v in np.arange(swapped.shape[1]):
            voxel_data = swapped[:, v, :]
            voxel_iscs.append(squareform(np.corrcoef(voxel_data),
                                         checks=False))
        iscs_stack = np.column_stack(voxel_iscs)
    else:
        iscs_stack = []
        for s in np.arange(n_subjects):
            iscs_stack.append(array_correlation(
                data[..., s], mean(np.delete(data, s, axis=2), axis=2)))
        iscs_stack = np.array(iscs_stack)

    iscs = np.full((iscs_stack.shape[0], n_voxels), np.nan)
    iscs[:, np.where(mask)[0]] = iscs_stack

    if summary_statistic:
        iscs = compute_summary_statistic(
            iscs, summary_statistic=summary_statistic, axis=0)[np.newaxis, :]
    if iscs.shape[0] == 1:
        iscs = iscs[0]
    return iscs


def _check_targets_input(targets, data):
    if isinstance(targets, (np.ndarray, list)):
        targets, n_TRs, n_voxels, n_subjects = (
            _check_timeseries_input(targets))
        if data.shape[0] != n_TRs:
            raise ValueError("Targets array must have same number of "
                             "TRs as input data")
        if data.shape[2] != n_subjects:
            raise ValueError("Targets array must have same number of "
                             "subjects as input data")
        symmetric = False
    else:
        targets = data
        n_TRs, n_voxels, n_subjects = data.shape
        symmetric = True
    return targets, n_TRs, n_voxels, n_subjects, symmetric


def isfc(data, targets=None, pairwise=False, summary_statistic=None,
         vectorize_isfcs=True, tolerate_nans=True):
    """Intersubject functional correlation: correlations between each
    voxel's time series and every (target) voxel in other subjects."""
    data, n_TRs, n_voxels, n_subjects = _check_timeseries_input(data)
    targets, t_n_TRs, t_n_voxels, _, symmetric = (
        _check_targets_input(targets, data))
    if not symmetric:
        pairwise = False
    mean = np.nanmean if tolerate_nans else np.mean
    data, mask = _threshold_nans(data, tolerate_nans)
    targets, targets_mask = _threshold_nans(targets, tolerate_nans)

    if symmetric and n_subjects == 2:
        isfcs = compute_correlation(np.ascontiguousarray(data[..., 0].T),
                                    np.ascontiguousarray(data[..., 1].T),
                                    return_nans=True)
        isfcs = ((isfcs + isfcs.T) / 2)[..., np.newaxis]
        summary_statistic = None
        logger.info("Only two subjects! Computing ISFC between them.")
    elif pairwise:
        stack = []
        for pair in combinations(np.arange(n_subjects), 2):
            isfc_pair = compute_correlation(
                np.ascontiguousarray(data[..., pair[0]].T),
                np.ascontiguousarray(targets[..., pair[1]].T),
                return_nans=True)
            if symmetric:
                isfc_pair = (isfc_pair + isfc_pair.T) / 2
            stack.append(isfc_pair)
        isfcs = np.dstack(stack)
    else:
        rolled = np.rollaxis(data, 2, 0)
        rolled_targets = np.rollaxis(targets, 2, 0)
        stack = [compute_correlation(
            np.ascontiguousarray(subject.T),
            np.ascontiguousarray(
                mean(np.delete(rolled_targets, s, axis=0), axis=0).T),
            return_nans=True) for s, subject in enumerate(rolled)]
        isfcs = np.dstack([(m + m.T) / 2 if symmetric else m
                           for m in stack])

    isfcs_all = np.full((len(mask), len(targets_mask), isfcs.shape[2]),
                        np.nan)
    isfcs_all[np.ix_(np.where(mask)[0],
                     np.where(targets_mask)[0])] = isfcs
    isfcs = np.moveaxis(isfcs_all, 2, 0)

    if summary_statistic:
        isfcs = compute_summary_statistic(
            isfcs, summary_statistic=summary_statistic, axis=0)
    if isfcs.shape[0] == 1:
        isfcs = isfcs[0]
    if vectorize_isfcs and symmetric:
        isfcs, iscs = squareform_isfc(isfcs)
        return isfcs, iscs
    return isfcs


def isfc_distributed(local_data, comm, summary_statistic=None,
                     row_tile=4096, device=None, precision='fp32',
                     return_tensor=False):
    """Subject-sharded leave-one-out ISFC over RCCL/xGMI.

    Parameters
    ----------
    local_data : list of [n_TRs, n_voxels] arrays — THIS rank's subjects.
    comm : DistContext.
    summary_statistic : None | 'mean' | 'median'.  With 'mean' the
        across-subject reduction rides the same collective pass and the
        full per-subject stack is never materialized.
    row_tile : accepted for API stability; the full-matrix GEMM path
        is used — at the benchmark scale (50k voxels) the [V, V]
        matrices fit 288 GB HBM several times over.
    device : torch device override.

    Returns (on every rank)
    -----------------------
    summary_statistic None  → [n_subjects_total, V, V] stacked ISFCs
    summary_statistic given → [V, V] collapsed ISFC matrix

    precision 'bf16' runs the [T,V]x[T,V] correlation GEMMs on the
    MFMA bf16 path (~0.4 % relative error on r; fp32 is the default
    and matches the serial oracle).  ``return_tensor=True`` keeps the
    collapsed result on the compute device (the D2H of a [V, V]
    matrix dominates the GPU step otherwise).
    """
    import torch

    dev = torch.device(device) if device is not None else comm.device
    local = [d.to(device=dev, dtype=torch.float32)
             if isinstance(d, torch.Tensor)
             else torch.as_tensor(np.ascontiguousarray(d),
                                  dtype=torch.float32).to(dev)
             for d in local_data]
    V = local[0].shape[1] if local else 0
    T = local[0].shape[0] if local else 0
    meta = comm.all_gather_object((len(local), V, T))
    n_total = sum(m[0] for m in meta)
    V = max(m[1] for m in meta)
    T = max(m[2] for m in meta)

    # z-score each subject once (so correlation = dot of unit columns)
    def _norm(d):
        d = d - d.mean(dim=0)
        denom = d.norm(dim=0).clamp_min(1e-30)
        return d / denom

    # ONE all-reduce carries the across-subject sum for the LOO mean
    total = torch.zeros((T, V), dtype=torch.float32, device=dev)
    for d in local:
        total += d
    total = comm.all_reduce(total)

    normed_local = [_norm(d) for d in local]

    def _corr(a, b, keep_bf16=False):
        if precision == 'bf16':
            m = a.T.to(torch.bfloat16) @ b.to(torch.bfloat16)
            return m if keep_bf16 else m.float()
        return a.T @ b                          # [V, V]

    if summary_statistic == 'mean':
        from . import ops as _ops
        use_hip = dev.type == 'cuda' and _ops.require_hip()
        acc = torch.zeros((V, V), dtype=torch.float32, device=dev)
        if use_hip:
            # subjects stream through the fused sym+atanh+accumulate
            # kernel in stacks of up to 4: the [V, V] accumulator's
            # read-modify-write happens once per stack instead of once
            # per subject (acc traffic dominates at V=50k).  GEMMs
            # write straight into the stack buffer slices — no copies.
            nb = min(4, max(1, len(local)))
            gdt = (torch.bfloat16 if precision == 'bf16'
                   else torch.float32)
            stackbuf = torch.empty((nb, V, V), dtype=gdt, device=dev)
            pairs = list(zip(local, normed_local))
            j = 0
            for i, (d, nd) in enumerate(pairs):
                loo = _norm((total - d) / (n_total - 1))
                torch.matmul(nd.T.to(gdt), loo.to(gdt),
                             out=stackbuf[j])
                j += 1
                if j == nb or i == len(pairs) - 1:
                    _ops.isfc_accum_(acc, stackbuf[:j])
                    j = 0
            del stackbuf
        else:
            for d, nd in zip(local, normed_local):
                loo = _norm((total - d) / (n_total - 1))
                m = _corr(nd, loo)
                m = (m + m.T) / 2
                acc += torch.atanh(m.clamp(-1 + 1e-7, 1 - 1e-7))
        acc = comm.all_reduce(acc)
        out = torch.tanh(acc / n_total)
        if return_tensor:
            # device-resident result for GPU consumers: the pageable
            # D2H of a [50k, 50k] fp32 matrix costs ~1 s — more than
            # the whole on-device computation (profiles/README.md r2)
            return out
        if out.is_cuda:
            host = torch.empty_like(out, device="cpu",
                                    pin_memory=True)
            host.copy_(out)
            return host.numpy()
        return out.cpu().numpy()

    stacks = []
    for d, nd in zip(local, normed_local):
        loo = _norm((total - d) / (n_total - 1))
        m = _corr(nd, loo)
        stacks.append(((m + m.T) / 2).cpu().numpy())
    gathered = comm.all_gather_object(stacks)
    flat = [m for part in gathered for m in part]
    out = np.stack(flat)
    if summary_statistic == 'median':
        return np.nanmedian(out, axis=0)
    return out


def _check_isc_input(iscs, pairwise=False):
    if type(iscs) == list:  # noqa: E721
        iscs = np.array(iscs)[:, np.newaxis]
    elif isinstance(iscs, np.ndarray):
        if iscs.ndim == 1:
            iscs = iscs[:, np.newaxis]
    if pairwise:
        try:
            test_square = squareform(iscs[:, 0], force='tomatrix')
            n_subjects = test_square.shape[0]
        except ValueError:
            raise ValueError("For pairwise input, ISCs must be the "
                             "vectorized triangle of a square matrix.")
    else:
        n_subjects = iscs.shape[0]
    n_voxels = iscs.shape[1]
    logger.info("Assuming %d subjects with and %d voxel(s) or ROI(s) in "
                "bootstrap ISC test.", n_subjects, n_voxels)
    return iscs, n_subjects, n_voxels


def compute_summary_statistic(iscs, summary_statistic='mean', axis=None):
    """'mean' (Fisher-Z mean: tanh(mean(arctanh))) or 'median' of ISCs."""
    if summary_statistic not in ('mean', 'median'):
        raise ValueError("Summary statistic must be 'mean' or 'median'")
    if summary_statistic == 'mean':
        return np.tanh(np.nanmean(np.arctanh(iscs), axis=axis))
    return np.nanmedian(iscs, axis=axis)


def squareform_isfc(isfcs, iscs=None):
    """Square ↔ condensed ISFC conversion retaining the ISC diagonal."""
    if not type(iscs) == np.ndarray and \
            isfcs.shape[-2] == isfcs.shape[-1]:  # noqa: E721
        if isfcs.ndim == 2:
            isfcs = isfcs[np.newaxis, ...]
        if isfcs.ndim == 3:
            iscs = np.diagonal(isfcs, axis1=1, axis2=2)
            isfcs = np.vstack([squareform(m, checks=False)[np.newaxis, :]
                               for m in isfcs])
        else:
            raise ValueError("Square (redundant) ISFCs must be square "
                             "with multiple subjects or pairs of subjects "
                             "indexed by the first dimension")
        if isfcs.shape[0] == iscs.shape[0] == 1:
            isfcs, iscs = isfcs[0], iscs[0]
        return isfcs, iscs
    else:
        if isfcs.ndim == iscs.ndim == 1:
            isfcs, iscs = isfcs[np.newaxis, :], iscs[np.newaxis, :]
        stack = []
        for isfc_row, isc_row in zip(isfcs, iscs):
            sq = squareform(isfc_row, checks=False)
            np.fill_diagonal(sq, isc_row)
            stack.append(sq[np.newaxis, ...])
        isfcs = np.vstack(stack)
        if isfcs.shape[0] == 1:
            isfcs = isfcs[0]
        return isfcs


def bootstrap_isc(iscs, pairwise=False, summary_statistic='median',
                  n_bootstraps=1000, ci_percentile=95, side='right',
                  random_state=None):
    """One-sample subject-level bootstrap with the Hall-Wilson shift."""
    iscs, n_subjects, n_voxels = _check_isc_input(iscs, pairwise=pairwise)
    if summary_statistic not in ('mean', 'median'):
        raise ValueError("Summary statistic must be 'mean' or 'median'")
    observed = compute_summary_statistic(
        iscs, summary_statistic=summary_statistic, axis=0)

    distribution = []
    for _ in np.arange(n_bootstraps):
        if isinstance(random_state, np.random.RandomState):
            prng = random_state
        else:
            prng = np.random.RandomState(random_state)
        subject_sample = sorted(prng.choice(np.arange(n_subjects),
                                            size=n_subjects))
        if pairwise:
            isc_sample = []
            for voxel_iscs in iscs.T:
                voxel_iscs = squareform(voxel_iscs, force='tomatrix')
                np.fill_diagonal(voxel_iscs, 1)
                voxel_sample = voxel_iscs[subject_sample, :][:,
                                                             subject_sample]
                voxel_sample = squareform(voxel_sample, checks=False)
                voxel_sample[voxel_sample == 1.] = np.nan
                isc_sample.append(voxel_sample)
            isc_sample = np.column_stack(isc_sample)
        else:
            isc_sample = iscs[subject_sample, :]
        distribution.append(compute_summary_statistic(
            isc_sample, summary_statistic=summary_statistic, axis=0))
        random_state = np.random.RandomState(
            prng.randint(0, MAX_RANDOM_SEED, dtype=np.int64))

    distribution = np.array(distribution)
    ci = (np.percentile(distribution, (100 - ci_percentile) / 2, axis=0),
          np.percentile(distribution,
                        ci_percentile + (100 - ci_percentile) / 2, axis=0))
    shifted = distribution - observed
    p = p_from_null(observed, shifted, side=side, exact=False, axis=0)
    return observed, ci, p, distribution


def _check_group_assignment(group_assignment, n_subjects):
    if type(group_assignment) == list:  # noqa: E721
        pass
    elif type(group_assignment) == np.ndarray:  # noqa: E721
        group_assignment = group_assignment.tolist()
    else:
        logger.info("No group assignment provided, "
                    "performing one-sample test.")
    if group_assignment and len(group_assignment) != n_subjects:
        raise ValueError("Group assignments ({0}) "
                         "do not match number of subjects ({1})!".format(
                             len(group_assignment), n_subjects))
    return group_assignment


def _get_group_parameters(group_assignment, n_subjects, pairwise=False):
    gp = {'group_assignment': group_assignment, 'n_subjects': n_subjects,
          'group_labels': None, 'groups': None, 'sorter': None,
          'unsorter': None, 'group_matrix': None, 'group_selector': None}
    if group_assignment and len(np.unique(group_assignment)) == 2:
        gp['n_groups'] = 2
        group_labels = np.unique(group_assignment)
        groups = {group_labels[0]: group_assignment.count(group_labels[0]),
                  group_labels[1]: group_assignment.count(group_labels[1])}
        if pairwise:
            sorter = np.array(group_assignment).argsort()
            unsorter = sorter.argsort()
            ul = np.full((groups[group_labels[0]],) * 2, group_labels[0])
            ur = np.full((groups[group_labels[0]],
                          groups[group_labels[1]]), np.nan)
            ll = np.full((groups[group_labels[1]],
                          groups[group_labels[0]]), np.nan)
            lr = np.full((groups[group_labels[1]],) * 2, group_labels[1])
            group_matrix = np.vstack((np.hstack((ul, ur)),
                                      np.hstack((ll, lr))))
            np.fill_diagonal(group_matrix, np.nan)
            gp['group_matrix'] = group_matrix
            gp['group_selector'] = squareform(
                group_matrix[unsorter, :][:, unsorter], checks=False)
            gp['sorter'] = sorter
            gp['unsorter'] = unsorter
        else:
            gp['group_selector'] = group_assignment
        gp['groups'] = groups
        gp['group_labels'] = group_labels
    elif not group_assignment or len(np.unique(group_assignment)) == 1:
        gp['n_groups'] = 1
        if pairwise:
            gp['group_matrix'] = np.ones((n_subjects, n_subjects))
    elif len(np.unique(group_assignment)) > 2:
        raise ValueError("This test is not valid for more than "
                         "2 groups! (got {0})".format(
                             len(np.unique(group_assignment))))
    else:
        raise ValueError("Invalid group assignments!")
    return gp


def _permute_one_sample_iscs(iscs, group_parameters, i, pairwise=False,
                             summary_statistic='median', group_matrix=None,
                             exact_permutations=None, prng=None):
    if exact_permutations:
        sign_flipper = np.array(exact_permutations[i])
    else:
        sign_flipper = prng.choice(
            [-1, 1], size=group_parameters['n_subjects'], replace=True)
    if pairwise:
        matrix_flipped = (group_parameters['group_matrix'] * sign_flipper
                          * sign_flipper[:, np.newaxis])
        sign_flipper = squareform(matrix_flipped, checks=False)
    isc_flipped = iscs * sign_flipper[:, np.newaxis]
    return compute_summary_statistic(
        isc_flipped, summary_statistic=summary_statistic, axis=0)


def _permute_two_sample_iscs(iscs, group_parameters, i, pairwise=False,
                             summary_statistic='median',
                             exact_permutations=None, prng=None):
    if exact_permutations:
        group_shuffler = np.array(exact_permutations[i])
    elif pairwise:
        group_shuffler = prng.permutation(np.arange(
            len(np.array(group_parameters['group_assignment'])[
                group_parameters['sorter']])))
    else:
        group_shuffler = prng.permutation(np.arange(
            len(group_parameters['group_assignment'])))

    if pairwise:
        group_shuffled = group_parameters['group_matrix'][
            group_shuffler, :][:, group_shuffler]
        group_selector = squareform(
            group_shuffled[group_parameters['unsorter'], :]
            [:, group_parameters['unsorter']], checks=False)
    else:
        group_selector = np.array(
            group_parameters['group_assignment'])[group_shuffler]

    labels = group_parameters['group_labels']
    return (compute_summary_statistic(
        iscs[group_selector == labels[0], :],
        summary_statistic=summary_statistic, axis=0)
        - compute_summary_statistic(
            iscs[group_selector == labels[1], :],
            summary_statistic=summary_statistic, axis=0))


def permutation_isc(iscs, group_assignment=None, pairwise=False,  # noqa: C901
                    summary_statistic='median', n_permutations=1000,
                    side='right', random_state=None):
    """One-sample (sign-flip) / two-sample (label-shuffle) permutation test;
    exact enumeration when the permutation space fits in n_permutations."""
    iscs, n_subjects, n_voxels = _check_isc_input(iscs, pairwise=pairwise)
    if summary_statistic not in ('mean', 'median'):
        raise ValueError("Summary statistic must be 'mean' or 'median'")
    group_assignment = _check_group_assignment(group_assignment, n_subjects)
    gp = _get_group_parameters(group_assignment, n_subjects,
                               pairwise=pairwise)

    if gp['n_groups'] == 1:
        if n_permutations < 2 ** n_subjects:
            exact_permutations = None
        else:
            exact_permutations = list(product([-1, 1], repeat=n_subjects))
            n_permutations = 2 ** n_subjects
    else:
        if n_permutations < math.factorial(n_subjects):
            exact_permutations = None
        else:
            exact_permutations = list(permutations(np.arange(
                len(group_assignment))))
            n_permutations = math.factorial(n_subjects)

    if gp['n_groups'] == 1:
        observed = compute_summary_statistic(
            iscs, summary_statistic=summary_statistic,
            axis=0)[np.newaxis, :]
    else:
        labels = gp['group_labels']
        observed = np.array(
            compute_summary_statistic(
                iscs[np.asarray(gp['group_selector']) == labels[0], :],
                summary_statistic=summary_statistic, axis=0)
            - compute_summary_statistic(
                iscs[np.asarray(gp['group_selector']) == labels[1], :],
                summary_statistic=summary_statistic, axis=0))

    distribution = []
    for i in np.arange(n_permutations):
        if exact_permutations:
            prng = None
        elif isinstance(random_state, np.random.RandomState):
            prng = random_state
        else:
            prng = np.random.RandomState(random_state)
        if gp['n_groups'] == 1:
            isc_sample = _permute_one_sample_iscs(
                iscs, gp, i, pairwise=pairwise,
                summary_statistic=summary_statistic,
                exact_permutations=exact_permutations, prng=prng)
        else:
            isc_sample = _permute_two_sample_iscs(
                iscs, gp, i, pairwise=pairwise,
                summary_statistic=summary_statistic,
                exact_permutations=exact_permutations, prng=prng)
        distribution.append(isc_sample)
        if not exact_permutations:
            random_state = np.random.RandomState(
                prng.randint(0, MAX_RANDOM_SEED, dtype=np.int64))

    distribution = np.array(distribution)
    p = p_from_null(observed, distribution, side=side,
                    exact=bool(exact_permutations), axis=0)
    return observed, p, distribution


def timeshift_isc(data, pairwise=False, summary_statistic='median',
                  n_shifts=1000, side='right', tolerate_nans=True,
                  random_state=None):
    """Circular time-shift null distribution for a one-sample ISC test."""
    data, n_TRs, n_voxels, n_subjects = _check_timeseries_input(data)
    observed = isc(data, pairwise=pairwise,
                   summary_statistic=summary_statistic,
                   tolerate_nans=tolerate_nans)

    rolled = np.rollaxis(data, 2, 0) if pairwise else data

    distribution = []
    for _ in np.arange(n_shifts):
        if isinstance(random_state, np.random.RandomState):
            prng = random_state
        else:
            prng = np.random.RandomState(random_state)
        shifts = prng.choice(np.arange(n_TRs), size=n_subjects,
                             replace=True)
        if pairwise:
            shifted_data = []
            for subject, shift in zip(rolled, shifts):
                shifted_data.append(np.concatenate(
                    (subject[-shift:, :], subject[:-shift, :])))
            shifted_data = np.dstack(shifted_data)
            shifted_isc = isc(shifted_data, pairwise=True,
                              summary_statistic=summary_statistic,
                              tolerate_nans=tolerate_nans)
        else:
            shifted_isc = []
            for s, shift in enumerate(shifts):
                shifted_subject = np.concatenate(
                    (data[-shift:, :, s], data[:-shift, :, s]))
                nonshifted_mean = np.mean(np.delete(data, s, 2), axis=2)
                loo_isc = isc(np.dstack((shifted_subject, nonshifted_mean)),
                              pairwise=False, summary_statistic=None,
                              tolerate_nans=tolerate_nans)
                shifted_isc.append(loo_isc)
            shifted_isc = compute_summary_statistic(
                np.dstack(shifted_isc),
                summary_statistic=summary_statistic, axis=2)
        distribution.append(shifted_isc)
        random_state = np.random.RandomState(
            prng.randint(0, MAX_RANDOM_SEED, dtype=np.int64))

    distribution = np.array(distribution).reshape(n_shifts, n_voxels)
    p = p_from_null(observed, distribution, side=side, exact=False, axis=0)
    return observed, p, distribution


def phaseshift_isc(data, pairwise=False, summary_statistic='median',
                   n_shifts=1000, side='right', tolerate_nans=True,
                   random_state=None):
    """FFT phase-randomization null distribution for an ISC test."""
    data, n_TRs, n_voxels, n_subjects = _check_timeseries_input(data)
    observed = isc(data, pairwise=pairwise,
                   summary_statistic=summary_statistic,
                   tolerate_nans=tolerate_nans)

    distribution = []
    for _ in np.arange(n_shifts):
        if isinstance(random_state, np.random.RandomState):
            prng = random_state
        else:
            prng = np.random.RandomState(random_state)
        shifted_data = phase_randomize(data, random_state=prng)
        if pairwise:
            shifted_isc = isc(shifted_data, pairwise=True,
                              summary_statistic=summary_statistic,
                              tolerate_nans=tolerate_nans)
        else:
            rolled = np.rollaxis(shifted_data, 2, 0)
            shifted_isc = []
            for s, shifted_subject in enumerate(rolled):
                nonshifted_mean = np.mean(np.delete(data, s, axis=2),
                                          axis=2)
                loo_isc = isc(np.dstack((shifted_subject,
                                         nonshifted_mean)),
                              pairwise=False, summary_statistic=None,
                              tolerate_nans=tolerate_nans)
                shifted_isc.append(loo_isc)
            shifted_isc = compute_summary_statistic(
                np.dstack(shifted_isc),
                summary_statistic=summary_statistic, axis=2)
        distribution.append(shifted_isc)
        random_state = np.random.RandomState(
            prng.randint(0, MAX_RANDOM_SEED, dtype=np.int64))

    distribution = np.array(distribution).reshape(n_shifts, n_voxels)
    p = p_from_null(observed, distribution, side=side, exact=False, axis=0)
    return observed, p, distribution
