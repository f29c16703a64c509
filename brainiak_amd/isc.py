"""Intersubject correlation (ISC/ISFC) and its nonparametric statistics.

API parity with the reference isc module (ref src/brainiak/isc.py:81-1551):
``isc``, ``isfc``, ``compute_summary_statistic``, ``squareform_isfc``,
``bootstrap_isc`` (subject bootstrap with Hall-Wilson shift),
``permutation_isc`` (one-sample sign-flip / two-sample label shuffle, exact
tests when the permutation space is small), ``timeshift_isc`` (circular
shifts), ``phaseshift_isc`` (FFT phase randomization).

The implementations are vectorized rather than per-iteration loops:
pairwise ISC is one einsum over z-scored series, leave-one-out means
use the sum trick (total minus subject) instead of ``np.delete``, the
bootstrap resamples all iterations with one fancy-index, and the
one-sample permutation test applies every sign-flip as a single
matrix product.

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
import os
from itertools import combinations, permutations

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


def _as_rng(random_state):
    """Coerce None / int seed / RandomState into a RandomState."""
    if isinstance(random_state, np.random.RandomState):
        return random_state
    return np.random.RandomState(random_state)


def _drop_bad_voxels(data, tolerate_nans):
    """Voxel NaN policy shared by isc/isfc.

    A voxel is dropped when every subject has a NaN somewhere in its
    series; a float policy in [0, 1] additionally requires that
    fraction of subjects to be NaN-free.  Returns (filtered data,
    keep mask over voxels).
    """
    nan_somewhere = np.any(np.isnan(data), axis=0)      # [voxel, subject]
    keep = ~np.all(nan_somewhere, axis=1)
    if tolerate_nans is True:
        logger.info("ISC: averaging will tolerate all NaNs")
    elif isinstance(tolerate_nans, float):
        if not 0.0 <= tolerate_nans <= 1.0:
            raise ValueError("If threshold to tolerate NaNs is a float, "
                             "it must be between 0.0 and 1.0; got "
                             f"{tolerate_nans}")
        clean_subjects = np.sum(~nan_somewhere, axis=1)
        keep &= clean_subjects >= data.shape[-1] * tolerate_nans
        logger.info("ISC: voxels need a %s fraction of NaN-free "
                    "subjects; %d voxels dropped", tolerate_nans,
                    int(np.sum(~keep)))
    else:
        logger.info("ISC: averaging will not tolerate NaNs")
    return data[:, keep, :], keep


def _loo_means(data, use_nanmean):
    """Leave-one-out across-subject means, [TR, voxel, subject], via
    the sum trick (one pass instead of n_subjects ``np.delete``s)."""
    n = data.shape[-1]
    if use_nanmean:
        finite = np.isfinite(data)
        totals = np.nansum(data, axis=2, keepdims=True)
        counts = finite.sum(axis=2, keepdims=True)
        num = totals - np.where(finite, data, 0.0)
        den = counts - finite
        with np.errstate(invalid="ignore", divide="ignore"):
            return num / den
    totals = data.sum(axis=2, keepdims=True)
    return (totals - data) / (n - 1)


def _pairwise_voxel_correlations(data):
    """r for every subject pair at every voxel in one einsum.

    data [TR, voxel, subject] → [pair, voxel] with pairs in
    ``combinations`` (row-major upper-triangle) order.
    """
    centered = data - data.mean(axis=0)
    norms = np.sqrt(np.einsum("tvs,tvs->vs", centered, centered))
    with np.errstate(invalid="ignore", divide="ignore"):
        unit = centered / norms
    gram = np.einsum("tvi,tvj->vij", unit, unit)        # [voxel, s, s]
    iu = np.triu_indices(data.shape[-1], k=1)
    return gram[:, iu[0], iu[1]].T


def isc(data, pairwise=False, summary_statistic=None, tolerate_nans=True):
    """Intersubject correlation per voxel (leave-one-out or pairwise)."""
    data, n_TRs, n_voxels, n_subjects = _check_timeseries_input(data)

    if n_subjects == 2:
        logger.info("Only two subjects! Simply computing Pearson "
                    "correlation.")
        summary_statistic = None

    data, keep = _drop_bad_voxels(data, tolerate_nans)

    if n_subjects == 2:
        rows = array_correlation(data[..., 0], data[..., 1])[None, :]
    elif pairwise:
        rows = _pairwise_voxel_correlations(data)
    else:
        loo = _loo_means(data, use_nanmean=bool(tolerate_nans))
        rows = np.stack([array_correlation(data[..., s], loo[..., s])
                         for s in range(n_subjects)])

    iscs = np.full((rows.shape[0], n_voxels), np.nan)
    iscs[:, keep] = rows

    if summary_statistic:
        iscs = compute_summary_statistic(
            iscs, summary_statistic=summary_statistic, axis=0)[None, :]
    return iscs[0] if iscs.shape[0] == 1 else iscs


def _resolve_targets(targets, data):
    """ISFC target handling: no targets → self-ISFC (symmetric)."""
    if not isinstance(targets, (np.ndarray, list)):
        t, v, s = data.shape
        return data, t, v, s, True
    targets, n_TRs, n_voxels, n_subjects = (
        _check_timeseries_input(targets))
    if data.shape[0] != n_TRs:
        raise ValueError("Targets array must have same number of "
                         "TRs as input data")
    if data.shape[2] != n_subjects:
        raise ValueError("Targets array must have same number of "
                         "subjects as input data")
    return targets, n_TRs, n_voxels, n_subjects, False


def isfc(data, targets=None, pairwise=False, summary_statistic=None,
         vectorize_isfcs=True, tolerate_nans=True):
    """Intersubject functional correlation: correlations between each
    voxel's time series and every (target) voxel in other subjects."""
    data, n_TRs, n_voxels, n_subjects = _check_timeseries_input(data)
    targets, _, _, _, symmetric = _resolve_targets(targets, data)
    if not symmetric:
        pairwise = False
    data, keep = _drop_bad_voxels(data, tolerate_nans)
    targets, keep_t = _drop_bad_voxels(targets, tolerate_nans)

    def corr(a, b):
        return compute_correlation(np.ascontiguousarray(a.T),
                                   np.ascontiguousarray(b.T),
                                   return_nans=True)

    if symmetric and n_subjects == 2:
        logger.info("Only two subjects! Computing ISFC between them.")
        m = corr(data[..., 0], data[..., 1])
        mats = [(m + m.T) / 2]
        summary_statistic = None
    elif pairwise:
        mats = []
        for a, b in combinations(range(n_subjects), 2):
            m = corr(data[..., a], targets[..., b])
            mats.append((m + m.T) / 2 if symmetric else m)
    else:
        use_nan = bool(tolerate_nans)
        loo = _loo_means(targets, use_nanmean=use_nan)
        mats = []
        for s in range(n_subjects):
            m = corr(data[..., s], loo[..., s])
            mats.append((m + m.T) / 2 if symmetric else m)

    expanded = np.full((len(mats), len(keep), len(keep_t)), np.nan)
    expanded[np.ix_(np.arange(len(mats)), keep, keep_t)] = \
        np.stack(mats)
    isfcs = expanded

    if summary_statistic:
        isfcs = compute_summary_statistic(
            isfcs, summary_statistic=summary_statistic, axis=0)
    if isfcs.shape[0] == 1:
        isfcs = isfcs[0]
    if vectorize_isfcs and symmetric:
        return squareform_isfc(isfcs)
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
        if use_hip and precision == 'bf16' and local \
                and os.environ.get("BRAINIAK_ISFC_FUSED"):
            # fully fused path (OPT-IN): correlation tile pairs,
            # symmetrize, atanh and accumulation in ONE kernel — the
            # per-subject [V, V] matrix never touches HBM.  Measured
            # 270 vs 259 ms against the GEMM+accum pipeline at 50k x
            # 32: the hand MFMA schedule gives back more than the M
            # round trip saves (hipBLASLt's GEMM is ~3x our MFMA
            # utilization), so the GEMM path stays the default —
            # profiles/NEXT.md lists the tiling work that would flip
            # it.
            Zs = torch.stack([nd.T.contiguous().to(torch.bfloat16)
                              for nd in normed_local])
            Zm = torch.stack([
                _norm((total - d) / (n_total - 1)).T.contiguous()
                .to(torch.bfloat16) for d in local])
            _ops.isfc_fused_(acc, Zs, Zm)
            del Zs, Zm
        elif use_hip:
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


def compute_summary_statistic(iscs, summary_statistic='mean', axis=None):
    """'mean' (Fisher-Z mean: tanh(mean(arctanh))) or 'median' of ISCs."""
    reducers = {
        'mean': lambda a: np.tanh(np.nanmean(np.arctanh(a), axis=axis)),
        'median': lambda a: np.nanmedian(a, axis=axis),
    }
    if summary_statistic not in reducers:
        raise ValueError("Summary statistic must be 'mean' or 'median'")
    return reducers[summary_statistic](iscs)


def squareform_isfc(isfcs, iscs=None):
    """Square ↔ condensed ISFC conversion retaining the ISC diagonal.

    Square [.., V, V] input (no iscs) → (condensed off-diagonals,
    diagonal ISCs); condensed input + iscs → square matrices.
    """
    have_iscs = isinstance(iscs, np.ndarray)
    square_in = not have_iscs and isfcs.shape[-2] == isfcs.shape[-1]

    if square_in:
        stack = isfcs[None] if isfcs.ndim == 2 else isfcs
        if stack.ndim != 3:
            raise ValueError("Square (redundant) ISFCs must be square "
                             "with multiple subjects or pairs of "
                             "subjects indexed by the first dimension")
        V = stack.shape[-1]
        iu = np.triu_indices(V, k=1)
        condensed = stack[:, iu[0], iu[1]]
        diags = np.diagonal(stack, axis1=1, axis2=2)
        if condensed.shape[0] == 1:
            return condensed[0], diags[0]
        return condensed, diags

    flat = isfcs[None] if isfcs.ndim == 1 else isfcs
    diag = iscs[None] if iscs.ndim == 1 else iscs
    V = diag.shape[-1]
    iu = np.triu_indices(V, k=1)
    squares = np.zeros((flat.shape[0], V, V), dtype=flat.dtype)
    squares[:, iu[0], iu[1]] = flat
    squares += np.transpose(squares, (0, 2, 1))
    squares[:, np.arange(V), np.arange(V)] = diag
    return squares[0] if squares.shape[0] == 1 else squares


# --------------------------------------------------------------------------
# nonparametric statistics
# --------------------------------------------------------------------------

def _coerce_iscs(iscs, pairwise):
    """Normalize ISC input to [row, voxel] and infer subject count."""
    iscs = np.asarray(iscs)
    if iscs.ndim == 1:
        iscs = iscs[:, None]
    if pairwise:
        try:
            n_subjects = squareform(iscs[:, 0],
                                    force='tomatrix').shape[0]
        except ValueError:
            raise ValueError("For pairwise input, ISCs must be the "
                             "vectorized triangle of a square matrix.")
    else:
        n_subjects = iscs.shape[0]
    logger.info("Nonparametric ISC test: %d subjects, %d voxel(s)/"
                "ROI(s)", n_subjects, iscs.shape[1])
    return iscs, n_subjects, iscs.shape[1]


def bootstrap_isc(iscs, pairwise=False, summary_statistic='median',
                  n_bootstraps=1000, ci_percentile=95, side='right',
                  random_state=None):
    """One-sample subject-level bootstrap with the Hall-Wilson shift.

    All ``n_bootstraps`` subject resamples are drawn up front and the
    summary statistic is applied to the whole [boot, subject, voxel]
    stack at once.  Resampled pairs of a subject with itself (pairwise
    mode) carry no information and are excluded as NaN.
    """
    iscs, n_subjects, n_voxels = _coerce_iscs(iscs, pairwise)
    if summary_statistic not in ('mean', 'median'):
        raise ValueError("Summary statistic must be 'mean' or 'median'")
    observed = compute_summary_statistic(
        iscs, summary_statistic=summary_statistic, axis=0)

    rng = _as_rng(random_state)
    draws = np.sort(rng.randint(0, n_subjects,
                                size=(n_bootstraps, n_subjects)), axis=1)

    if pairwise:
        # square per-voxel matrices once, then each bootstrap is a
        # row/column gather + upper-triangle read
        V = n_voxels
        iu = np.triu_indices(n_subjects, k=1)
        sq = np.zeros((n_subjects, n_subjects, V))
        sq[iu[0], iu[1]] = iscs
        sq += np.transpose(sq, (1, 0, 2))
        gathered = sq[draws[:, :, None],
                      draws[:, None, :]]        # [boot, s, s, V]
        samples = gathered[:, iu[0], iu[1]]     # [boot, pair, V]
        self_pair = draws[:, iu[0]] == draws[:, iu[1]]
        samples[self_pair] = np.nan
    else:
        samples = iscs[draws]                   # [boot, subject, voxel]

    distribution = compute_summary_statistic(
        samples, summary_statistic=summary_statistic, axis=1)

    half_alpha = (100 - ci_percentile) / 2
    ci = (np.percentile(distribution, half_alpha, axis=0),
          np.percentile(distribution, 100 - half_alpha, axis=0))
    p = p_from_null(observed, distribution - observed, side=side,
                    exact=False, axis=0)
    return observed, ci, p, distribution


class _GroupLayout:
    """Two-group bookkeeping for permutation_isc: label order, the
    pairwise label matrix, and the observed group selector."""

    def __init__(self, assignment, n_subjects, pairwise):
        self.assignment = assignment
        self.n_subjects = n_subjects
        uniques = np.unique(assignment) if assignment else np.array([])
        self.n_groups = max(1, len(uniques))
        if self.n_groups > 2:
            raise ValueError("This test is not valid for more than "
                             f"2 groups! (got {self.n_groups})")
        self.labels = uniques if self.n_groups == 2 else None
        self.matrix = None
        self.selector = None
        if self.n_groups == 2:
            a = np.asarray(assignment)
            if pairwise:
                # label matrix: within-group pairs carry the group
                # label, cross-group pairs are NaN
                lab = np.where(a[:, None] == a[None, :],
                               np.broadcast_to(a, (n_subjects,
                                                   n_subjects)),
                               np.nan).astype(float)
                np.fill_diagonal(lab, np.nan)
                self.matrix = lab
                self.selector = squareform(lab, checks=False)
            else:
                self.selector = a
        elif pairwise:
            self.matrix = np.ones((n_subjects, n_subjects))


def _normalize_assignment(group_assignment, n_subjects):
    if isinstance(group_assignment, np.ndarray):
        group_assignment = group_assignment.tolist()
    elif not isinstance(group_assignment, list):
        logger.info("No group assignment provided, "
                    "performing one-sample test.")
        group_assignment = None
    if group_assignment and len(group_assignment) != n_subjects:
        raise ValueError(
            f"Group assignments ({len(group_assignment)}) do not "
            f"match number of subjects ({n_subjects})!")
    return group_assignment


def _group_difference(iscs, selector, labels, summary_statistic):
    sel = np.asarray(selector)
    first = compute_summary_statistic(
        iscs[sel == labels[0]], summary_statistic=summary_statistic,
        axis=0)
    second = compute_summary_statistic(
        iscs[sel == labels[1]], summary_statistic=summary_statistic,
        axis=0)
    return first - second


def _sign_flip_rows(layout, flips, pairwise):
    """Per-permutation row multipliers from subject sign flips.

    flips [n_perm, n_subjects] → [n_perm, n_rows]: in pairwise mode a
    pair's sign is the product of its two subjects' signs.
    """
    if not pairwise:
        return flips
    iu = np.triu_indices(layout.n_subjects, k=1)
    return flips[:, iu[0]] * flips[:, iu[1]]


def permutation_isc(iscs, group_assignment=None, pairwise=False,
                    summary_statistic='median', n_permutations=1000,
                    side='right', random_state=None):
    """One-sample (sign-flip) / two-sample (label-shuffle) permutation
    test; exact enumeration when the permutation space fits in
    n_permutations."""
    iscs, n_subjects, n_voxels = _coerce_iscs(iscs, pairwise)
    if summary_statistic not in ('mean', 'median'):
        raise ValueError("Summary statistic must be 'mean' or 'median'")
    assignment = _normalize_assignment(group_assignment, n_subjects)
    layout = _GroupLayout(assignment, n_subjects, pairwise)
    rng = _as_rng(random_state)

    one_sample = layout.n_groups == 1
    space = 2 ** n_subjects if one_sample \
        else math.factorial(n_subjects)
    exact = n_permutations >= space

    if one_sample:
        observed = compute_summary_statistic(
            iscs, summary_statistic=summary_statistic, axis=0)[None, :]
        if exact:
            # every sign pattern via the bits of 0..2^n-1
            codes = np.arange(space)[:, None] >> np.arange(n_subjects)
            flips = 1.0 - 2.0 * (codes & 1)
            n_permutations = space
        else:
            flips = rng.choice([-1.0, 1.0],
                               size=(n_permutations, n_subjects))
        row_signs = _sign_flip_rows(layout, flips, pairwise)
        # [perm, row, voxel] in one broadcast multiply
        flipped = iscs[None] * row_signs[:, :, None]
        distribution = compute_summary_statistic(
            flipped, summary_statistic=summary_statistic, axis=1)
    else:
        observed = np.array(_group_difference(
            iscs, layout.selector, layout.labels, summary_statistic))
        if exact:
            orders = list(permutations(range(n_subjects)))
            n_permutations = space
        else:
            orders = [rng.permutation(n_subjects)
                      for _ in range(n_permutations)]
        rows = []
        base = np.asarray(layout.assignment)
        for order in orders:
            if pairwise:
                shuffled_matrix = layout.matrix[np.ix_(order, order)]
                selector = squareform(shuffled_matrix, checks=False)
            else:
                selector = base[np.asarray(order)]
            rows.append(_group_difference(
                iscs, selector, layout.labels, summary_statistic))
        distribution = np.array(rows)

    p = p_from_null(observed, distribution, side=side, exact=exact,
                    axis=0)
    return observed, p, distribution


def _null_from_resamples(data, observed, n_iter, side, draw_fn):
    """Shared shell of the timeshift/phaseshift tests: build the null
    distribution row by row from draw_fn(iteration) and convert to p."""
    rows = [np.asarray(draw_fn(i)).ravel() for i in range(n_iter)]
    distribution = np.vstack(rows)
    p = p_from_null(observed, distribution, side=side, exact=False,
                    axis=0)
    return observed, p, distribution


def _loo_null_isc(surrogate, data, summary_statistic, tolerate_nans):
    """Leave-one-out null: each surrogate subject vs the UNshifted
    mean of the others."""
    totals = data.sum(axis=2)
    n = data.shape[-1]
    cols = []
    for s in range(n):
        others_mean = (totals - data[..., s]) / (n - 1)
        cols.append(isc(np.dstack((surrogate[..., s], others_mean)),
                        pairwise=False, summary_statistic=None,
                        tolerate_nans=tolerate_nans))
    return compute_summary_statistic(
        np.dstack(cols), summary_statistic=summary_statistic, axis=2)


def timeshift_isc(data, pairwise=False, summary_statistic='median',
                  n_shifts=1000, side='right', tolerate_nans=True,
                  random_state=None):
    """Circular time-shift null distribution for a one-sample ISC test."""
    data, n_TRs, n_voxels, n_subjects = _check_timeseries_input(data)
    observed = isc(data, pairwise=pairwise,
                   summary_statistic=summary_statistic,
                   tolerate_nans=tolerate_nans)
    rng = _as_rng(random_state)

    def one_draw(_):
        shifts = rng.randint(0, n_TRs, size=n_subjects)
        rolled = np.dstack([np.roll(data[..., s], shifts[s], axis=0)
                            for s in range(n_subjects)])
        if pairwise:
            return isc(rolled, pairwise=True,
                       summary_statistic=summary_statistic,
                       tolerate_nans=tolerate_nans)
        return _loo_null_isc(rolled, data, summary_statistic,
                             tolerate_nans)

    return _null_from_resamples(data, observed, n_shifts, side, one_draw)


def phaseshift_isc(data, pairwise=False, summary_statistic='median',
                   n_shifts=1000, side='right', tolerate_nans=True,
                   random_state=None):
    """FFT phase-randomization null distribution for an ISC test."""
    data, n_TRs, n_voxels, n_subjects = _check_timeseries_input(data)
    observed = isc(data, pairwise=pairwise,
                   summary_statistic=summary_statistic,
                   tolerate_nans=tolerate_nans)
    rng = _as_rng(random_state)

    def one_draw(_):
        surrogate = phase_randomize(data, random_state=rng)
        if pairwise:
            return isc(surrogate, pairwise=True,
                       summary_statistic=summary_statistic,
                       tolerate_nans=tolerate_nans)
        return _loo_null_isc(surrogate, data, summary_statistic,
                             tolerate_nans)

    return _null_from_resamples(data, observed, n_shifts, side, one_draw)
