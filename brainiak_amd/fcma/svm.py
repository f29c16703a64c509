"""Batched precomputed-kernel SVM cross-validation.

The reference scores each voxel's [E, E] kernel matrix with sklearn SVC
inside a multiprocessing pool (ref src/brainiak/fcma/voxelselector.py:423-465)
— thousands of tiny sequential QPs on CPU.  Here the per-(voxel, fold)
dual problems are solved *batched* on the GPU with a vectorized SMO
(maximal-violating-pair working-set selection), so the whole chunk's CV
is a handful of tensor ops per iteration.  The CPU path keeps sklearn
for exact reference parity.
"""

import multiprocessing
from typing import List, Optional, Tuple

import numpy as np
import torch

__all__ = ["cross_validate_voxels", "smo_batch_train", "stratified_folds"]


def stratified_folds(labels: np.ndarray, num_folds: int
                     ) -> List[Tuple[np.ndarray, np.ndarray]]:
    """StratifiedKFold(shuffle=False) splits, same as the reference uses
    (voxelselector.py:44-48).  Delegates to sklearn for exactness."""
    from sklearn.model_selection import StratifiedKFold
    skf = StratifiedKFold(n_splits=num_folds, shuffle=False)
    return list(skf.split(np.zeros(len(labels)), labels))


def smo_batch_train(K: torch.Tensor, y: torch.Tensor, C: float = 1.0,
                    tol: float = 1e-3, max_iter: int = 2000
                    ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Solve B independent C-SVC duals with precomputed kernels.

    Parameters
    ----------
    K : [B, n, n] kernel matrices.
    y : [B, n] labels in {-1, +1} (float).
    C : box constraint.

    Returns
    -------
    alpha : [B, n] dual coefficients.
    b : [B] biases.

    One SMO iteration updates the maximal-violating pair of every
    problem simultaneously; problems that reached the KKT gap tolerance
    stay fixed (their step size is zero).
    """
    B, n, _ = K.shape
    device = K.device
    alpha = torch.zeros((B, n), dtype=K.dtype, device=device)
    grad = -torch.ones((B, n), dtype=K.dtype, device=device)  # Qα - 1
    Q = K * (y[:, :, None] * y[:, None, :])
    arangeB = torch.arange(B, device=device)
    NEG = torch.finfo(K.dtype).min / 4

    for _ in range(max_iter):
        # -y*grad, maximized over I_up, minimized over I_low
        score = -y * grad
        up_mask = ((y > 0) & (alpha < C - 1e-12)) | \
                  ((y < 0) & (alpha > 1e-12))
        low_mask = ((y > 0) & (alpha > 1e-12)) | \
                   ((y < 0) & (alpha < C - 1e-12))
        up_score = torch.where(up_mask, score, torch.full_like(score, NEG))
        low_score = torch.where(low_mask, score, torch.full_like(score, -NEG))
        gmax, i = up_score.max(dim=1)
        gmin, j = low_score.min(dim=1)
        gap = gmax - gmin
        if bool((gap < tol).all()):
            break
        active = gap >= tol

        yi = y[arangeB, i]
        yj = y[arangeB, j]
        Kii = Q[arangeB, i, i]
        Kjj = Q[arangeB, j, j]
        Kij = Q[arangeB, i, j]
        # η = K_ii + K_jj − 2K_ij expressed through Q (Q_ij = y_i y_j K_ij)
        eta = (Kii + Kjj - 2.0 * Kij * (yi * yj)).clamp_min(1e-12)

        # unconstrained optimal step along (Δα_i, Δα_j) = (y_i, −y_j)·t
        t = (gmax - gmin) / eta
        t = torch.where(active, t, torch.zeros_like(t))
        # clip by box constraints
        ai, aj = alpha[arangeB, i], alpha[arangeB, j]
        t_max_i = torch.where(yi > 0, C - ai, ai)
        t_max_j = torch.where(yj > 0, aj, C - aj)
        t = torch.minimum(t, torch.minimum(t_max_i, t_max_j)).clamp_min(0.0)

        dai = yi * t
        daj = -yj * t
        alpha[arangeB, i] += dai
        alpha[arangeB, j] += daj
        # rank-2 gradient update
        grad += Q[arangeB, i, :] * dai[:, None] + \
            Q[arangeB, j, :] * daj[:, None]

    # bias: average of the violating boundary scores
    score = -y * grad
    up_mask = ((y > 0) & (alpha < C - 1e-12)) | ((y < 0) & (alpha > 1e-12))
    low_mask = ((y > 0) & (alpha > 1e-12)) | ((y < 0) & (alpha < C - 1e-12))
    up_score = torch.where(up_mask, score, torch.full_like(score, NEG))
    low_score = torch.where(low_mask, score, torch.full_like(score, -NEG))
    b = (up_score.max(dim=1).values + low_score.min(dim=1).values) / 2.0
    return alpha, b


class FoldPlan:
    """Device-resident stratified-fold index tables, built once and
    reused for every chunk's batched-SMO launch (the per-call rebuild
    plus the max-fold-size .item() sync would otherwise serialize the
    overlapped CV stream against the duo sweep)."""

    def __init__(self, labels: np.ndarray, num_folds: int, device):
        labels = np.asarray(labels)
        E = len(labels)
        classes = np.unique(labels)
        if len(classes) != 2:
            raise ValueError("GPU batched SVM supports binary labels; "
                             f"got {len(classes)} classes")
        y_np = np.where(labels == classes[1], 1.0,
                        -1.0).astype(np.float32)
        folds = stratified_folds(labels, num_folds)
        F = len(folds)
        train_idx = np.zeros((F, E), dtype=np.int32)
        test_idx = np.zeros((F, E), dtype=np.int32)
        n_train = np.zeros(F, dtype=np.int32)
        n_test = np.zeros(F, dtype=np.int32)
        for f, (tr, te) in enumerate(folds):
            train_idx[f, :len(tr)] = tr
            test_idx[f, :len(te)] = te
            n_train[f] = len(tr)
            n_test[f] = len(te)
        self.max_n = int(max(n_train.max(), n_test.max()))
        self.y = torch.as_tensor(y_np, device=device)
        self.train_idx = torch.as_tensor(train_idx, device=device)
        self.test_idx = torch.as_tensor(test_idx, device=device)
        self.n_train = torch.as_tensor(n_train, device=device)
        self.n_test = torch.as_tensor(n_test, device=device)
        self.n_test_f = self.n_test.to(torch.float32)


def svm_cv_device(kernels: torch.Tensor, plan: FoldPlan, C: float,
                  tol: float) -> torch.Tensor:
    """Batched-SMO CV on the CURRENT stream, returning the per-voxel
    mean accuracy as a DEVICE tensor — no host sync, so it can be
    enqueued on a side stream while the duo sweep continues."""
    from .. import ops
    correct = ops.svm_cv(
        kernels.to(torch.float32).contiguous(),
        plan.y, plan.train_idx, plan.test_idx,
        plan.n_train, plan.n_test, C=C, tol=tol,
        max_n=plan.max_n)
    return (correct.to(torch.float32) / plan.n_test_f).mean(dim=1)


def _accuracy_gpu_hip(kernels: torch.Tensor, labels: np.ndarray,
                      num_folds: int, C: float, tol: float) -> np.ndarray:
    """Whole-CV-in-one-launch path: one wavefront per (voxel, fold) QP
    (ops.svm_cv HIP kernel)."""
    plan = FoldPlan(labels, num_folds, kernels.device)
    return svm_cv_device(kernels, plan, C, tol).cpu().numpy()


def _accuracy_gpu(kernels: torch.Tensor, labels: np.ndarray, num_folds: int,
                  C: float, tol: float) -> np.ndarray:
    """Batched k-fold CV accuracy for kernels [Cvox, E, E]."""
    device = kernels.device
    dtype = torch.float32
    n_vox, E, _ = kernels.shape
    classes = np.unique(labels)
    if len(classes) != 2:
        raise ValueError("GPU batched SVM supports binary labels; got "
                         f"{len(classes)} classes")
    from .. import ops
    if kernels.is_cuda and ops.require_hip():
        folds = stratified_folds(labels, num_folds)
        if max(len(tr) for tr, _ in folds) <= 128 and \
                max(len(te) for _, te in folds) <= 128:
            return _accuracy_gpu_hip(kernels, labels, num_folds, C, tol)
    y_np = np.where(labels == classes[1], 1.0, -1.0)
    folds = stratified_folds(labels, num_folds)

    correct = torch.zeros(n_vox, device=device, dtype=dtype)
    total = 0
    for train_idx, test_idx in folds:
        tr = torch.as_tensor(train_idx, device=device)
        te = torch.as_tensor(test_idx, device=device)
        Ktr = kernels.index_select(1, tr).index_select(2, tr).to(dtype)
        y_tr = torch.as_tensor(y_np[train_idx], device=device,
                               dtype=dtype).expand(n_vox, -1).contiguous()
        alpha, b = smo_batch_train(Ktr, y_tr, C=C, tol=tol)
        # decision for test rows: K[test, train] @ (alpha * y) + b
        Kte = kernels.index_select(1, te).index_select(2, tr).to(dtype)
        coef = alpha * y_tr
        dec = torch.bmm(Kte, coef[:, :, None]).squeeze(2) + b[:, None]
        y_te = torch.as_tensor(y_np[test_idx], device=device, dtype=dtype)
        pred = torch.where(dec > 0, 1.0, -1.0)
        correct += (pred == y_te[None, :]).to(dtype).sum(dim=1)
        total += len(test_idx)
    return (correct / total).cpu().numpy()


def _accuracy_one_voxel_sklearn(args):
    kernel, labels, num_folds, C, class_weight = args
    from sklearn import model_selection, svm
    clf = svm.SVC(kernel='precomputed', shrinking=False, C=C,
                  class_weight=class_weight)
    skf = model_selection.StratifiedKFold(n_splits=num_folds, shuffle=False)
    scores = model_selection.cross_val_score(clf, kernel, y=labels, cv=skf,
                                             n_jobs=1)
    return float(scores.mean())


def cross_validate_voxels(kernels: torch.Tensor, labels: np.ndarray,
                          num_folds: int, C: float = 1.0,
                          tol: float = 1e-3,
                          process_num: Optional[int] = 0) -> np.ndarray:
    """Mean CV accuracy per voxel for kernels [Cvox, E, E].

    GPU tensors use the batched SMO; CPU uses sklearn (optionally in a
    process pool, mirroring the reference's use_multiprocessing switch).
    """
    labels = np.asarray(labels)
    if kernels.is_cuda:
        return _accuracy_gpu(kernels, labels, num_folds, C, tol)
    k_np = kernels.cpu().numpy().astype(np.float64)
    jobs = [(k_np[i], labels, num_folds, C, None)
            for i in range(k_np.shape[0])]
    if process_num and len(jobs) > 8:
        with multiprocessing.Pool(process_num) as pool:
            accs = pool.map(_accuracy_one_voxel_sklearn, jobs)
    else:
        accs = [_accuracy_one_voxel_sklearn(j) for j in jobs]
    return np.asarray(accs)
