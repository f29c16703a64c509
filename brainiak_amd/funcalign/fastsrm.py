"""FastSRM: atlas-accelerated SRM for large datasets.

API parity with the reference (ref src/brainiak/funcalign/fastsrm.py:
1252-1767): data are first projected onto an atlas (probabilistic
[n_supervoxels, n_voxels] or deterministic label [n_voxels]), the
shared response is fit in the reduced space (DetSRM-style BCD), and
each subject's full-resolution basis is recovered from S·Xᵀ by
orthogonal Procrustes.  Inputs may be arrays, lists of arrays (one per
subject), lists of lists (subject × session) or paths to .npy files;
``temp_dir`` spills bases to disk (the reference's low_ram mode — at
288 GB HBM/host RAM this is rarely needed, but the path API is kept).

Citation: [Richard2019] "Fast shared response model for fMRI data",
arXiv 1909.12537.
"""

import logging
import os
import uuid

import numpy as np

from .srm import DetSRM, NotFittedError, _polar_orthogonal

logger = logging.getLogger(__name__)

__all__ = ["FastSRM"]


def safe_load(data):
    """Load if a path, pass through if an array."""
    if isinstance(data, (str, os.PathLike)):
        return np.load(data)
    return data


def _compute_subject_basis(corr_mat):
    """Procrustes basis from the [n_components, n_voxels] correlation
    S·Xᵀ: argmin_W ||X - Wᵀ... || with WWᵀ = I (ref fastsrm.py:925)."""
    import torch
    A = torch.as_tensor(np.ascontiguousarray(corr_mat.T),
                        dtype=torch.float64)   # [v, k]
    W = _polar_orthogonal(A, perturb=0.0)
    return W.numpy().T                         # [k, v]


class FastSRM:
    """FastSRM estimator.

    Parameters (reference-compatible): atlas, n_components, n_iter,
    temp_dir, low_ram, seed, n_jobs, verbose, aggregate ('mean'|None).
    Attribute after fit: ``basis_list`` — per-subject [k, n_voxels]
    bases (arrays, or paths when temp_dir is set).
    """

    def __init__(self, atlas=None, n_components=20, n_iter=100,
                 temp_dir=None, low_ram=False, seed=None, n_jobs=1,
                 verbose="warn", aggregate="mean"):
        self.atlas = atlas
        self.n_components = n_components
        self.n_iter = n_iter
        self.low_ram = low_ram
        self.seed = seed if seed is not None else 0
        self.n_jobs = n_jobs
        self.verbose = verbose
        if aggregate is not None and aggregate != "mean":
            raise ValueError("aggregate can have only value mean or None")
        self.aggregate = aggregate
        self.basis_list = None
        if temp_dir is None:
            self.temp_dir = None
        else:
            self.temp_dir = os.path.join(str(temp_dir),
                                         "fastsrm" + str(uuid.uuid4()))

    # -- helpers -----------------------------------------------------------

    @staticmethod
    def _canonicalize(imgs):
        """→ list (subjects) of lists (sessions) of [V, T] arrays/paths,
        plus a flag describing the input style."""
        if isinstance(imgs, np.ndarray) and imgs.dtype.kind in 'US':
            return [list(row) for row in imgs], 'array'
        if isinstance(imgs, np.ndarray) and imgs.ndim == 3:
            return [[imgs[i]] for i in range(imgs.shape[0])], 'list'
        if isinstance(imgs, list):
            if len(imgs) == 0:
                raise ValueError("imgs is empty")
            if isinstance(imgs[0], (list, tuple)):
                return [list(s) for s in imgs], 'list_of_list'
            return [[s] for s in imgs], 'list'
        raise ValueError("Unrecognized imgs input")

    def _reduce(self, data):
        """Project [V, T] data onto the atlas → [T, n_supervoxels]."""
        data = safe_load(data)
        if self.atlas is None:
            return data.T.copy()
        atlas = safe_load(self.atlas)
        if atlas.ndim == 2:        # probabilistic [n_supervoxels, V]
            inv = np.linalg.pinv(atlas)
            return data.T @ inv
        # deterministic labels [V]: mean within each parcel (label 0
        # ignored)
        labels = atlas.astype(int)
        n_parcels = labels.max()
        out = np.zeros((data.shape[1], n_parcels))
        for p in range(1, n_parcels + 1):
            m = labels == p
            if m.any():
                out[:, p - 1] = data[m].mean(axis=0)
        return out

    def clean(self):
        """Remove the temp dir contents created by this estimator."""
        if self.temp_dir is not None and os.path.exists(self.temp_dir):
            for f in os.listdir(self.temp_dir):
                os.remove(os.path.join(self.temp_dir, f))
            os.rmdir(self.temp_dir)
        self.basis_list = None

    # -- fitting -----------------------------------------------------------

    def fit(self, imgs):
        """Fit the shared response and per-subject bases."""
        subjects, _ = self._canonicalize(imgs)
        n_subjects = len(subjects)
        n_sessions = len(subjects[0])
        for s in subjects:
            if len(s) != n_sessions:
                raise ValueError(
                    "All subjects need the same number of sessions")
        if self.temp_dir is not None:
            os.makedirs(self.temp_dir, exist_ok=True)

        # 1. atlas reduction
        reduced = [[self._reduce(sess) for sess in subj]
                   for subj in subjects]

        # 2. shared response in reduced space (DetSRM over concatenated
        #    sessions)
        slices = []
        cur = 0
        for m in range(n_sessions):
            t = reduced[0][m].shape[0]
            slices.append(slice(cur, cur + t))
            cur += t
        X = [np.concatenate(subj, axis=0).T for subj in reduced]
        det = DetSRM(n_iter=self.n_iter, features=self.n_components,
                     rand_seed=self.seed, device="cpu")
        det.fit(X)
        concatenated_s = np.mean(det.transform(X), axis=0).T  # [T, k]
        shared_sessions = [concatenated_s[sl] for sl in slices]

        # 3. full-resolution bases from S·Xᵀ per subject
        basis = []
        for i, subj in enumerate(subjects):
            corr = None
            for j, sess in enumerate(subj):
                data = safe_load(sess)          # [V, T]
                c = shared_sessions[j].T @ data.T   # [k, V]
                corr = c if corr is None else corr + c
            basis_i = _compute_subject_basis(corr)
            if self.temp_dir is None:
                basis.append(basis_i)
            else:
                path = os.path.join(self.temp_dir, "basis_%i.npy" % i)
                np.save(path, basis_i)
                basis.append(path)
        self.basis_list = basis
        self._n_sessions = n_sessions
        return self

    def fit_transform(self, imgs, subjects_indexes=None):
        self.fit(imgs)
        return self.transform(imgs, subjects_indexes=subjects_indexes)

    def transform(self, imgs, subjects_indexes=None):
        """Project imgs into the shared space with the fitted bases."""
        if self.basis_list is None:
            raise NotFittedError("The model fit has not been run yet.")
        subjects, style = self._canonicalize(imgs)
        if subjects_indexes is None:
            subjects_indexes = list(range(len(subjects)))

        per_subject = []
        for subj, idx in zip(subjects, subjects_indexes):
            basis = safe_load(self.basis_list[idx])
            sessions = [basis @ safe_load(s) for s in subj]  # [k, T]
            per_subject.append(sessions)

        n_sessions = len(per_subject[0])
        if self.aggregate == "mean":
            agg = [np.mean([p[j] for p in per_subject], axis=0)
                   for j in range(n_sessions)]
            if style == 'list' and n_sessions == 1:
                return agg[0]
            return agg
        if style == 'list' and n_sessions == 1:
            return [p[0] for p in per_subject]
        return per_subject

    def inverse_transform(self, shared_response, subjects_indexes=None,
                          sessions_indexes=None):
        """Reconstruct voxel-space data from a shared response."""
        if self.basis_list is None:
            raise NotFittedError("The model fit has not been run yet.")
        if subjects_indexes is None:
            subjects_indexes = list(range(len(self.basis_list)))
        single_session = isinstance(shared_response, np.ndarray)
        sessions = [shared_response] if single_session else \
            list(shared_response)
        if sessions_indexes is not None:
            sessions = [sessions[j] for j in sessions_indexes]
        out = []
        for idx in subjects_indexes:
            basis = safe_load(self.basis_list[idx])
            recon = [basis.T @ s for s in sessions]
            out.append(recon[0] if single_session else recon)
        return out[0] if len(out) == 1 and single_session else out

    def add_subjects(self, imgs, shared_response):
        """Compute bases for new subjects against an existing shared
        response and append them to basis_list."""
        subjects, _ = self._canonicalize(imgs)
        single = isinstance(shared_response, np.ndarray)
        sessions_s = [shared_response] if single else list(shared_response)
        # accept [k, T] or [T, k]
        sessions_s = [s if s.shape[0] != self.n_components else s.T
                      for s in sessions_s]
        start = len(self.basis_list) if self.basis_list else 0
        if self.basis_list is None:
            self.basis_list = []
        for i, subj in enumerate(subjects):
            corr = None
            for j, sess in enumerate(subj):
                data = safe_load(sess)
                c = sessions_s[j].T @ data.T
                corr = c if corr is None else corr + c
            basis_i = _compute_subject_basis(corr)
            if self.temp_dir is None:
                self.basis_list.append(basis_i)
            else:
                os.makedirs(self.temp_dir, exist_ok=True)
                path = os.path.join(self.temp_dir,
                                    "basis_%i.npy" % (start + i))
                np.save(path, basis_i)
                self.basis_list.append(path)
        return self

    def get_params(self, deep=True):
        return {"atlas": self.atlas, "n_components": self.n_components,
                "n_iter": self.n_iter, "low_ram": self.low_ram,
                "seed": self.seed, "n_jobs": self.n_jobs,
                "aggregate": self.aggregate}

    def set_params(self, **params):
        for k, v in params.items():
            setattr(self, k, v)
        return self
