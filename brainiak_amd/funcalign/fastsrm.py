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

__all__ = ["FastSRM", "assert_array_2axis", "assert_non_empty_list",
           "assert_valid_index", "check_atlas", "check_imgs",
           "check_indexes", "check_shared_response", "create_temp_dir",
           "fast_srm", "get_shape", "lowram_srm", "reduce_data",
           "reduce_data_single", "safe_encode", "safe_load"]


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

    def _validate(self, subjects):
        """Input-consistency checks (the reference's checker suite,
        condensed to the load-bearing assertions: ref
        fastsrm.py:69-576)."""
        if len(subjects) == 0:
            raise ValueError("imgs is empty")
        shapes = [[np.asarray(safe_load(sess)).shape for sess in subj]
                  for subj in subjects]
        v0 = shapes[0][0][0]
        for i, subj in enumerate(shapes):
            if len(subj) != len(shapes[0]):
                raise ValueError(
                    "Subject %d has %d sessions but subject 0 has %d"
                    % (i, len(subj), len(shapes[0])))
            for j, sh in enumerate(subj):
                if len(sh) != 2:
                    raise ValueError(
                        "imgs[%d][%d] must be 2-D, got shape %s"
                        % (i, j, (sh,)))
                if sh[0] != v0:
                    raise ValueError(
                        "Subject %d session %d has %d voxels; expected "
                        "%d" % (i, j, sh[0], v0))
                if sh[1] != shapes[0][j][1]:
                    raise ValueError(
                        "Session %d timeframe counts differ across "
                        "subjects (%d vs %d)" % (j, sh[1],
                                                 shapes[0][j][1]))
        if self.atlas is not None:
            atlas = safe_load(self.atlas)
            n_regions = (atlas.shape[0] if atlas.ndim == 2
                         else int(atlas.max()))
            n_vox_atlas = (atlas.shape[1] if atlas.ndim == 2
                           else atlas.shape[0])
            if atlas.ndim > 2:
                raise ValueError("Atlas has %d axes; expected 1 or 2"
                                 % atlas.ndim)
            if n_regions < self.n_components:
                raise ValueError(
                    "Number of regions in the atlas is lower than the "
                    "number of components (%d < %d)"
                    % (n_regions, self.n_components))
            if n_vox_atlas != v0:
                raise ValueError(
                    "Atlas covers %d voxels but data has %d"
                    % (n_vox_atlas, v0))

    def fit(self, imgs):
        """Fit the shared response and per-subject bases."""
        subjects, _ = self._canonicalize(imgs)
        self._validate(subjects)
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
        if sessions_s and isinstance(sessions_s[0], (list, tuple)):
            # aggregate=None responses: [subjects][sessions] → average
            # across subjects per session (the reference's
            # check_shared_response collapse, ref fastsrm.py:365-393)
            n_sess = len(sessions_s[0])
            sessions_s = [np.mean([np.asarray(subj[j])
                                   for subj in sessions_s], axis=0)
                          for j in range(n_sess)]
        else:
            sessions_s = [np.asarray(s) for s in sessions_s]
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


# -- public module-level helpers (reference fastsrm.py:32-1073 API) --------

def get_shape(path):
    """Shape of a saved ``.npy`` array without loading it."""
    return np.load(path, mmap_mode='r').shape


def safe_encode(img):
    """Stable hash name for an image (array contents or path string)."""
    import hashlib
    if isinstance(img, np.ndarray):
        return hashlib.md5(img.tobytes()).hexdigest()
    return hashlib.md5(str(img).encode()).hexdigest()


def assert_non_empty_list(input_list, list_name):
    if len(input_list) == 0:
        raise ValueError("%s is a list of length 0 which is not valid"
                         % list_name)


def assert_array_2axis(array, name_array):
    if not isinstance(array, np.ndarray):
        raise ValueError("%s should be of type np.ndarray but is of "
                         "type %s" % (name_array, type(array)))
    if array.ndim != 2:
        raise ValueError("%s must have exactly 2 axes but has %i axes"
                         % (name_array, array.ndim))


def assert_valid_index(indexes, max_value, name_indexes):
    for i, ind in enumerate(indexes):
        if ind < 0 or ind >= max_value:
            raise ValueError(
                "Index %i of %s has value %i whereas value should be "
                "between 0 and %i" % (i, name_indexes, ind,
                                      max_value - 1))


def check_indexes(indexes, name):
    if not (indexes is None or isinstance(indexes, (list, np.ndarray))):
        raise ValueError("%s should be either a list, an array or None "
                         "but received type %s" % (name, type(indexes)))


def check_atlas(atlas, n_components=None):
    """Validate a probabilistic/deterministic atlas (array or path);
    returns its shape, or None for no atlas."""
    if atlas is None:
        return None
    if isinstance(atlas, (str, os.PathLike)):
        shape = get_shape(atlas)
    elif isinstance(atlas, np.ndarray):
        shape = atlas.shape
    else:
        raise ValueError(
            "Atlas is stored using type %s which is neither np.ndarray "
            "or str" % type(atlas))
    if len(shape) == 1:          # deterministic label atlas
        labels = safe_load(atlas)
        n_sup = int(np.max(labels))
    elif len(shape) == 2:        # probabilistic
        n_sup = shape[0]
    else:
        raise ValueError("Atlas has %i axes; it should have either 1 "
                         "or 2 axes" % len(shape))
    if n_components is not None and n_sup < n_components:
        raise ValueError(
            "Number of atlas supervoxels (%i) is less than the number "
            "of components (%i)" % (n_sup, n_components))
    return shape


def check_imgs(imgs, n_components=None, atlas_shape=None,
               ignore_nans=False):
    """Validate input images (array of paths / list of arrays / list of
    lists); returns (reshaped [subject][session] list, n_subjects,
    n_sessions)."""
    subjects, _ = FastSRM._canonicalize(imgs)
    n_subjects = len(subjects)
    n_sessions = len(subjects[0])
    for s in subjects:
        if len(s) != n_sessions:
            raise ValueError("All subjects need the same number of "
                             "sessions")
    v0 = None
    for s in subjects:
        for img in s:
            arr = safe_load(img)
            assert_array_2axis(arr, "imgs element")
            if v0 is None:
                v0 = arr.shape[0]
            elif arr.shape[0] != v0:
                raise ValueError("All images must have the same number "
                                 "of voxels")
    return subjects, n_subjects, n_sessions


def check_shared_response(shared_response, aggregate="mean",
                          n_components=None, input_format=None,
                          n_timeframes=None):
    """Validate a shared response (array, list of sessions, or list of
    lists subject x session); returns the list-of-sessions form."""
    if isinstance(shared_response, np.ndarray):
        assert_array_2axis(shared_response, "shared_response")
        out = [shared_response]
    elif isinstance(shared_response, list):
        assert_non_empty_list(shared_response, "shared_response")
        if isinstance(shared_response[0], list):
            # subject x session: aggregate by mean over subjects
            n_sess = len(shared_response[0])
            out = [np.mean([subj[j] for subj in shared_response], axis=0)
                   for j in range(n_sess)]
        else:
            for s in shared_response:
                assert_array_2axis(s, "shared_response session")
            out = list(shared_response)
    else:
        raise ValueError("shared_response should be an array or a list")
    if n_components is not None:
        for s in out:
            if n_components not in s.shape:
                raise ValueError("shared response does not match "
                                 "n_components=%i" % n_components)
    return out


def create_temp_dir(temp_dir):
    """Create ``temp_dir`` if needed; error if it already exists (use
    ``.clean()`` between runs)."""
    if temp_dir is None:
        return None
    if not os.path.exists(temp_dir):
        os.makedirs(temp_dir)
    else:
        raise ValueError(
            "Path %s already exists. When a model is used, filesystem "
            "should be cleaned by using the .clean() method" % temp_dir)


def reduce_data_single(subject_index, session_index, img, atlas=None,
                       inv_atlas=None, low_ram=False, temp_dir=None):
    """Atlas-project one [V, T] image -> [T, n_supervoxels] (array, or
    path when ``low_ram``)."""
    data = safe_load(img)
    if inv_atlas is not None:
        reduced = data.T @ inv_atlas
    elif atlas is not None:
        labels = safe_load(atlas).astype(int)
        n_parcels = int(labels.max())
        reduced = np.zeros((data.shape[1], n_parcels))
        for p in range(1, n_parcels + 1):
            m = labels == p
            if m.any():
                reduced[:, p - 1] = data[m].mean(axis=0)
    else:
        reduced = data.T.copy()
    if low_ram and temp_dir is not None:
        path = os.path.join(
            temp_dir, "reduced_data_%i_%i.npy"
            % (subject_index, session_index))
        np.save(path, reduced)
        return path
    return reduced


def reduce_data(imgs, atlas, n_jobs=1, low_ram=False, temp_dir=None):
    """Atlas-project all images -> [n_subjects][n_sessions] reduced
    data ([T, n_supervoxels] arrays or paths when ``low_ram``)."""
    subjects, n_subjects, n_sessions = check_imgs(imgs)
    inv_atlas = None
    label_atlas = None
    if atlas is not None:
        a = safe_load(atlas)
        if a.ndim == 2:
            inv_atlas = np.linalg.pinv(a)
        else:
            label_atlas = a
    return [
        [reduce_data_single(i, j, subjects[i][j], atlas=label_atlas,
                            inv_atlas=inv_atlas, low_ram=low_ram,
                            temp_dir=temp_dir)
         for j in range(n_sessions)]
        for i in range(n_subjects)]


def fast_srm(reduced_data_list, n_iter=10, n_components=None,
             low_ram=False, seed=0):
    """Shared response in reduced space (list of [k, T] per session)
    from [n_subjects][n_sessions] reduced data."""
    if isinstance(reduced_data_list, np.ndarray) and \
            reduced_data_list.ndim == 3:
        reduced_data_list = [[r] for r in reduced_data_list]
    n_subjects = len(reduced_data_list)
    n_sessions = len(reduced_data_list[0])
    # concatenate sessions in time, run DetSRM in the reduced space
    data = [np.concatenate(
        [safe_load(reduced_data_list[i][j]) for j in range(n_sessions)],
        axis=0).T for i in range(n_subjects)]    # [n_sup, T_total]
    srm = DetSRM(n_iter=n_iter, features=n_components, rand_seed=seed)
    srm.fit(data)
    s = srm.s_                                   # [k, T_total]
    bounds = np.cumsum(
        [0] + [safe_load(reduced_data_list[0][j]).shape[0]
               for j in range(n_sessions)])
    return [s[:, bounds[j]:bounds[j + 1]] for j in range(n_sessions)]


def lowram_srm(reduced_data_list, n_iter=10, n_components=None):
    """Memory-lean variant of ``fast_srm`` (same computation here —
    288 GB HBM / host RAM makes the reference's disk-streamed variant
    unnecessary, the API is kept for parity)."""
    return fast_srm(reduced_data_list, n_iter=n_iter,
                    n_components=n_components)
