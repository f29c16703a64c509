"""Topographical Factor Analysis (single subject).

API parity with the reference TFA (ref src/brainiak/factoranalysis/
tfa.py:52-1024): same constructor, the packed prior/posterior vector
layout (centers | widths | centers_mean_cov | widths_mean_var with
``map_offset``), K-means template init, alternating ridge weights /
nonlinear-least-squares center+width estimation with voxel/TR
subsampling, and Hungarian prior↔posterior matching.

The two hot native ops (ref tfa_extension.cpp:28-239: the RBF factor
matrix and the scaled reconstruction residual) are
``brainiak_amd.ops.tfa_factor`` / ``tfa_recon`` HIP kernels on GPU and
vectorized numpy on CPU — the separable unique-coordinate trick the
reference uses is an x86 cache optimization that a gfx950 kernel does
not need (the direct distance form is bandwidth-trivial).
"""

import logging
import math
import os

import numpy as np
import torch
from scipy.optimize import least_squares, linear_sum_assignment
from scipy.spatial import distance

from .. import ops
from ..utils.utils import from_sym_2_tri, from_tri_2_sym

logger = logging.getLogger(__name__)

__all__ = ["TFA"]


class TFA:
    """Single-subject TFA: X ≈ F(centers, widths) · W with RBF factors.

    Constructor parameters identical to the reference (max_iter,
    threshold, K, nlss_method, nlss_loss, jac, x_scale, tr_solver,
    weight_method ('rr'|'ols'), upper_ratio, lower_ratio, max_num_tr,
    max_num_voxel, seed, verbose) plus ``device`` for the factor/recon
    kernels.
    """

    def __init__(self, max_iter=10, threshold=1.0, K=50,
                 nlss_method='trf', nlss_loss='soft_l1', jac='analytic',
                 x_scale='jac', tr_solver=None, weight_method='rr',
                 upper_ratio=1.8, lower_ratio=0.02, max_num_tr=500,
                 max_num_voxel=5000, seed=100, verbose=False, device=None,
                 nlss_max_nfev=10):
        self.miter = max_iter
        self.threshold = threshold
        self.K = K
        self.nlss_method = nlss_method
        self.nlss_loss = nlss_loss
        self.jac = jac
        self.nlss_max_nfev = nlss_max_nfev
        self.x_scale = x_scale
        self.tr_solver = tr_solver
        self.weight_method = weight_method
        self.upper_ratio = upper_ratio
        self.lower_ratio = lower_ratio
        self.max_num_tr = max_num_tr
        self.max_num_voxel = max_num_voxel
        self.seed = seed
        self.verbose = verbose
        self.device = device

    # -- small setters (reference API) -------------------------------------

    def set_K(self, K):
        self.K = K
        return self

    def set_prior(self, prior):
        self.local_prior = prior
        return self

    def set_seed(self, seed):
        self.seed = seed
        return self

    def init_prior(self, R):
        centers, widths = self.init_centers_widths(R)
        prior = np.zeros(self.K * (self.n_dim + 1))
        self.set_centers(prior, centers)
        self.set_widths(prior, widths)
        self.set_prior(prior)
        return self

    # -- packed-vector accessors -------------------------------------------

    def get_map_offset(self):
        nfield = 4
        self.map_offset = np.zeros(nfield).astype(int)
        field_size = self.K * np.array(
            [self.n_dim, 1, self.cov_vec_size, 1])
        for i in np.arange(nfield - 1) + 1:
            self.map_offset[i] = self.map_offset[i - 1] + field_size[i - 1]
        return self.map_offset

    def set_centers(self, estimation, centers):
        estimation[0:self.map_offset[1]] = centers.ravel()

    def set_widths(self, estimation, widths):
        estimation[self.map_offset[1]:self.map_offset[2]] = widths.ravel()

    def set_centers_mean_cov(self, estimation, centers_mean_cov):
        estimation[self.map_offset[2]:self.map_offset[3]] = \
            centers_mean_cov.ravel()

    def set_widths_mean_var(self, estimation, widths_mean_var):
        estimation[self.map_offset[3]:] = widths_mean_var.ravel()

    def get_centers(self, estimation):
        return estimation[0:self.map_offset[1]].reshape(self.K, self.n_dim)

    def get_widths(self, estimation):
        return estimation[self.map_offset[1]:self.map_offset[2]].reshape(
            self.K, 1)

    def get_centers_mean_cov(self, estimation):
        return estimation[self.map_offset[2]:self.map_offset[3]].reshape(
            self.K, self.cov_vec_size)

    def get_widths_mean_var(self, estimation):
        return estimation[self.map_offset[3]:].reshape(self.K, 1)

    # -- initialization -----------------------------------------------------

    def init_centers_widths(self, R):
        """K-means centers + max-sigma widths.

        The clustering only SEEDS the NLSS (centers are refined every
        iteration), so it runs single-init on a <=8k-voxel subsample —
        full-brain 10-restart Lloyd was 32 % of an entire HTFA fit
        (profiles/htfa_cprofile.txt) for identical end fits.  On GPU
        the Lloyd loop itself runs in torch (sklearn's CPU kmeans was
        still ~40 % of a 100k-voxel HTFA global iteration)."""
        pts = R
        if R.shape[0] > 8000:
            sel = np.random.RandomState(100).choice(
                R.shape[0], 8000, replace=False)
            pts = R[sel]
        if self._use_gpu():
            centers = self._kmeans_torch(pts, self.K, seed=100)
        else:
            from sklearn.cluster import KMeans
            kmeans = KMeans(init='k-means++', n_clusters=self.K,
                            n_init=1, max_iter=50, random_state=100)
            kmeans.fit(pts)
            centers = kmeans.cluster_centers_
        widths = self._get_max_sigma(R) * np.ones((self.K, 1))
        return centers, widths

    @staticmethod
    def _kmeans_torch(pts, K, seed=100, iters=25):
        """Seeding-quality Lloyd on device: k-means++ init (numpy,
        deterministic) then fixed-iteration assignments in torch."""
        rng = np.random.RandomState(seed)
        P = np.asarray(pts, dtype=np.float64)
        # k-means++ seeding
        centers = [P[rng.randint(len(P))]]
        d2 = ((P - centers[0]) ** 2).sum(1)
        for _ in range(K - 1):
            probs = d2 / d2.sum()
            centers.append(P[rng.choice(len(P), p=probs)])
            d2 = np.minimum(d2, ((P - centers[-1]) ** 2).sum(1))
        C = torch.as_tensor(np.asarray(centers), dtype=torch.float32,
                            device="cuda")
        X = torch.as_tensor(P, dtype=torch.float32, device="cuda")
        x_sq = (X * X).sum(1, keepdim=True)
        for _ in range(iters):
            # expanded-norm distances: torch.cdist measured 5 ms/call
            # at [8k, 3] x [K, 3] vs ~50 us for the GEMM form
            d2 = x_sq + (C * C).sum(1)[None, :] - 2.0 * (X @ C.T)
            assign = d2.argmin(dim=1)
            one_hot = torch.zeros((X.shape[0], K), device="cuda")
            one_hot.scatter_(1, assign[:, None], 1.0)
            counts = one_hot.sum(0).clamp_min(1.0)
            C = (one_hot.T @ X) / counts[:, None]
        return C.double().cpu().numpy()

    def get_template(self, R):
        """Template prior (centers | widths | centers cov | widths var)."""
        centers, widths = self.init_centers_widths(R)
        template_prior = np.zeros(
            self.K * (self.n_dim + 2 + self.cov_vec_size))
        template_centers_cov = np.cov(R.T) * math.pow(self.K, -2 / 3.0)
        template_widths_var = self._get_max_sigma(R)
        centers_cov_all = np.tile(from_sym_2_tri(template_centers_cov),
                                  self.K)
        widths_var_all = np.tile(template_widths_var, self.K)
        self.set_centers(template_prior, centers)
        self.set_widths(template_prior, widths)
        self.set_centers_mean_cov(template_prior, centers_cov_all)
        self.set_widths_mean_var(template_prior, widths_var_all)
        return template_prior, template_centers_cov, template_widths_var

    def _get_max_sigma(self, R):
        return 2.0 * math.pow(np.nanmax(np.std(R, axis=0)), 2)

    def get_bounds(self, R):
        max_sigma = self._get_max_sigma(R)
        lower = np.zeros(self.K * (self.n_dim + 1))
        lower[0:self.K * self.n_dim] = np.tile(np.nanmin(R, axis=0),
                                               self.K)
        lower[self.K * self.n_dim:] = np.repeat(
            self.lower_ratio * max_sigma, self.K)
        upper = np.zeros(self.K * (self.n_dim + 1))
        upper[0:self.K * self.n_dim] = np.tile(np.nanmax(R, axis=0),
                                               self.K)
        upper[self.K * self.n_dim:] = np.repeat(
            self.upper_ratio * max_sigma, self.K)
        return (lower, upper)

    # -- compute core --------------------------------------------------------

    def get_unique_R(self, R):
        """Per-dimension unique coordinate tables (reference API)."""
        unique_R, inds = [], []
        for d in np.arange(self.n_dim):
            u, i = np.unique(R[:, d], return_inverse=True)
            unique_R.append(u)
            inds.append(i)
        return unique_R, inds

    def _use_gpu(self):
        if self.device is not None:
            return torch.device(self.device).type == "cuda"
        return torch.cuda.is_available()

    def get_factors(self, unique_R, inds, centers, widths):
        """RBF factor matrix F [n_voxel, K] (N8 equivalent)."""
        coords = np.column_stack(
            [unique_R[d][inds[d]] for d in range(self.n_dim)]).astype(
                np.float64)
        if self._use_gpu() and self.n_dim == 3:
            return self._get_factors_dev(coords, centers,
                                         widths).double().cpu().numpy()
        d2 = distance.cdist(coords, centers, 'sqeuclidean')
        return np.exp(-d2 / widths.ravel()[None, :])

    @staticmethod
    def _to_dev_f32(a):
        """numpy -> fp32 cuda with the dtype cast on DEVICE: as_tensor
        with dtype=float32 converts fp64 arrays on the host first
        (single-threaded, and doubles the work before an 80 MB/subject
        upload — measured as the top cost of an HTFA step)."""
        t = torch.from_numpy(np.ascontiguousarray(a))
        return t.to("cuda", non_blocking=False).float()

    def _get_factors_dev(self, coords, centers, widths):
        """Device-resident factor matrix (fp32 cuda) — lets callers
        chain into the weight solve without a host round trip."""
        return ops.tfa_factor(
            self._to_dev_f32(np.asarray(centers)),
            self._to_dev_f32(np.asarray(widths).ravel()),
            self._to_dev_f32(coords))

    def get_weights(self, data, F):
        """Ridge ('rr') or OLS weights W [K, n_tr] (device GEMMs when
        a GPU is available — the [V, K] normal-equation products are
        the only O(V) work here)."""
        if self._use_gpu():
            Ft = self._to_dev_f32(F)
            Xt = self._to_dev_f32(data)
            return self._get_weights_dev(Xt, Ft)
        beta = np.var(data)
        trans_F = F.T.copy()
        if self.weight_method == 'rr':
            W = np.linalg.solve(trans_F.dot(F)
                                + beta * np.identity(self.K),
                                trans_F.dot(data))
        else:
            W = np.linalg.solve(trans_F.dot(F), trans_F.dot(data))
        return W

    def _get_weights_dev(self, Xt, Ft):
        """Ridge/OLS weight solve: the O(V) normal-equation GEMMs run
        on device; the [K, K] solve runs on the HOST in fp64 (at
        K~20-50 a hipSOLVER call is ~1.2 ms of launch/sync overhead vs
        ~50 us in numpy, and every caller downloads W anyway).  The
        ridge beta (= var of the data, ddof 0, matching np.var) comes
        from the already-uploaded tensor — the CPU np.var over the
        full [V, T] array was half of HTFA's _update_weight wall.
        Returns W as a float64 numpy array."""
        G = Ft.T @ Ft
        B = Ft.T @ Xt
        if self.weight_method == 'rr':
            beta = Xt.double().var(correction=0)
            G = G + beta.float() * torch.eye(self.K, device=Ft.device)
        G_h = G.cpu().numpy().astype(np.float64)
        B_h = B.cpu().numpy().astype(np.float64)
        return np.linalg.solve(G_h, B_h)

    def _recon_err(self, X, F, W, data_sigma):
        """Scaled flattened residual data_sigma*(X - F·W) (N9)."""
        if self._use_gpu():
            return ops.tfa_recon(
                torch.as_tensor(X, dtype=torch.float32, device="cuda"),
                torch.as_tensor(W, dtype=torch.float32, device="cuda"),
                torch.as_tensor(F, dtype=torch.float32, device="cuda"),
                float(data_sigma)).double().cpu().numpy()
        return (data_sigma * (X - F.dot(W))).ravel()

    def _residual_multivariate(self, estimate, unique_R, inds, X, W,
                               template_centers,
                               template_centers_mean_cov,
                               template_widths,
                               template_widths_mean_var_reci, data_sigma):
        centers = self.get_centers(estimate)
        widths = self.get_widths(estimate)
        recon = X.size
        other_err = 0 if template_centers is None else (2 * self.K)
        final_err = np.zeros(recon + other_err)
        F = self.get_factors(unique_R, inds, centers, widths)
        final_err[0:recon] = self._recon_err(X, F, W, data_sigma)

        if other_err > 0:
            for k in np.arange(self.K):
                diff = centers[k] - template_centers[k]
                cov = from_tri_2_sym(template_centers_mean_cov[k],
                                     self.n_dim)
                cov = cov + cov.T - np.diag(np.diag(cov))
                final_err[recon + k] = math.sqrt(
                    self.sample_scaling
                    * diff.dot(np.linalg.solve(cov, diff.T)))
            base = recon + self.K
            dist = template_widths_mean_var_reci * \
                (widths - template_widths) ** 2
            final_err[base:] = np.sqrt(self.sample_scaling * dist).ravel()
        return final_err

    def _jacobian_multivariate(self, estimate, unique_R, inds, X, W,
                               template_centers,
                               template_centers_mean_cov,
                               template_widths,
                               template_widths_mean_var_reci,
                               data_sigma):
        """Closed-form Jacobian of ``_residual_multivariate``.

        The reference differentiates by finite differences (~K*(dim+1)+1
        residual evaluations per NLSS iteration, each an N8+N9 kernel
        pass); the RBF residual is analytic in centers and widths:

            r[v,t]        = sigma*(X[v,t] - sum_k F[v,k] W[k,t])
            dF/dc_{k,d}   = F[v,k] * 2 (coord[v,d] - c[k,d]) / w_k
            dF/dw_k       = F[v,k] * d2[v,k] / w_k**2

        so ONE factor evaluation yields the whole [m, K*(dim+1)] matrix.
        Parameter order matches ``estimate``: centers.ravel() then
        widths.
        """
        centers = self.get_centers(estimate)
        widths = self.get_widths(estimate)
        coords = np.column_stack(
            [unique_R[d][inds[d]] for d in range(self.n_dim)]).astype(
                np.float64)
        K, D = self.K, self.n_dim
        V, T = X.shape
        n_par = K * (D + 1)
        dev = "cuda" if self._use_gpu() else "cpu"
        co = torch.as_tensor(coords, dtype=torch.float32, device=dev)
        ce = torch.as_tensor(centers, dtype=torch.float32, device=dev)
        wd = torch.as_tensor(widths.ravel(), dtype=torch.float32,
                             device=dev)
        Wt = torch.as_tensor(W, dtype=torch.float32, device=dev)
        diff = co[:, None, :] - ce[None, :, :]            # [V, K, D]
        d2 = (diff * diff).sum(-1)                        # [V, K]
        F = torch.exp(-d2 / wd[None, :])
        dF = torch.empty((V, K, D + 1), dtype=torch.float32, device=dev)
        dF[:, :, :D] = F[:, :, None] * 2.0 * diff / wd[None, :, None]
        dF[:, :, D] = F * d2 / (wd * wd)[None, :]
        # J_recon[v,t, k,j] = -sigma * W[k,t] * dF[v,k,j]
        Jr = -float(data_sigma) * torch.einsum('kt,vkj->vtkj', Wt, dF)
        recon = V * T
        other = 0 if template_centers is None else 2 * K
        # fp32 J: the trust-region solves live in an 80-dim subspace
        # where single precision is ample (FD was far noisier), and
        # scipy's per-iteration SVD of [m, n_par] runs 3.5x faster on
        # fp32 (measured; quality identical to the fp64 J)
        J = np.zeros((recon + other, n_par), dtype=np.float32)
        Jrc = Jr.reshape(recon, K, D + 1).cpu().numpy()
        # column order: centers (k*D + d), then widths (K*D + k)
        J[:recon, :K * D] = Jrc[:, :, :D].reshape(recon, K * D)
        J[:recon, K * D:] = Jrc[:, :, D]
        if other > 0:
            S = self.sample_scaling
            for k in range(K):
                dfk = centers[k] - template_centers[k]
                cov = from_tri_2_sym(template_centers_mean_cov[k],
                                     self.n_dim)
                cov = cov + cov.T - np.diag(np.diag(cov))
                solved = np.linalg.solve(cov, dfk)
                e = math.sqrt(max(S * dfk.dot(solved), 1e-30))
                J[recon + k, k * D:(k + 1) * D] = S * solved / e
            reci = np.asarray(template_widths_mean_var_reci).ravel()
            tw = np.asarray(template_widths).ravel()
            w = widths.ravel()
            J[recon + K:, K * D:][np.arange(K), np.arange(K)] = \
                np.sqrt(S * reci) * np.sign(w - tw)
        return J

    def _estimate_centers_widths_torch_lm(
            self, unique_R, inds, X, W, init_estimate, data_sigma,
            template_centers, template_widths,
            template_centers_mean_cov, template_widths_mean_var_reci):
        """Device-resident bounded Levenberg-Marquardt with a soft_l1
        IRLS reweighting.

        The scipy trf solver spends its time in host-side SVD/QR of
        the [V*T, K*(dim+1)] Jacobian — 87 % of an HTFA fit's wall
        time (profiles/htfa_profile.json).  The normal-equation form
        J^T J (an [n_par, n_par] product) is a single GEMM on device,
        so the whole solve stays on the GPU; bounds are enforced by
        projection (clamp) as in scipy's 'tr' reflective strategy's
        simple limit.
        """
        dev = "cuda"
        K, D = self.K, self.n_dim
        n_par = K * (D + 1)
        coords = np.column_stack(
            [unique_R[d][inds[d]] for d in range(D)]).astype(np.float32)
        co = torch.as_tensor(coords, device=dev)
        Xt = self._to_dev_f32(X)
        Wt = self._to_dev_f32(W)
        lb = torch.as_tensor(self.bounds[0], dtype=torch.float32,
                             device=dev)
        ub = torch.as_tensor(self.bounds[1], dtype=torch.float32,
                             device=dev)
        sigma = float(data_sigma)
        have_prior = template_centers is not None
        if have_prior:
            tc = torch.as_tensor(np.asarray(template_centers,
                                            dtype=np.float32), device=dev)
            tw = torch.as_tensor(np.asarray(template_widths,
                                            dtype=np.float32).ravel(),
                                 device=dev)
            reci = torch.as_tensor(
                np.asarray(template_widths_mean_var_reci,
                           dtype=np.float32).ravel(), device=dev)
            cov_inv = []
            for k in range(K):
                cov = from_tri_2_sym(template_centers_mean_cov[k], D)
                cov = cov + cov.T - np.diag(np.diag(cov))
                cov_inv.append(np.linalg.inv(cov))
            cov_inv = torch.as_tensor(np.asarray(cov_inv,
                                                 dtype=np.float32),
                                      device=dev)       # [K, D, D]
            S = float(self.sample_scaling)

        def unpack(theta):
            return theta[:K * D].reshape(K, D), theta[K * D:]

        def residual_and_jac(theta, want_jac=True):
            ce, wd = unpack(theta)
            diff = co[:, None, :] - ce[None, :, :]        # [V, K, D]
            d2 = (diff * diff).sum(-1)
            F = torch.exp(-d2 / wd[None, :])
            R = sigma * (Xt - F @ Wt)                     # [V, T]
            parts = [R.reshape(-1)]
            Js = None
            if want_jac:
                dF = torch.empty((co.shape[0], K, D + 1),
                                 dtype=torch.float32, device=dev)
                dF[:, :, :D] = F[:, :, None] * 2.0 * diff                     / wd[None, :, None]
                dF[:, :, D] = F * d2 / (wd * wd)[None, :]
                Jr = -sigma * torch.einsum('kt,vkj->vtkj', Wt, dF)
                Jr = Jr.reshape(-1, K, D + 1)
                Jcols = torch.cat([Jr[:, :, :D].reshape(-1, K * D),
                                   Jr[:, :, D]], dim=1)  # [VT, n_par]
                Js = [Jcols]
            if have_prior:
                dfc = ce - tc                             # [K, D]
                solved = torch.einsum('kij,kj->ki', cov_inv, dfc)
                qc = (S * (dfc * solved).sum(-1)).clamp_min(1e-30)
                rc = qc.sqrt()                            # [K]
                rw_sq = (S * reci * (wd - tw) ** 2).clamp_min(0.0)
                rw = rw_sq.sqrt()
                parts += [rc, rw]
                if want_jac:
                    Jp = torch.zeros((2 * K, n_par),
                                     dtype=torch.float32, device=dev)
                    grad_c = S * solved / rc[:, None]     # [K, D]
                    for k in range(K):
                        Jp[k, k * D:(k + 1) * D] = grad_c[k]
                    Jp[K + torch.arange(K, device=dev),
                       K * D + torch.arange(K, device=dev)] =                         torch.sqrt(S * reci) * torch.sign(wd - tw)
                    Js.append(Jp)
            r = torch.cat(parts)
            J = torch.cat(Js, dim=0) if want_jac else None
            return r, J

        def soft_l1_cost(r):
            z = r * r
            return float(2.0 * ((1 + z).sqrt() - 1).sum())

        theta = torch.as_tensor(np.asarray(init_estimate,
                                           dtype=np.float32),
                                device=dev).clamp(lb, ub)
        lam = 1e-3
        max_nfev = self.nlss_max_nfev or 20
        r, J = residual_and_jac(theta)
        cost = soft_l1_cost(r)
        for _ in range(int(max_nfev)):
            # soft_l1 IRLS weights: sqrt(rho'(z)) scaling
            wgt = (1 + r * r).pow(-0.25)
            rw_ = r * wgt
            Jw = J * wgt[:, None]
            A = Jw.T @ Jw                                 # [n, n] GEMM
            g = Jw.T @ rw_
            # the [n_par, n_par] damped solve runs on the HOST: n_par
            # is ~80, and a hipSOLVER call at that size costs ~1.2 ms
            # of launch/sync overhead per damping try (1500+ tries per
            # HTFA fit) vs ~50 us in fp64 numpy after a 26 KB download
            A_h = A.cpu().numpy().astype(np.float64)
            g_h = g.cpu().numpy().astype(np.float64)
            dA_h = np.clip(np.diag(A_h), 1e-12, None)
            improved = False
            for _try in range(6):
                damp = A_h + lam * np.diag(dA_h)
                try:
                    delta_h = np.linalg.solve(damp, -g_h)
                except np.linalg.LinAlgError:
                    lam *= 10
                    continue
                delta = torch.as_tensor(delta_h, dtype=torch.float32,
                                        device=dev)
                cand = (theta + delta).clamp(lb, ub)
                rc_, _ = residual_and_jac(cand, want_jac=False)
                c2 = soft_l1_cost(rc_)
                if c2 < cost:
                    theta, cost = cand, c2
                    lam = max(lam / 3, 1e-8)
                    improved = True
                    break
                lam *= 4
            if not improved:
                break
            r, J = residual_and_jac(theta)
        return theta.cpu().numpy().astype(np.float64), 0.5 * cost

    def _estimate_centers_widths(self, unique_R, inds, X, W, init_centers,
                                 init_widths, template_centers,
                                 template_widths,
                                 template_centers_mean_cov,
                                 template_widths_mean_var_reci):
        init_estimate = np.hstack((init_centers.ravel(),
                                   init_widths.ravel()))
        data_sigma = 1.0 / math.sqrt(2.0) * np.std(X)
        if self._use_gpu() and self.n_dim == 3 and \
                not os.environ.get("BRAINIAK_TFA_SCIPY"):
            return self._estimate_centers_widths_torch_lm(
                unique_R, inds, X, W, init_estimate, data_sigma,
                template_centers, template_widths,
                template_centers_mean_cov,
                template_widths_mean_var_reci)
        # 'analytic' (default): the closed-form Jacobian — one factor
        # pass instead of ~K*(dim+1) FD residual evaluations.  FD
        # schemes remain selectable via jac='2-point'/'3-point'.
        # nlss_max_nfev bounds the inner trf iterations: the exact
        # Jacobian keeps finding descent long past the point where the
        # outer TFA alternation makes the refinement moot.
        jac = self._jacobian_multivariate if self.jac == 'analytic' \
            else self.jac
        final_estimate = least_squares(
            self._residual_multivariate, init_estimate,
            args=(unique_R, inds, X, W, template_centers,
                  template_centers_mean_cov, template_widths,
                  template_widths_mean_var_reci, data_sigma),
            method=self.nlss_method, loss=self.nlss_loss,
            bounds=self.bounds, verbose=0, x_scale=self.x_scale,
            tr_solver=self.tr_solver, jac=jac,
            max_nfev=self.nlss_max_nfev)
        return final_estimate.x, final_estimate.cost

    # -- convergence ---------------------------------------------------------

    def _assign_posterior(self):
        """Hungarian match of posterior factors onto the prior ordering."""
        prior_centers = self.get_centers(self.local_prior)
        posterior_centers = self.get_centers(self.local_posterior_)
        posterior_widths = self.get_widths(self.local_posterior_)
        cost = distance.cdist(prior_centers, posterior_centers, 'euclidean')
        _, col_ind = linear_sum_assignment(cost)
        self.set_centers(self.local_posterior_, posterior_centers[col_ind])
        self.set_widths(self.local_posterior_, posterior_widths[col_ind])
        return self

    def _converged(self):
        diff = self.local_prior - self.local_posterior_
        max_diff = np.max(np.fabs(diff))
        if self.verbose:
            _, mse = self._mse_converged()
            diff_ratio = np.sum(diff ** 2) / np.sum(
                self.local_posterior_ ** 2)
            logger.info('tfa prior posterior max diff %f mse %f '
                        'diff_ratio %f', max_diff, mse, diff_ratio)
        return (max_diff <= self.threshold), max_diff

    def _mse_converged(self):
        mse = np.mean((self.local_prior - self.local_posterior_) ** 2)
        return (mse <= self.threshold), mse

    # -- fitting ---------------------------------------------------------------

    def _prepare_fit(self, X, R, template_prior):
        """Shared fit preamble (dims, offsets, bounds, subsample
        scaling, prior init) — also used by HTFA's batched local
        fit."""
        self.n_dim = R.shape[1]
        self.cov_vec_size = np.sum(np.arange(self.n_dim) + 1)
        self.map_offset = self.get_map_offset()
        # bounds depend only on R; HTFA re-enters here every global
        # iteration with the same coordinate matrix (np.std over the
        # full [V, 3] array per subject per iteration otherwise)
        cache = getattr(self, "_bounds_cache", None)
        if cache is not None and cache[0] is R:
            self.bounds = cache[1]
        else:
            self.bounds = self.get_bounds(R)
            self._bounds_cache = (R, self.bounds)
        n_voxel, n_tr = X.shape
        self.sample_scaling = 0.5 * float(
            min(self.max_num_voxel, n_voxel)
            * min(self.max_num_tr, n_tr)) / float(n_voxel * n_tr)
        if template_prior is None:
            self.init_prior(R)
        else:
            self.local_prior = template_prior[0:self.map_offset[2]]
        return self

    def _fit_tfa(self, data, R, template_prior=None):
        if template_prior is None:
            template_centers = None
            template_widths = None
            template_centers_mean_cov = None
            template_widths_mean_var_reci = None
        else:
            template_centers = self.get_centers(template_prior)
            template_widths = self.get_widths(template_prior)
            template_centers_mean_cov = self.get_centers_mean_cov(
                template_prior)
            template_widths_mean_var_reci = 1.0 / self.get_widths_mean_var(
                template_prior)
        inner_converged = False
        np.random.seed(self.seed)
        n = 0
        while n < self.miter and not inner_converged:
            self._fit_tfa_inner(data, R, template_centers,
                                template_widths,
                                template_centers_mean_cov,
                                template_widths_mean_var_reci)
            self._assign_posterior()
            inner_converged, _ = self._converged()
            if not inner_converged:
                self.local_prior = self.local_posterior_
            else:
                logger.info("TFA converged at %d iteration.", n)
            n += 1
        return self

    def _fit_tfa_inner(self, data, R, template_centers, template_widths,
                       template_centers_mean_cov,
                       template_widths_mean_var_reci):
        nfeature = data.shape[0]
        nsample = data.shape[1]
        n_vox = min(self.max_num_voxel, nfeature)
        n_tr = min(self.max_num_tr, nsample)
        feature_indices = np.random.choice(nfeature, n_vox, replace=False)
        samples_indices = np.random.choice(nsample, n_tr, replace=False)
        curr_data = data[feature_indices][:, samples_indices].copy()
        curr_R = R[feature_indices].copy()
        centers = self.get_centers(self.local_prior)
        widths = self.get_widths(self.local_prior)
        unique_R, inds = self.get_unique_R(curr_R)
        F = self.get_factors(unique_R, inds, centers, widths)
        W = self.get_weights(curr_data, F)
        self.local_posterior_, self.total_cost = \
            self._estimate_centers_widths(
                unique_R, inds, curr_data, W, centers, widths,
                template_centers, template_widths,
                template_centers_mean_cov, template_widths_mean_var_reci)
        return self

    def fit(self, X, R, template_prior=None):
        """Fit TFA to one subject's [n_voxel, n_tr] data with coordinates
        R [n_voxel, n_dim]."""
        if self.verbose:
            logger.info('Start to fit TFA')
        if not isinstance(X, np.ndarray):
            raise TypeError("Input data should be an array")
        if X.ndim != 2:
            raise TypeError("Input data should be 2D array")
        if not isinstance(R, np.ndarray):
            raise TypeError("Input coordinate matrix should be an array")
        if R.ndim != 2:
            raise TypeError("Input coordinate matrix should be 2D array")
        if X.shape[0] != R.shape[0]:
            raise TypeError(
                "The number of voxels should be the same in X and R!")
        if self.weight_method not in ('rr', 'ols'):
            raise ValueError(
                "only 'rr' and 'ols' are accepted as weight_method!")

        self._prepare_fit(X, R, template_prior)
        self._fit_tfa(X, R, template_prior)
        if template_prior is None:
            centers = self.get_centers(self.local_posterior_)
            widths = self.get_widths(self.local_posterior_)
            unique_R, inds = self.get_unique_R(R)
            self.F_ = self.get_factors(unique_R, inds, centers, widths)
            self.W_ = self.get_weights(X, self.F_)
        return self

    # sklearn-style params
    def get_params(self, deep=True):
        return {"max_iter": self.miter, "threshold": self.threshold,
                "K": self.K, "nlss_method": self.nlss_method,
                "nlss_loss": self.nlss_loss, "jac": self.jac,
                "nlss_max_nfev": self.nlss_max_nfev,
                "x_scale": self.x_scale, "tr_solver": self.tr_solver,
                "weight_method": self.weight_method,
                "upper_ratio": self.upper_ratio,
                "lower_ratio": self.lower_ratio,
                "max_num_tr": self.max_num_tr,
                "max_num_voxel": self.max_num_voxel,
                "seed": self.seed, "verbose": self.verbose}

    def set_params(self, **params):
        mapping = {"max_iter": "miter"}
        for k, v in params.items():
            setattr(self, mapping.get(k, k), v)
        return self
