"""Hierarchical Topographical Factor Analysis (multi-subject).

API parity with the reference HTFA (ref src/brainiak/factoranalysis/
htfa.py:62-841): subjects sharded across ranks, per-iteration global
template broadcast, local per-subject TFA fits, posterior gather to
root, MAP update of the global template with Hungarian factor matching,
and convergence flag broadcast.

Communication is the DistContext (RCCL over xGMI / gloo) instead of
mpi4py: the reference's Bcast + Gatherv + Bcast triple
(htfa.py:730, 548-557, 754) becomes broadcast → gather_object →
broadcast_object per global iteration (posterior vectors are
K*(n_dim+1) doubles — latency-, not bandwidth-bound).
"""

import logging

import numpy as np
from scipy.optimize import linear_sum_assignment
from scipy.spatial import distance

from ..parallel import DistContext
from ..utils.utils import from_sym_2_tri, from_tri_2_sym
from .tfa import TFA

logger = logging.getLogger(__name__)

__all__ = ["HTFA"]


class HTFA(TFA):
    """Multi-subject HTFA with a global template prior.

    Constructor matches the reference: K, n_subj, max_global_iter,
    max_local_iter, threshold, nlss_method, nlss_loss, jac, x_scale,
    tr_solver, weight_method, upper_ratio, lower_ratio, voxel_ratio,
    tr_ratio, max_voxel, max_tr, comm (a DistContext), verbose.
    """

    def __init__(self, K, n_subj, max_global_iter=10, max_local_iter=10,
                 threshold=1.0, nlss_method='trf', nlss_loss='soft_l1',
                 jac='2-point', x_scale='jac', tr_solver=None,
                 weight_method='rr', upper_ratio=1.8, lower_ratio=0.02,
                 voxel_ratio=0.25, tr_ratio=0.1, max_voxel=5000,
                 max_tr=500, comm=None, verbose=False, device=None):
        self.K = K
        self.n_subj = n_subj
        self.max_global_iter = max_global_iter
        self.max_local_iter = max_local_iter
        self.threshold = threshold
        self.nlss_method = nlss_method
        self.nlss_loss = nlss_loss
        self.jac = jac
        self.x_scale = x_scale
        self.tr_solver = tr_solver
        self.weight_method = weight_method
        self.upper_ratio = upper_ratio
        self.lower_ratio = lower_ratio
        self.voxel_ratio = voxel_ratio
        self.tr_ratio = tr_ratio
        self.max_voxel = max_voxel
        self.max_tr = max_tr
        self.comm = comm if isinstance(comm, DistContext) else None
        self.verbose = verbose
        self.device = device

    def _ctx(self):
        if self.comm is None:
            self.comm = DistContext()
        return self.comm

    # -- convergence on the global template ---------------------------------

    def _converged(self):
        prior = self.global_prior_[0:self.prior_size]
        posterior = self.global_posterior_[0:self.prior_size]
        diff = prior - posterior
        max_diff = np.max(np.fabs(diff))
        if self.verbose:
            _, mse = self._mse_converged()
            diff_ratio = np.sum(diff ** 2) / np.sum(posterior ** 2)
            logger.info('htfa prior posterior max diff %f mse %f '
                        'diff_ratio %f', max_diff, mse, diff_ratio)
        return (max_diff <= self.threshold), max_diff

    def _mse_converged(self):
        prior = self.global_prior_[0:self.prior_size]
        posterior = self.global_posterior_[0:self.prior_size]
        mse = np.mean((prior - posterior) ** 2)
        return (mse <= self.threshold), mse

    # -- MAP update ----------------------------------------------------------

    @staticmethod
    def _map_update(prior_mean, prior_cov, global_cov_scaled,
                    new_observation):
        """Gaussian MAP combination of the prior mean with the mean of the
        per-subject observations."""
        common = np.linalg.inv(prior_cov + global_cov_scaled)
        observation_mean = np.mean(new_observation, axis=1)
        posterior_mean = prior_cov.dot(common.dot(observation_mean)) + \
            global_cov_scaled.dot(common.dot(prior_mean))
        posterior_cov = prior_cov.dot(common.dot(global_cov_scaled))
        return posterior_mean, posterior_cov

    def _map_update_posterior(self):
        self.global_posterior_ = self.global_prior_.copy()
        prior_centers = self.get_centers(self.global_prior_)
        prior_widths = self.get_widths(self.global_prior_)
        prior_centers_mean_cov = self.get_centers_mean_cov(
            self.global_prior_)
        prior_widths_mean_var = self.get_widths_mean_var(
            self.global_prior_)
        center_size = self.K * self.n_dim
        posterior_size = center_size + self.K
        for k in np.arange(self.K):
            next_centers = np.zeros((self.n_dim, self.n_subj))
            next_widths = np.zeros(self.n_subj)
            for s in np.arange(self.n_subj):
                center_start = s * posterior_size
                width_start = center_start + center_size
                start_idx = center_start + k * self.n_dim
                end_idx = center_start + (k + 1) * self.n_dim
                next_centers[:, s] = \
                    self.gather_posterior[start_idx:end_idx].copy()
                next_widths[s] = self.gather_posterior[width_start + k]

            cov = from_tri_2_sym(prior_centers_mean_cov[k], self.n_dim)
            cov = cov + cov.T - np.diag(np.diag(cov))
            posterior_mean, posterior_cov = self._map_update(
                prior_centers[k].T.copy(), cov,
                self.global_centers_cov_scaled, next_centers)
            self.global_posterior_[k * self.n_dim:(k + 1) * self.n_dim] = \
                posterior_mean.T
            start_idx = int(self.map_offset[2]) + k * self.cov_vec_size
            end_idx = int(self.map_offset[2]) + (k + 1) * self.cov_vec_size
            self.global_posterior_[start_idx:end_idx] = \
                from_sym_2_tri(posterior_cov)

            pw_var = float(prior_widths_mean_var[k])
            common = 1.0 / (pw_var + self.global_widths_var_scaled)
            observation_mean = np.mean(next_widths)
            tmp = common * self.global_widths_var_scaled
            self.global_posterior_[int(self.map_offset[1]) + k] = \
                pw_var * common * observation_mean + \
                tmp * float(prior_widths[k])
            self.global_posterior_[int(self.map_offset[3]) + k] = \
                pw_var * tmp
        return self

    def _assign_posterior(self):
        """Hungarian-match global posterior factors onto the prior order."""
        prior_centers = self.get_centers(self.global_prior_)
        posterior_centers = self.get_centers(self.global_posterior_)
        posterior_widths = self.get_widths(self.global_posterior_)
        posterior_centers_mean_cov = self.get_centers_mean_cov(
            self.global_posterior_)
        posterior_widths_mean_var = self.get_widths_mean_var(
            self.global_posterior_)
        cost = distance.cdist(prior_centers, posterior_centers,
                              'euclidean')
        _, col_ind = linear_sum_assignment(cost)
        self.set_centers(self.global_posterior_,
                         posterior_centers[col_ind])
        self.set_widths(self.global_posterior_, posterior_widths[col_ind])
        self.set_centers_mean_cov(self.global_posterior_,
                                  posterior_centers_mean_cov[col_ind])
        self.set_widths_mean_var(self.global_posterior_,
                                 posterior_widths_mean_var[col_ind])
        return self

    # -- subject metadata ----------------------------------------------------

    def _get_subject_info(self, n_local_subj, data):
        max_sample_tr = np.zeros(n_local_subj).astype(int)
        max_sample_voxel = np.zeros(n_local_subj).astype(int)
        for idx in np.arange(n_local_subj):
            nvoxel, ntr = data[idx].shape
            max_sample_voxel[idx] = min(self.max_voxel,
                                        int(self.voxel_ratio * nvoxel))
            max_sample_tr[idx] = min(self.max_tr,
                                     int(self.tr_ratio * ntr))
        return max_sample_tr, max_sample_voxel

    def _get_weight_size(self, data, n_local_subj):
        weight_size = np.zeros(1).astype(int)
        local_weight_offset = np.zeros(n_local_subj).astype(int)
        for idx, subj_data in enumerate(data):
            if idx > 0:
                local_weight_offset[idx] = weight_size[0]
            weight_size[0] += self.K * subj_data.shape[1]
        return weight_size, local_weight_offset

    def _init_prior_posterior(self, ctx, R, n_local_subj):
        if ctx.is_root:
            idx = np.random.choice(n_local_subj, 1)
            self.global_prior_, self.global_centers_cov, \
                self.global_widths_var = self.get_template(R[idx[0]])
            self.global_centers_cov_scaled = \
                self.global_centers_cov / float(self.n_subj)
            self.global_widths_var_scaled = \
                self.global_widths_var / float(self.n_subj)
            self.gather_posterior = np.zeros(
                self.n_subj * self.prior_size)
            self.global_posterior_ = np.zeros(self.prior_size)
        else:
            self.global_prior_ = np.zeros(self.prior_bcast_size)
            self.global_posterior_ = None
            self.gather_posterior = None
        return self

    def _update_weight(self, data, R, n_local_subj, local_weight_offset):
        for s, subj_data in enumerate(data):
            base = s * self.prior_size
            centers = self.local_posterior_[
                base:base + self.K * self.n_dim].reshape(
                    (self.K, self.n_dim))
            widths = self.local_posterior_[
                base + self.K * self.n_dim:base + self.prior_size].reshape(
                    (self.K, 1))
            unique_R, inds = self.get_unique_R(R[s])
            F = self.get_factors(unique_R, inds, centers, widths)
            start_idx = local_weight_offset[s]
            if s == n_local_subj - 1:
                self.local_weights_[start_idx:] = \
                    self.get_weights(subj_data, F).ravel()
            else:
                end_idx = local_weight_offset[s + 1]
                self.local_weights_[start_idx:end_idx] = \
                    self.get_weights(subj_data, F).ravel()
        return self

    # -- main loop ------------------------------------------------------------

    def _fit_htfa(self, data, R):
        ctx = self._ctx()
        n_local_subj = len(R)
        max_sample_tr, max_sample_voxel = self._get_subject_info(
            n_local_subj, data)

        tfa = []
        for s in range(n_local_subj):
            tfa.append(TFA(
                max_iter=self.max_local_iter, threshold=self.threshold,
                K=self.K, nlss_method=self.nlss_method,
                nlss_loss=self.nlss_loss, x_scale=self.x_scale,
                tr_solver=self.tr_solver,
                weight_method=self.weight_method,
                upper_ratio=self.upper_ratio,
                lower_ratio=self.lower_ratio, verbose=self.verbose,
                max_num_tr=int(max_sample_tr[s]),
                max_num_voxel=int(max_sample_voxel[s]),
                device=self.device))

        self.local_posterior_ = np.zeros(n_local_subj * self.prior_size)
        self._init_prior_posterior(ctx, R, n_local_subj)
        node_weight_size, local_weight_offset = self._get_weight_size(
            data, n_local_subj)
        self.local_weights_ = np.zeros(node_weight_size[0])

        m = 0
        outer_converged = np.array([0])
        while m < self.max_global_iter and not outer_converged[0]:
            if self.verbose:
                logger.info("HTFA global iter %d", m)
            self.global_prior_ = ctx.broadcast(self.global_prior_)
            for s, subj_data in enumerate(data):
                tfa[s].set_prior(
                    self.global_prior_[0:self.prior_size].copy())
                tfa[s].set_seed(m * self.max_local_iter)
                tfa[s].fit(subj_data, R=R[s],
                           template_prior=self.global_prior_.copy())
                tfa[s]._assign_posterior()
                self.local_posterior_[
                    s * self.prior_size:(s + 1) * self.prior_size] = \
                    tfa[s].local_posterior_

            gathered = ctx.gather_object(self.local_posterior_)
            if ctx.is_root:
                self.gather_posterior = np.concatenate(gathered)
                self._map_update_posterior()
                self._assign_posterior()
                is_converged, _ = self._converged()
                if is_converged:
                    logger.info("converged at %d outer iter", m)
                    outer_converged[0] = 1
                else:
                    self.global_prior_ = self.global_posterior_
            outer_converged = ctx.broadcast_object(outer_converged)
            m += 1

        self._update_weight(data, R, n_local_subj, local_weight_offset)
        return self

    def _check_input(self, X, R):
        if not isinstance(X, list):
            raise TypeError("Input data should be a list")
        if not isinstance(R, list):
            raise TypeError("Coordinates should be a list")
        if len(X) < 1:
            raise ValueError("Need at leat one subject to train the model."
                             " Got {0:d}".format(len(X)))
        for idx, x in enumerate(X):
            if not isinstance(x, np.ndarray):
                raise TypeError("Each subject data should be an array")
            if x.ndim != 2:
                raise TypeError("Each subject data should be 2D array")
            if not isinstance(R[idx], np.ndarray):
                raise TypeError(
                    "Each scanner coordinate matrix should be an array")
            if R[idx].ndim != 2:
                raise TypeError(
                    "Each scanner coordinate matrix should be 2D array")
            if x.shape[0] != R[idx].shape[0]:
                raise TypeError(
                    "n_voxel should be the same in X[idx] and R[idx]")
        return self

    def fit(self, X, R):
        """Fit HTFA: X/R are THIS rank's subjects (lists of [V_i, T_i]
        data and [V_i, n_dim] coordinates)."""
        self._check_input(X, R)
        if self.verbose:
            logger.info("Start to fit HTFA")
        self.n_dim = R[0].shape[1]
        self.cov_vec_size = np.sum(np.arange(self.n_dim) + 1)
        self.prior_size = self.K * (self.n_dim + 1)
        self.prior_bcast_size = \
            self.K * (self.n_dim + 2 + self.cov_vec_size)
        self.get_map_offset()
        self._fit_htfa(X, R)
        return self
