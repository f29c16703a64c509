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
import os
import math

import numpy as np
import torch
from scipy.optimize import linear_sum_assignment
from scipy.spatial import distance

from ..parallel import DistContext
from ..utils.utils import from_sym_2_tri, from_tri_2_sym
from .tfa import TFA

logger = logging.getLogger(__name__)

__all__ = ["HTFA"]


# ---------------------------------------------------------------------------
# batched local TFA fit: one Levenberg-Marquardt solve over ALL local
# subjects per inner iteration (the sequential per-subject loop pays
# ~0.5 ms of Python + stream syncs per residual evaluation x ~2000
# evaluations per global iteration; batching divides that by the
# subject count).  Math per subject is identical to
# TFA._estimate_centers_widths_torch_lm: per-subject damping factor,
# per-subject accept/reject, soft_l1 IRLS weights, bound clamps.
# ---------------------------------------------------------------------------

def _soft_l1_cost_batch(r):
    """2*((1+r^2)^.5 - 1) summed per subject; r [S, n_res] -> [S]."""
    z = r * r
    return (2.0 * ((1 + z).sqrt() - 1)).sum(dim=1)


class _BatchedLM:
    """State for the batched-across-subjects LM solve."""

    def __init__(self, tfas, coords, X, W, init, sigma, template_prior):
        dev = "cuda"
        t0 = tfas[0]
        self.K, self.D = t0.K, t0.n_dim
        self.n_par = self.K * (self.D + 1)
        S = len(tfas)
        self.S = S
        self.co = torch.stack([TFA._to_dev_f32(c) for c in coords])
        self.X = torch.stack([TFA._to_dev_f32(x) for x in X])
        self.W = W                                   # [S, K, T] device
        self.lb = torch.stack([
            torch.as_tensor(t.bounds[0], dtype=torch.float32,
                            device=dev) for t in tfas])
        self.ub = torch.stack([
            torch.as_tensor(t.bounds[1], dtype=torch.float32,
                            device=dev) for t in tfas])
        self.sigma = torch.as_tensor(np.asarray(sigma, np.float32),
                                     device=dev)     # [S]
        self.theta = torch.stack([
            torch.as_tensor(np.asarray(i, np.float32), device=dev)
            for i in init]).clamp(self.lb, self.ub)
        self.scale = torch.as_tensor(
            np.asarray([t.sample_scaling for t in tfas], np.float32),
            device=dev)
        self.max_nfev = t0.nlss_max_nfev or 20

        # shared template prior (HTFA broadcasts ONE global prior)
        tc_np = t0.get_centers(template_prior)
        tw_np = t0.get_widths(template_prior).ravel()
        cov_tri = t0.get_centers_mean_cov(template_prior)
        reci_np = (1.0 / t0.get_widths_mean_var(template_prior)).ravel()
        cov_inv = []
        for k in range(self.K):
            cov = from_tri_2_sym(cov_tri[k], self.D)
            cov = cov + cov.T - np.diag(np.diag(cov))
            cov_inv.append(np.linalg.inv(cov))
        as_f32 = lambda a: torch.as_tensor(  # noqa: E731
            np.asarray(a, np.float32), device=dev)
        self.tc, self.tw = as_f32(tc_np), as_f32(tw_np)
        self.reci, self.cov_inv = as_f32(reci_np), as_f32(cov_inv)

    def _split(self, theta):
        KD = self.K * self.D
        return (theta[:, :KD].reshape(self.S, self.K, self.D),
                theta[:, KD:])

    def residual_jac(self, theta, want_jac=True):
        S, K, D, n_par = self.S, self.K, self.D, self.n_par
        ce, wd = self._split(theta)
        diff = self.co[:, :, None, :] - ce[:, None, :, :]  # [S,V,K,D]
        d2 = (diff * diff).sum(-1)
        F = torch.exp(-d2 / wd[:, None, :])
        Rres = self.sigma[:, None, None] * (self.X
                                            - torch.bmm(F, self.W))
        parts = [Rres.reshape(S, -1)]
        J = None
        if want_jac:
            dF = torch.empty((S, F.shape[1], K, D + 1),
                             dtype=torch.float32, device=F.device)
            dF[..., :D] = F[..., None] * 2.0 * diff                 / wd[:, None, :, None]
            dF[..., D] = F * d2 / (wd * wd)[:, None, :]
            Jr = -self.sigma[:, None, None, None, None] * torch.einsum(
                'skt,svkj->svtkj', self.W, dF)
            Jr = Jr.reshape(S, -1, K, D + 1)
            J = torch.cat([Jr[..., :D].reshape(S, -1, K * D),
                           Jr[..., D]], dim=2)       # [S, VT, n_par]
        # template-prior residual rows (always present in HTFA)
        dfc = ce - self.tc[None]                     # [S,K,D]
        solved = torch.einsum('kij,skj->ski', self.cov_inv, dfc)
        qc = (self.scale[:, None]
              * (dfc * solved).sum(-1)).clamp_min(1e-30)
        rc = qc.sqrt()
        rw_sq = (self.scale[:, None] * self.reci
                 * (wd - self.tw) ** 2).clamp_min(0.0)
        rw = rw_sq.sqrt()
        parts += [rc, rw]
        r = torch.cat(parts, dim=1)
        if want_jac:
            Jp = torch.zeros((S, 2 * K, n_par), dtype=torch.float32,
                             device=F.device)
            grad_c = self.scale[:, None, None] * solved / rc[..., None]
            ar = torch.arange(K, device=F.device)
            for d in range(D):
                Jp[:, ar, ar * D + d] = grad_c[..., d]
            Jp[:, K + ar, K * D + ar] =                 (self.scale[:, None] * self.reci).sqrt()                 * torch.sign(wd - self.tw)
            J = torch.cat([J, Jp], dim=1)
        return r, J

    def eval_costs_subset(self, subj_idx, thetas):
        """soft_l1 costs for a [R*k, n_par] candidate stack covering
        subjects ``subj_idx`` (k candidates each, repeat-interleaved
        order) — one residual batch, one host sync."""
        N = thetas.shape[0]
        k = N // len(subj_idx)
        idx = torch.as_tensor(np.repeat(subj_idx, k),
                              device=self.theta.device)
        saved = (self.co, self.X, self.W, self.sigma, self.scale,
                 self.S)
        try:
            self.co, self.X, self.W = (self.co[idx], self.X[idx],
                                       self.W[idx])
            self.sigma = self.sigma[idx]
            self.scale = self.scale[idx]
            self.S = N
            r, _ = self.residual_jac(thetas, want_jac=False)
            return _soft_l1_cost_batch(r).double().cpu().numpy()
        finally:
            (self.co, self.X, self.W, self.sigma, self.scale,
             self.S) = saved

    def solve(self):
        """Run the batched LM; returns theta [S, n_par] fp64 numpy."""
        S = self.S
        lam = np.full(S, 1e-3)
        alive = np.ones(S, dtype=bool)
        r, J = self.residual_jac(self.theta)
        cost = _soft_l1_cost_batch(r).double().cpu().numpy()
        for _ in range(int(self.max_nfev)):
            if not alive.any():
                break
            wgt = (1 + r * r).pow(-0.25)
            rw_ = (r * wgt).unsqueeze(-1)
            Jw = J * wgt[..., None]
            A = torch.bmm(Jw.transpose(1, 2), Jw)
            g = torch.bmm(Jw.transpose(1, 2), rw_).squeeze(-1)
            A_h = A.double().cpu().numpy()
            g_h = g.double().cpu().numpy()
            dA_h = np.clip(np.einsum('sii->si', A_h), 1e-12, None)
            # hybrid damping: try the current lambda for every
            # subject first (the typical accept path — one solve, one
            # cost sync, exactly like the sequential loop's best
            # case); ONLY subjects that reject get the remaining
            # ladder (4*lam .. 4^5*lam) as one stacked solve + one
            # batched cost evaluation, instead of up to 5 more
            # sync rounds.
            NT = 6

            def ladder_deltas(As, gs, lams):
                n = self.n_par
                dAs = np.einsum('sii->si', As).clip(1e-12, None)
                damp = As + lams[:, None, None] \
                    * (np.eye(n)[None] * dAs[:, :, None])
                try:
                    return np.linalg.solve(
                        damp, -gs[:, :, None])[:, :, 0]
                except np.linalg.LinAlgError:
                    return np.stack([
                        np.linalg.lstsq(d, -r, rcond=None)[0]
                        for d, r in zip(damp, gs)])

            d0 = ladder_deltas(A_h, g_h, lam)
            cand0 = (self.theta + torch.as_tensor(
                d0, dtype=torch.float32,
                device=self.theta.device)).clamp(self.lb, self.ub)
            rc0, _ = self.residual_jac(cand0, want_jac=False)
            c0 = _soft_l1_cost_batch(rc0).double().cpu().numpy()
            improved = alive & (c0 < cost)
            if improved.any():
                m = torch.as_tensor(improved, device=cand0.device)
                self.theta = torch.where(m[:, None], cand0,
                                         self.theta)
                cost[improved] = c0[improved]
                lam[improved] = np.maximum(lam[improved] / 3, 1e-8)
            rej = np.nonzero(alive & ~improved)[0]
            if len(rej):
                R = len(rej)
                lads = lam[rej, None] * (4.0 ** np.arange(1, NT))
                dR = ladder_deltas(
                    np.repeat(A_h[rej], NT - 1, axis=0),
                    np.repeat(g_h[rej], NT - 1, axis=0),
                    lads.reshape(-1))
                dR = torch.as_tensor(
                    dR.reshape(R, NT - 1, self.n_par),
                    dtype=torch.float32, device=self.theta.device)
                rej_t = torch.as_tensor(rej,
                                        device=self.theta.device)
                candR = (self.theta[rej_t][:, None] + dR).clamp(
                    self.lb[rej_t][:, None], self.ub[rej_t][:, None])
                costsR = self.eval_costs_subset(
                    rej,
                    candR.reshape(R * (NT - 1), self.n_par)
                ).reshape(R, NT - 1)
                first_ok = np.full(R, -1)
                for t in range(NT - 1):
                    hit = (first_ok < 0) & (costsR[:, t] < cost[rej])
                    first_ok[hit] = t
                acc = first_ok >= 0
                if acc.any():
                    sel = torch.as_tensor(np.maximum(first_ok, 0),
                                          device=candR.device)
                    chosen = candR[
                        torch.arange(R, device=candR.device), sel]
                    mfull = np.zeros(S, dtype=bool)
                    mfull[rej[acc]] = True
                    mt = torch.as_tensor(mfull, device=candR.device)
                    full = torch.zeros_like(self.theta)
                    full[rej_t] = chosen
                    self.theta = torch.where(mt[:, None], full,
                                             self.theta)
                    t_acc = first_ok[acc]
                    cost[rej[acc]] = costsR[acc, t_acc]
                    lam[rej[acc]] = np.maximum(
                        lads[acc, t_acc] / 3, 1e-8)
                    improved[rej[acc]] = True
                still = rej[~acc]
                lam[still] *= 4.0 ** NT
            alive &= improved
            if alive.any():
                r, J = self.residual_jac(self.theta)
        return (self.theta.double().cpu().numpy(), 0.5 * cost)


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

    def _template_delta(self):
        """(max |Δ|, mse, relative energy) between prior and posterior
        templates over the mean block."""
        a = self.global_prior_[:self.prior_size]
        b = self.global_posterior_[:self.prior_size]
        d = a - b
        return (float(np.abs(d).max()), float(np.mean(d ** 2)),
                float(np.sum(d ** 2) / max(np.sum(b ** 2), 1e-30)))

    def _converged(self):
        max_diff, mse, ratio = self._template_delta()
        if self.verbose:
            logger.info('htfa prior posterior max diff %f mse %f '
                        'diff_ratio %f', max_diff, mse, ratio)
        return (max_diff <= self.threshold), max_diff

    def _mse_converged(self):
        _, mse, _ = self._template_delta()
        return (mse <= self.threshold), mse

    # -- MAP update ----------------------------------------------------------

    def _map_update_posterior(self):
        """One BATCHED Gaussian MAP update of all K factors.

        The gathered per-subject posteriors reshape to
        ``centers [n_subj, K, D]`` / ``widths [n_subj, K]``; the
        per-factor center update
            μ* = Σp (Σp + Σg)⁻¹ x̄  +  Σg (Σp + Σg)⁻¹ μp
            Σ* = Σp (Σp + Σg)⁻¹ Σg
        runs as one [K, D, D] batched inverse + einsum instead of the
        reference's per-k / per-subject index walk
        (ref htfa.py:246-341).
        """
        self.global_posterior_ = self.global_prior_.copy()
        K, D = self.K, self.n_dim
        stacked = self.gather_posterior.reshape(self.n_subj,
                                                self.prior_size)
        obs_centers = stacked[:, :K * D].reshape(self.n_subj, K, D)
        obs_widths = stacked[:, K * D:K * (D + 1)]

        mu_p = self.get_centers(self.global_prior_)        # [K, D]
        w_p = self.get_widths(self.global_prior_).ravel()  # [K]
        tri = self.get_centers_mean_cov(self.global_prior_)
        cov_p = np.stack([from_tri_2_sym(tri[k], D) for k in range(K)])
        cov_p = cov_p + np.transpose(cov_p, (0, 2, 1)) \
            - np.einsum('kij,ij->kij', cov_p, np.eye(D))
        wvar_p = np.asarray(self.get_widths_mean_var(
            self.global_prior_), dtype=np.float64).ravel()

        sig_g = self.global_centers_cov_scaled                # [D, D]
        common = np.linalg.inv(cov_p + sig_g[None])           # [K, D, D]
        x_bar = obs_centers.mean(axis=0)                      # [K, D]
        mu_star = np.einsum('kij,kj->ki', cov_p,
                            np.einsum('kij,kj->ki', common, x_bar)) \
            + np.einsum('ij,kj->ki', sig_g,
                        np.einsum('kij,kj->ki', common, mu_p))
        cov_star = cov_p @ common @ sig_g[None]               # [K, D, D]

        wv_g = self.global_widths_var_scaled
        denom = wvar_p + wv_g
        w_star = (wvar_p * obs_widths.mean(axis=0)
                  + wv_g * w_p) / denom
        wvar_star = wvar_p * wv_g / denom

        self.global_posterior_[:K * D] = mu_star.ravel()
        self.global_posterior_[int(self.map_offset[1]):
                               int(self.map_offset[1]) + K] = w_star
        cov_tri = np.concatenate([from_sym_2_tri(cov_star[k])
                                  for k in range(K)])
        self.global_posterior_[int(self.map_offset[2]):
                               int(self.map_offset[2])
                               + K * self.cov_vec_size] = cov_tri
        self.global_posterior_[int(self.map_offset[3]):
                               int(self.map_offset[3]) + K] = wvar_star
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
        """Per-subject voxel/TR subsampling caps (ratio-scaled, clipped
        at max_voxel/max_tr — the reference's sizing rule)."""
        caps_tr = [min(self.max_tr, int(self.tr_ratio * d.shape[1]))
                   for d in data]
        caps_vox = [min(self.max_voxel,
                        int(self.voxel_ratio * d.shape[0]))
                    for d in data]
        return (np.asarray(caps_tr, dtype=int),
                np.asarray(caps_vox, dtype=int))

    def _get_weight_size(self, data, n_local_subj):
        """Flattened-weight total size and per-subject offsets."""
        sizes = np.asarray([self.K * d.shape[1] for d in data],
                           dtype=int)
        offsets = np.concatenate([[0], np.cumsum(sizes)[:-1]])
        return np.array([sizes.sum()]), offsets

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
        """Full-resolution ridge weights per subject from the final
        posterior factors, packed into the flat local_weights_ buffer."""
        del local_weight_offset, n_local_subj
        posts = self.local_posterior_.reshape(-1, self.prior_size)
        pieces = []
        on_dev = self._use_gpu() and R and R[0].shape[1] == 3
        for subj_data, coords, post in zip(data, R, posts):
            centers = self.get_centers(post)
            widths = self.get_widths(post)
            if on_dev:
                # factor matrix, ridge beta and weight solve all stay
                # on device: ONE [V, T] upload and one [K, T] download
                # per subject (the numpy path re-derived np.var and
                # round-tripped F through the host)
                Ft = self._get_factors_dev(
                    np.ascontiguousarray(coords, dtype=np.float64),
                    centers, widths)
                Xt = self._to_dev_f32(subj_data)
                W = self._get_weights_dev(Xt, Ft)
                pieces.append(W.ravel())
                continue
            unique_R, inds = self.get_unique_R(coords)
            F = self.get_factors(unique_R, inds, centers, widths)
            pieces.append(self.get_weights(subj_data, F).ravel())
        self.local_weights_ = np.concatenate(pieces)
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
            if self._batched_local_ok(data, R):
                self._fit_local_batched(
                    tfa, data, R, m, self.global_prior_.copy())
            else:
                for s, subj_data in enumerate(data):
                    tfa[s].set_prior(
                        self.global_prior_[0:self.prior_size].copy())
                    tfa[s].set_seed(m * self.max_local_iter)
                    tfa[s].fit(subj_data, R=R[s],
                               template_prior=self.global_prior_.copy())
                    tfa[s]._assign_posterior()
            for s in range(n_local_subj):
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

    def _batched_local_ok(self, data, R):
        """Batched local fits need: GPU, 3-D coords, every local
        subject the same data/coord shape (one shared subsample index
        set), and the kill switch unset."""
        if os.environ.get("BRAINIAK_HTFA_SEQ"):
            return False
        if not self._use_gpu():
            return False
        shapes = {d.shape for d in data}
        rdims = {r.shape[1] for r in R}
        return (len(shapes) == 1 and rdims == {3}
                and len({r.shape[0] for r in R}) == 1)

    def _fit_local_batched(self, tfas, data, R, m, template_prior):
        """All local subjects' TFA inner loops with ONE batched LM
        solve per iteration (same per-subject math as the sequential
        path; subjects that converge drop out of later iterations)."""
        S = len(tfas)
        for t, d, r in zip(tfas, data, R):
            t.set_prior(template_prior[0:self.prior_size].copy())
            t.set_seed(m * self.max_local_iter)
            t._prepare_fit(d, r, template_prior)
        t0 = tfas[0]
        np.random.seed(t0.seed)
        nfeature, nsample = data[0].shape
        n_vox = min(t0.max_num_voxel, nfeature)
        n_tr = min(t0.max_num_tr, nsample)
        converged = np.zeros(S, dtype=bool)
        for _ in range(t0.miter):
            if converged.all():
                break
            feat = np.random.choice(nfeature, n_vox, replace=False)
            samp = np.random.choice(nsample, n_tr, replace=False)
            idxs = [s for s in range(S) if not converged[s]]
            coords, Xs, Ws, inits, sigmas = [], [], [], [], []
            for s in idxs:
                t = tfas[s]
                curr = data[s][feat][:, samp]
                curr_R = np.ascontiguousarray(R[s][feat],
                                              dtype=np.float64)
                centers = t.get_centers(t.local_prior)
                widths = t.get_widths(t.local_prior)
                Ft = t._get_factors_dev(curr_R, centers, widths)
                Xt = TFA._to_dev_f32(curr)
                W = t._get_weights_dev(Xt, Ft)         # fp64 numpy
                coords.append(curr_R)
                Xs.append(curr)
                Ws.append(torch.as_tensor(W, dtype=torch.float32,
                                          device="cuda"))
                inits.append(np.hstack((centers.ravel(),
                                        widths.ravel())))
                sigmas.append(1.0 / math.sqrt(2.0) * np.std(curr))
            lm = _BatchedLM([tfas[s] for s in idxs], coords, Xs,
                            torch.stack(Ws), inits, sigmas,
                            template_prior)
            theta, costs = lm.solve()
            for j, s in enumerate(idxs):
                t = tfas[s]
                t.local_posterior_ = theta[j]
                t.total_cost = costs[j]
                t._assign_posterior()
                is_conv, _ = t._converged()
                if is_conv:
                    converged[s] = True
                else:
                    t.local_prior = t.local_posterior_
        return self

    def _check_input(self, X, R):
        if not isinstance(X, list) or not isinstance(R, list):
            raise TypeError("X and R must be lists (one entry per "
                            "local subject)")
        if not X:
            raise ValueError("need at least one local subject")
        for idx, (x, r) in enumerate(zip(X, R)):
            if not (isinstance(x, np.ndarray) and x.ndim == 2):
                raise TypeError(
                    "X[%d] must be a 2-D [voxels, TRs] array" % idx)
            if not (isinstance(r, np.ndarray) and r.ndim == 2):
                raise TypeError(
                    "R[%d] must be a 2-D [voxels, dims] array" % idx)
            if x.shape[0] != r.shape[0]:
                raise TypeError(
                    "X[%d] and R[%d] disagree on the voxel count "
                    "(%d vs %d)" % (idx, idx, x.shape[0], r.shape[0]))
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
