"""Batched LARS / lasso-LARS feature pre-selection (K8).

shap 0.35.0's ``l1_reg`` branch fits a per-instance ``LassoLarsIC`` (or
``lars_path`` / ``Lasso``) on the weighted mask matrix — a host loop that is
O(instances) Python/sklearn calls (reference call chain
``explainers/kernel_shap.py:250`` -> shap ``solve``). Here the whole batch is
solved at once, in covariance form: the (m x m) weighted Gram matrix and
correlation vector are built ON DEVICE by the MFMA ``wls_gram`` kernel (the
S-dependent part — communication-free, one pass over the packed mask bits)
and the LARS path itself runs as a BATCHED torch fp64 iteration over
instances: each step appends (or drops) one feature per instance, maintaining
the inverse of the active-set Gram via Schur-complement rank-1 updates, so a
step costs O(b * m) tensor work and the full path is O(b * m^2) — independent
of nsamples.

Selection criteria match sklearn's (the CPU oracle):
* ``aic`` / ``bic``: lasso path, criterion ``RSS/sigma2 + factor*df`` with
  ``sigma2`` the OLS noise variance (sklearn>=1.2 ``LassoLarsIC``);
* ``num_features(k)``: plain LAR (no drops) stopped after k steps
  (sklearn ``lars_path(max_iter=k)``);
* float alpha: lasso path stopped at ``C/n <= alpha`` (the lars-lasso
  solution at that regularisation).
"""
from __future__ import annotations

from typing import Optional

import numpy as np

__all__ = ["batched_lars_select"]

_TINY = 1e-300


def batched_lars_select(
    G,
    c,
    yty,
    n_samples: int,
    mode: str = "aic",
    num_features: Optional[int] = None,
    alpha: Optional[float] = None,
    max_steps: Optional[int] = None,
    tol: float = 1e-10,
    zbar=None,
    ybar=None,
):
    """Select active features per instance from weighted normal equations.

    Parameters
    ----------
    G : (b, m, m) fp64 torch tensor — Z^T diag(w) Z.
    c : (b, m) fp64 torch tensor — Z^T diag(w) y.
    yty : (b,) fp64 torch tensor — sum w*y^2.
    n_samples : number of samples behind the normal equations.
    mode : 'aic' | 'bic' | 'num_features' | 'alpha'.
    zbar, ybar : optional (b, m) / (b,) means of the weighted design rows
        (sqrt(w)*Z) and response (sqrt(w)*y). When given, the normal
        equations are centred — sklearn's ``fit_intercept=True`` behaviour
        of ``LassoLarsIC`` / ``Lasso`` (``lars_path`` for num_features does
        NOT centre, so pass None there).

    Returns a (b, m) bool torch tensor — the selected support.
    """
    import torch as t

    b, m, _ = G.shape
    dev = G.device
    lasso = mode in ("aic", "bic", "alpha")
    if zbar is not None:
        # centre: (zw - 1 zbar)^T (zw - 1 zbar) = G - n zbar zbar^T, etc.
        n = float(n_samples)
        G = G - n * zbar.unsqueeze(2) * zbar.unsqueeze(1)
        c = c - n * zbar * ybar.unsqueeze(1)
        yty = yty - n * ybar * ybar
    kmax = m if num_features is None else min(num_features, m)
    if max_steps is None:
        max_steps = 8 * m if lasso else kmax

    K = t.zeros(b, m, m, dtype=t.float64, device=dev)   # inv(G_AA), slot order
    act = t.full((b, m), m, dtype=t.int64, device=dev)  # slot -> feature (m = pad)
    in_act = t.zeros(b, m, dtype=t.bool, device=dev)
    sgn = t.zeros(b, m, dtype=t.float64, device=dev)    # per slot
    beta = t.zeros(b, m, dtype=t.float64, device=dev)   # per feature
    r = c.clone()                                       # residual correlations
    k = t.zeros(b, dtype=t.int64, device=dev)
    frozen = t.zeros(b, dtype=t.bool, device=dev)
    drop_pending = t.zeros(b, dtype=t.bool, device=dev)
    drop_slot = t.zeros(b, dtype=t.int64, device=dev)

    crit_factor = float(np.log(n_samples)) if mode == "bic" else 2.0
    if mode in ("aic", "bic"):
        # sklearn LassoLarsIC noise variance: OLS residual / (n - m - 1)
        try:
            ols = t.linalg.solve(
                G + 1e-12 * t.eye(m, dtype=t.float64, device=dev), c.unsqueeze(2)
            ).squeeze(2)
        except Exception:
            ols = t.zeros_like(c)
        rss_ols = (yty - (c * ols).sum(1)).clamp_min(1e-300)
        dof = max(n_samples - m - 1, 1)
        sigma2 = (rss_ols / dof).clamp_min(1e-300)
        best_crit = yty / sigma2                         # k = 0 point
        best_support = t.zeros(b, m, dtype=t.bool, device=dev)
    else:
        sigma2 = None
        best_crit = None
        best_support = None

    gdiag = t.diagonal(G, dim1=1, dim2=2)                # (b, m)

    for _step in range(max_steps):
        live = ~frozen
        dropping = drop_pending & live
        appending = live & ~dropping
        # ONE host sync per step: all control flags in a single transfer
        flags = t.stack([
            live.any(), dropping.any(), appending.any()
        ]).cpu()
        if not bool(flags[0]):
            break

        if bool(flags[1]):
            # remove slot p: permute it to the last active slot, then reduce
            # K by the reverse Schur complement  K11 - k1 k1^T / k22
            di = t.nonzero(dropping).squeeze(1)
            for i in di.tolist():                        # drops are rare
                p = int(drop_slot[i])
                ki = int(k[i])
                perm = list(range(ki))
                perm.append(perm.pop(p))                 # move p to end
                pidx = t.tensor(perm, device=dev)
                Ki = K[i, :ki, :ki][pidx][:, pidx]
                k22 = Ki[-1, -1].clamp_min(_TINY)
                k1 = Ki[:-1, -1]
                Kr = Ki[:-1, :-1] - t.outer(k1, k1) / k22
                K[i] = 0.0
                K[i, : ki - 1, : ki - 1] = Kr
                feat = int(act[i, p])
                in_act[i, feat] = False
                beta[i, feat] = 0.0
                order = pidx[:-1]
                act_new = act[i, :ki][order]
                sgn_new = sgn[i, :ki][order]
                act[i] = m
                act[i, : ki - 1] = act_new
                sgn[i] = 0.0
                sgn[i, : ki - 1] = sgn_new
                k[i] = ki - 1
            drop_pending = drop_pending & ~dropping

        # most-correlated inactive feature
        rmask = r.masked_fill(in_act, 0.0)
        Cval, jstar = rmask.abs().max(dim=1)

        stop_now = appending & ((Cval < tol) | (k >= kmax))
        if mode == "alpha":
            stop_now = stop_now | (appending & (Cval / n_samples <= alpha))
        frozen = frozen | stop_now
        appending = appending & ~stop_now
        live = ~frozen

        # all slot-space work is confined to the leading kcap slots. k grows
        # by at most 1 per step, so min(step+1, m) bounds every live count
        # WITHOUT a device sync, and the padded slots are exact zeros.
        kcap = min(_step + 1, m)
        if bool(flags[2]):
            app = appending
            gj = t.gather(
                G, 2, jstar.view(b, 1, 1).expand(b, m, 1)
            ).squeeze(2)                                  # (b, m) = G[:, :, j*]
            actc = act[:, :kcap]
            slotmask = (actc < m)
            g_ord = t.gather(gj, 1, actc.clamp(max=m - 1)) * slotmask
            Kv = K[:, :kcap, :kcap]
            u = t.bmm(Kv, g_ord.unsqueeze(2)).squeeze(2)  # (b, kcap)
            gjj = t.gather(gdiag, 1, jstar.unsqueeze(1)).squeeze(1)
            schur = (gjj - (g_ord * u).sum(1)).clamp_min(1e-12)
            inv_s = t.where(app, 1.0 / schur, t.zeros_like(schur))
            # K <- [[K + u u^T/s, -u/s], [-u^T/s, 1/s]]  (u is 0 beyond k)
            Kv += inv_s.view(b, 1, 1) * u.unsqueeze(2) * u.unsqueeze(1)
            newrow = -u * inv_s.unsqueeze(1)
            ai = t.nonzero(app).squeeze(1)
            K[ai, k[ai], :kcap] = newrow[ai]
            K[ai, :kcap, k[ai]] = newrow[ai]
            K[ai, k[ai], k[ai]] = inv_s[ai]
            act[ai, k[ai]] = jstar[ai]
            rj = t.gather(r, 1, jstar.unsqueeze(1)).squeeze(1)
            sgn[ai, k[ai]] = t.sign(rj[ai])
            in_act[ai, jstar[ai]] = True
            k = k + app.long()
            kcap = min(_step + 2, m)   # still sync-free: k <= _step+1

        # equiangular direction over the (updated) active sets of live rows
        actc = act[:, :kcap]
        sgnc = sgn[:, :kcap]
        w = t.bmm(K[:, :kcap, :kcap], sgnc.unsqueeze(2)).squeeze(2)
        denom = ((sgnc * w).sum(1)).clamp_min(1e-300)
        AA = 1.0 / t.sqrt(denom)
        w = w * AA.unsqueeze(1)
        # scatter slot direction to feature space
        d_full = t.zeros(b, m + 1, dtype=t.float64, device=dev)
        d_full.scatter_(1, actc, t.where(actc < m, w, t.zeros_like(w)))
        d_full = d_full[:, :m]
        a = t.bmm(G, d_full.unsqueeze(2)).squeeze(2)      # (b, m)

        # the shared active correlation level (all active |r_j| are equal and
        # maximal along the path)
        Cv = r.abs().max(dim=1).values
        eps = 1e-12
        cand1 = (Cv.unsqueeze(1) - r) / (AA.unsqueeze(1) - a + eps)
        cand2 = (Cv.unsqueeze(1) + r) / (AA.unsqueeze(1) + a + eps)
        inf = t.full_like(cand1, float("inf"))
        cand1 = t.where((cand1 > eps) & ~in_act, cand1, inf)
        cand2 = t.where((cand2 > eps) & ~in_act, cand2, inf)
        gamma = t.minimum(cand1.min(dim=1).values, cand2.min(dim=1).values)
        gamma_max = Cv / AA                              # drive |r| to zero
        gamma = t.minimum(gamma, gamma_max)

        if lasso:
            beta_ord = t.gather(beta, 1, actc.clamp(max=m - 1))
            infk = t.full_like(w, float("inf"))
            gd = t.where(
                (actc < m) & (w.abs() > eps) & (-beta_ord / w > eps)
                & (beta_ord.abs() > 0),
                -beta_ord / w, infk,
            )
            gamma_d, pslot = gd.min(dim=1)
            do_drop = live & (gamma_d < gamma)
            gamma = t.where(do_drop, gamma_d, gamma)
            drop_pending = drop_pending | do_drop
            drop_slot = t.where(do_drop, pslot, drop_slot)

        alpha_hit = None
        if mode == "alpha":
            # the lasso solution at alpha sits BETWEEN breakpoints: stop with
            # the partial step that brings C down to exactly n*alpha
            target = alpha * n_samples
            g_star = ((Cv - target) / AA).clamp_min(0.0)
            alpha_hit = live & (Cv - gamma * AA < target)
            gamma = t.where(alpha_hit, t.minimum(gamma, g_star), gamma)
            drop_pending = drop_pending & ~alpha_hit

        gamma = t.where(live, gamma, t.zeros_like(gamma))
        beta = beta + gamma.unsqueeze(1) * d_full
        r = r - gamma.unsqueeze(1) * a

        # reaching gamma_max with no drop = end of path for that instance
        hit_end = live & ~drop_pending & (gamma >= gamma_max - 1e-15)
        if alpha_hit is not None:
            hit_end = hit_end | alpha_hit

        if mode in ("aic", "bic"):
            # RSS = yty - beta.c - beta.r   (since G beta = c - r)
            rss = (yty - (beta * c).sum(1) - (beta * r).sum(1)).clamp_min(0.0)
            nz = in_act & (beta.abs() > 1e-12)           # sklearn df = #nonzero
            crit = rss / sigma2 + crit_factor * nz.sum(1).double()
            improved = live & (crit < best_crit)
            best_crit = t.where(improved, crit, best_crit)
            best_support = t.where(improved.unsqueeze(1), nz, best_support)

        frozen = frozen | hit_end

    if mode in ("aic", "bic"):
        return best_support
    if mode == "alpha":
        return in_act & (beta.abs() > 1e-12)
    return in_act
