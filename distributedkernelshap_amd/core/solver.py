"""Constrained weighted-least-squares Shapley solve.

Reimplements the ``solve()`` step the reference delegates to shap 0.35.0
(driven from ``explainers/kernel_shap.py:250``): eliminate the last varying
feature using the local-accuracy constraint
``sum(phi) = link(f(x)) - link(fnull)``, solve the weighted normal equations
over the remaining M-1 features, and back-substitute.

The CPU implementation here doubles as the CI oracle for the HIP kernel
(K7 in SURVEY.md §2.4).
"""
from __future__ import annotations

from typing import Optional

import numpy as np


def solve_wls(
    masks: np.ndarray,
    kernel_weights: np.ndarray,
    ey_adj: np.ndarray,
    total: np.ndarray,
    nonzero_inds: Optional[np.ndarray] = None,
    m_full: Optional[int] = None,
) -> np.ndarray:
    """Solve the constrained Shapley WLS for one instance, all outputs at once.

    Parameters
    ----------
    masks : (S, M) {0,1} coalition matrix over the *varying* groups.
    kernel_weights : (S,) Shapley kernel weights.
    ey_adj : (S, n_out) ``link(ey) - link(fnull)`` per sample and output.
    total : (n_out,) ``link(fx) - link(fnull)`` per output.
    nonzero_inds : optional subset of mask columns to solve over (l1 path);
        defaults to all M columns.
    m_full : width of the returned phi (defaults to M).

    Returns
    -------
    phi : (m_full, n_out) float64.
    """
    s, m = masks.shape
    n_out = ey_adj.shape[1]
    if nonzero_inds is None:
        nonzero_inds = np.arange(m)
    if m_full is None:
        m_full = m
    phi = np.zeros((m_full, n_out), dtype=np.float64)
    if len(nonzero_inds) == 0:
        return phi
    if len(nonzero_inds) == 1:
        phi[nonzero_inds[0]] = total
        return phi

    z = masks[:, nonzero_inds].astype(np.float64)
    last = z[:, -1]
    # eyAdj2 = eyAdj - mask_last * total      (per output)
    ey2 = ey_adj - last[:, None] * total[None, :]
    # etmp = Z[:, :-1] - Z[:, -1:]
    etmp = z[:, :-1] - last[:, None]
    wz = etmp * kernel_weights[:, None]
    a = wz.T @ etmp  # (M-1, M-1), shared across outputs
    r = wz.T @ ey2  # (M-1, n_out)
    try:
        w = np.linalg.solve(a, r)
    except np.linalg.LinAlgError:
        w, *_ = np.linalg.lstsq(a, r, rcond=None)
    phi[nonzero_inds[:-1]] = w
    phi[nonzero_inds[-1]] = total - w.sum(axis=0)
    return phi
