"""WLS solver tests (CPU oracle of HIP kernel K7)."""
import numpy as np
import pytest

from distributedkernelshap_amd.core.sampler import plan_coalitions, sample_masks
from distributedkernelshap_amd.core.solver import solve_wls


def _random_problem(rng, m=8, s=200, n_out=2):
    masks = (rng.random((s, m)) > 0.5).astype(np.uint8)
    # avoid all-zero / all-one rows (never produced by the sampler)
    masks[masks.sum(axis=1) == 0, 0] = 1
    masks[masks.sum(axis=1) == m, 0] = 0
    kw = rng.random(s) + 0.1
    ey = rng.normal(size=(s, n_out))
    total = rng.normal(size=(n_out,))
    return masks, kw, ey, total


def test_matches_reference_normal_equations(rng):
    masks, kw, ey, total = _random_problem(rng)
    phi = solve_wls(masks, kw, ey, total)
    # independent reference: weighted lstsq on the eliminated system
    z = masks.astype(float)
    ey2 = ey - z[:, -1:] * total[None, :]
    etmp = z[:, :-1] - z[:, -1:]
    sw = np.sqrt(kw)
    w_ref, *_ = np.linalg.lstsq(etmp * sw[:, None], ey2 * sw[:, None], rcond=None)
    assert np.allclose(phi[:-1], w_ref, atol=1e-8)
    assert np.allclose(phi[-1], total - w_ref.sum(axis=0), atol=1e-8)


def test_constraint_exact(rng):
    masks, kw, ey, total = _random_problem(rng, m=12)
    phi = solve_wls(masks, kw, ey, total)
    assert np.allclose(phi.sum(axis=0), total)


def test_exact_shapley_linear_game():
    """Full enumeration + exact kernel weights recover the analytic Shapley
    values of an additive (linear) game exactly."""
    m = 8
    plan = plan_coalitions(m)  # fully enumerated for m=8
    masks, kw = sample_masks(plan, seed=0, instance_index=0)
    contrib = np.arange(1, m + 1, dtype=float)  # v(S) = sum_{i in S} c_i
    ey = (masks.astype(float) @ contrib).reshape(-1, 1)
    total = np.array([contrib.sum()])
    phi = solve_wls(masks, kw, ey, total)
    assert np.allclose(phi[:, 0], contrib, atol=1e-6)


def test_nonzero_inds_subset(rng):
    masks, kw, ey, total = _random_problem(rng, m=10)
    keep = np.array([1, 3, 4, 7, 9])
    phi = solve_wls(masks, kw, ey, total, nonzero_inds=keep)
    # excluded features get exactly zero; constraint still holds
    mask = np.ones(10, dtype=bool)
    mask[keep] = False
    assert np.all(phi[mask] == 0)
    assert np.allclose(phi.sum(axis=0), total)


def test_single_feature():
    phi = solve_wls(
        np.array([[1]], dtype=np.uint8),
        np.array([1.0]),
        np.array([[0.5]]),
        np.array([0.7]),
    )
    assert np.allclose(phi, [[0.7]])


def test_singular_gram_falls_back_to_lstsq():
    """Duplicate mask columns make the Gram singular; the solver must not
    raise (np.linalg fallback, mirroring shap's failure modes)."""
    s, m = 50, 4
    rng = np.random.default_rng(0)
    masks = (rng.random((s, m)) > 0.5).astype(np.uint8)
    masks[:, 1] = masks[:, 0]  # identical columns -> singular normal matrix
    kw = np.ones(s)
    ey = rng.normal(size=(s, 1))
    total = np.array([1.0])
    phi = solve_wls(masks, kw, ey, total)
    assert np.all(np.isfinite(phi))
    assert np.allclose(phi.sum(axis=0), total)


def test_plan_rejects_m_below_two():
    with pytest.raises(ValueError):
        plan_coalitions(1)
