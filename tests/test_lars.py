"""Batched LARS pre-selection (core.lars) vs the sklearn calls it replaces:
the GPU engine's l1_reg path must select the SAME features as the CPU
oracle's per-instance sklearn fits (engine._l1_select)."""
import numpy as np
import pytest
import torch
from sklearn import linear_model

from distributedkernelshap_amd.core.lars import batched_lars_select


def _prep(Z, y, w=None):
    s = Z.shape[0]
    if w is None:
        w = np.ones(s)
    sw = np.sqrt(w)
    zw = Z * sw[:, None]
    yw = y * sw
    G = torch.tensor(zw.T @ zw)[None]
    c = torch.tensor(zw.T @ yw)[None]
    yty = torch.tensor([(yw * yw).sum()])
    zbar = torch.tensor(zw.mean(0))[None]
    ybar = torch.tensor([yw.mean()])
    return zw, yw, G, c, yty, zbar, ybar


@pytest.mark.parametrize("crit", ["aic", "bic"])
def test_ic_selection_matches_sklearn(crit):
    rng = np.random.default_rng(3)
    s, m = 600, 12
    for trial in range(6):
        Z = (rng.random((s, m)) < 0.5).astype(float)
        w = 10.0 ** (-2 * rng.random(s))
        bt = np.zeros(m)
        bt[rng.choice(m, 4, replace=False)] = rng.normal(size=4)
        y = Z @ bt + 0.5 * rng.normal(size=s) + (3.0 if trial % 2 else 0.0)
        zw, yw, G, c, yty, zbar, ybar = _prep(Z, y, w)
        sup = batched_lars_select(
            G, c, yty, n_samples=s, mode=crit, zbar=zbar, ybar=ybar
        )[0].numpy()
        skm = linear_model.LassoLarsIC(criterion=crit)
        skm.fit(zw, yw)
        assert np.array_equal(sup, np.abs(skm.coef_) > 0), trial


def test_num_features_matches_lars_path():
    rng = np.random.default_rng(4)
    s, m = 500, 14
    for trial in range(4):
        Z = (rng.random((s, m)) < 0.5).astype(float)
        y = rng.normal(size=s)
        _, _, G, c, yty, _, _ = _prep(Z, y)
        sup = batched_lars_select(
            G, c, yty, n_samples=s, mode="num_features", num_features=5
        )[0].numpy()
        coefs = linear_model.lars_path(Z, y, max_iter=5)[2]
        assert np.array_equal(sup, np.abs(coefs[:, -1]) > 0), trial


@pytest.mark.parametrize("alpha", [0.01, 0.05, 0.2])
def test_alpha_matches_lasso(alpha):
    rng = np.random.default_rng(5)
    s, m = 600, 12
    Z = (rng.random((s, m)) < 0.5).astype(float)
    y = (Z @ (rng.normal(size=m) * (rng.random(m) < 0.4))
         + 0.3 * rng.normal(size=s) + 1.0)
    _, _, G, c, yty, zbar, ybar = _prep(Z, y)
    sup = batched_lars_select(
        G, c, yty, n_samples=s, mode="alpha", alpha=alpha,
        zbar=zbar, ybar=ybar,
    )[0].numpy()
    las = linear_model.Lasso(alpha=alpha)
    las.fit(Z, y)
    assert np.array_equal(sup, np.abs(las.coef_) > 1e-9)


def test_batched_heterogeneous_instances():
    """One batched call over instances with different supports/offsets gives
    per-instance answers identical to solo calls."""
    rng = np.random.default_rng(6)
    s, m, b = 500, 10, 5
    Gs, cs, ys, zb, yb, solo = [], [], [], [], [], []
    for i in range(b):
        Z = (rng.random((s, m)) < 0.5).astype(float)
        bt = np.zeros(m)
        bt[rng.choice(m, 1 + i % 4, replace=False)] = rng.normal(size=1 + i % 4)
        y = Z @ bt + 0.3 * rng.normal(size=s) + i
        _, _, G, c, yty, zbar, ybar = _prep(Z, y)
        Gs.append(G); cs.append(c); ys.append(yty); zb.append(zbar); yb.append(ybar)
        solo.append(batched_lars_select(
            G, c, yty, n_samples=s, mode="aic", zbar=zbar, ybar=ybar
        )[0].numpy())
    sup = batched_lars_select(
        torch.cat(Gs), torch.cat(cs), torch.cat(ys), n_samples=s,
        mode="aic", zbar=torch.cat(zb), ybar=torch.cat(yb),
    ).numpy()
    for i in range(b):
        assert np.array_equal(sup[i], solo[i]), i
