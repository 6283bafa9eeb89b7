"""End-to-end engine tests against analytic oracles (SURVEY.md §4 rebuild
test strategy: linear closed form + local-accuracy property)."""
import numpy as np
import pytest

from distributedkernelshap_amd.core.engine import KernelShapEngine
from distributedkernelshap_amd.models import LinearPredictor, make_adult_like


def test_linear_closed_form(rng):
    """For a linear model (identity link), KernelSHAP has the closed form
    phi_g = sum_{j in g} w_j * (x_j - E_bg[x_j])."""
    d, n_bg, b = 10, 30, 5
    w = rng.normal(size=(1, d))
    pred = LinearPredictor(w, np.zeros(1), activation="none")
    bg = rng.normal(size=(n_bg, d))
    X = rng.normal(size=(b, d))
    eng = KernelShapEngine(pred, bg, link="identity", seed=0, device="cpu")
    sv = eng.shap_values(X)  # full enumeration (m=10 <= cap)
    expect = (X - bg.mean(axis=0)) * w[0]
    assert np.allclose(sv[0], expect, atol=1e-7)


def test_linear_closed_form_grouped(rng):
    d = 12
    groups = [[0, 1], [2], [3, 4, 5], [6], [7, 8], [9, 10, 11]]
    w = rng.normal(size=(1, d))
    pred = LinearPredictor(w, np.zeros(1), activation="none")
    bg = rng.normal(size=(40, d))
    X = rng.normal(size=(3, d))
    eng = KernelShapEngine(pred, bg, groups=groups, link="identity", device="cpu")
    sv = eng.shap_values(X)
    diff = (X - bg.mean(axis=0)) * w[0]
    expect = np.stack([diff[:, g].sum(axis=1) for g in groups], axis=1)
    assert np.allclose(sv[0], expect, atol=1e-7)


def test_local_accuracy_sampled(adult_like, linear_predictor):
    """Sum of phi equals link(f(x)) - link(fnull) exactly (constrained solve),
    even in the sampled (non-enumerated) regime."""
    from distributedkernelshap_amd.core.links import logit

    eng = KernelShapEngine(
        linear_predictor,
        adult_like.background,
        groups=adult_like.groups,
        link="logit",
        seed=0,
        device="cpu",
    )
    X = adult_like.X[:4]
    sv = eng.shap_values(X)
    fx = logit(linear_predictor(X))
    for o in range(2):
        total = sv[o].sum(axis=1) + eng.expected_value[o]
        assert np.allclose(total, fx[:, o], atol=1e-10)


def test_weighted_background(rng):
    """Non-uniform background weights shift fnull and phi (kmeans-weights path)."""
    d = 6
    w = rng.normal(size=(1, d))
    pred = LinearPredictor(w, np.zeros(1), activation="none")
    bg = rng.normal(size=(20, d))
    bw = rng.random(20)
    eng = KernelShapEngine(pred, bg, bg_weights=bw, link="identity", device="cpu")
    X = rng.normal(size=(2, d))
    sv = eng.shap_values(X)
    mu = (bw / bw.sum()) @ bg
    expect = (X - mu) * w[0]
    assert np.allclose(sv[0], expect, atol=1e-7)


def test_no_varying_groups(rng):
    d = 5
    pred = LinearPredictor(rng.normal(size=(1, d)), np.zeros(1), activation="none")
    bg = np.tile(rng.normal(size=(1, d)), (10, 1))
    eng = KernelShapEngine(pred, bg, link="identity", device="cpu")
    sv = eng.shap_values(bg[:1])  # x == background everywhere -> phi = 0
    assert np.allclose(sv[0], 0.0)


def test_single_varying_group(rng):
    d = 4
    w = rng.normal(size=(1, d))
    pred = LinearPredictor(w, np.zeros(1), activation="none")
    base = rng.normal(size=(1, d))
    bg = np.tile(base, (10, 1))
    x = base.copy()
    x[0, 2] += 3.0  # only feature 2 varies
    eng = KernelShapEngine(pred, bg, link="identity", device="cpu")
    sv = eng.shap_values(x)
    expect = np.zeros(d)
    expect[2] = 3.0 * w[0, 2]
    assert np.allclose(sv[0][0], expect, atol=1e-10)


def test_instance_offset_reproducibility(adult_like, linear_predictor):
    """Sharded computation (instance_offset) reproduces the full batch —
    the invariant the collective DP path relies on."""
    eng = KernelShapEngine(
        linear_predictor,
        adult_like.background,
        groups=adult_like.groups,
        link="logit",
        seed=0,
        device="cpu",
    )
    X = adult_like.X[:6]
    full = eng.shap_values(X)
    part1 = eng.shap_values(X[:3], instance_offset=0)
    part2 = eng.shap_values(X[3:], instance_offset=3)
    # masks are bitwise identical; the only allowed divergence is BLAS
    # blocking on different predict batch shapes (last-ulp rounding)
    for o in range(2):
        assert np.allclose(
            full[o], np.concatenate([part1[o], part2[o]]), rtol=0, atol=1e-10
        )


def test_nsamples_kwarg(adult_like, linear_predictor):
    eng = KernelShapEngine(
        linear_predictor,
        adult_like.background,
        groups=adult_like.groups,
        link="logit",
        device="cpu",
    )
    sv = eng.shap_values(adult_like.X[:2], nsamples=500)
    assert sv[0].shape == (2, 12)


def test_l1_reg_num_features(adult_like, linear_predictor):
    eng = KernelShapEngine(
        linear_predictor,
        adult_like.background,
        groups=adult_like.groups,
        link="logit",
        device="cpu",
    )
    sv = eng.shap_values(adult_like.X[:2], l1_reg="num_features(5)")
    # at most 5 nonzero features per instance (plus constraint back-substitution)
    assert (np.abs(sv[0]) > 1e-12).sum(axis=1).max() <= 6


def test_single_instance_and_1d_input(adult_like, linear_predictor):
    eng = KernelShapEngine(
        linear_predictor, adult_like.background, groups=adult_like.groups,
        link="logit", device="cpu",
    )
    sv = eng.shap_values(adult_like.X[0])  # 1-D input
    assert sv[0].shape == (1, 12)


def test_linear_fast_path_matches_general(rng):
    """The algebraic linear CPU path must agree with the generic synth path
    (same predictor wrapped as a plain callable loses linear_params)."""
    d = 8
    w = rng.normal(size=(2, d))
    b = rng.normal(size=2)
    pred = LinearPredictor(w, b, activation="softmax")
    plain = lambda X: pred(X)  # noqa: E731  (drops linear_params)
    bg = rng.normal(size=(20, d))
    X = rng.normal(size=(3, d))
    fast = KernelShapEngine(pred, bg, link="logit", seed=0, device="cpu")
    slow = KernelShapEngine(plain, bg, link="logit", seed=0, device="cpu")
    svf = fast.shap_values(X)
    svs = slow.shap_values(X)
    for o in range(2):
        assert np.allclose(svf[o], svs[o], atol=1e-10)


def test_sigmoid_single_output_predictor(rng):
    """n_out=1 sigmoid predictor (binary LR parameterised as one logit)."""
    from distributedkernelshap_amd.core.links import logit

    d = 7
    w = rng.normal(size=(1, d))
    pred = LinearPredictor(w, np.zeros(1), activation="sigmoid")
    bg = rng.normal(size=(25, d))
    X = rng.normal(size=(3, d))
    eng = KernelShapEngine(pred, bg, link="logit", seed=0, device="cpu")
    sv = eng.shap_values(X)
    assert len(sv) == 1 and sv[0].shape == (3, d)
    total = sv[0].sum(axis=1) + eng.expected_value[0]
    assert np.allclose(total, logit(pred(X))[:, 0], atol=1e-9)
    # with a SINGLE background row, logit(sigmoid(z)) == z exactly, so the
    # game is additive and the linear closed form holds (a multi-row
    # background is NOT linear: logit of the mean of sigmoids)
    bg1 = rng.normal(size=(1, d))
    eng1 = KernelShapEngine(pred, bg1, link="logit", seed=0, device="cpu")
    sv1 = eng1.shap_values(X)
    expect = (X - bg1[0]) * w[0]
    assert np.allclose(sv1[0], expect, atol=1e-6)


def test_shap_values_as_tensor_cpu(rng):
    """as_tensor=True returns the (B, n_groups, n_out) fp64 torch tensor the
    distributed gather consumes, matching the per-class list bitwise."""
    import torch

    d = 6
    pred = LinearPredictor.random(d, 2, seed=3)
    bg = rng.normal(size=(15, d))
    X = rng.normal(size=(4, d))
    eng = KernelShapEngine(pred, bg, link="logit", seed=0, device="cpu")
    sv = eng.shap_values(X)
    t = eng.shap_values(X, as_tensor=True)
    assert torch.is_tensor(t) and t.dtype == torch.float64
    assert tuple(t.shape) == (4, d, 2)
    for o in range(2):
        assert np.array_equal(t[:, :, o].numpy(), sv[o])
