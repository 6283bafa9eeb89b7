"""Model/data generator tests (CPU)."""
import numpy as np

from distributedkernelshap_amd.models import (
    LinearPredictor,
    make_adult_like,
    make_tabular,
)
from distributedkernelshap_amd.models.resnet import make_superpixel_problem


def test_adult_like_shapes():
    d = make_adult_like(n_instances=10, n_background=5, seed=0)
    assert d.X.shape == (10, 50)
    assert d.background.shape == (5, 50)
    assert len(d.groups) == 12
    assert sorted(c for g in d.groups for c in g) == list(range(50))
    # one-hot blocks: at most one hot column per categorical group
    for g in d.groups[4:]:
        assert d.X[:, g].sum(axis=1).max() <= 1.0


def test_adult_like_deterministic():
    a = make_adult_like(n_instances=4, n_background=3, seed=5)
    b = make_adult_like(n_instances=4, n_background=3, seed=5)
    assert np.array_equal(a.X, b.X)
    c = make_adult_like(n_instances=4, n_background=3, seed=6)
    assert not np.array_equal(a.X, c.X)


def test_tabular_groups_are_singletons():
    d = make_tabular(n_features=7, n_instances=3, n_background=2)
    assert d.groups == [[j] for j in range(7)]


def test_linear_predictor_softmax_rows_sum_to_one(rng):
    p = LinearPredictor.random(6, 3, seed=0)
    out = p(rng.normal(size=(5, 6)))
    assert np.allclose(out.sum(axis=1), 1.0)
    w, b, act = p.linear_params()
    assert w.shape == (3, 6) and act == "softmax"


def test_superpixel_problem_structure():
    X, bg, groups, names = make_superpixel_problem(n_instances=2, hw=64, patch=32)
    assert X.shape == (2, 3 * 64 * 64)
    assert bg.shape == (1, 3 * 64 * 64)
    assert len(groups) == 4 and len(names) == 4
    cols = sorted(c for g in groups for c in g)
    assert cols == list(range(3 * 64 * 64))


def test_resnet18_forward_flat_rows():
    import torch

    from distributedkernelshap_amd.models.resnet import resnet18

    m = resnet18(num_classes=5, seed=0)
    with torch.no_grad():
        out = m(torch.rand(2, 3 * 64 * 64))
    assert out.shape == (2, 5)
    assert torch.allclose(out.sum(dim=1), torch.ones(2), atol=1e-5)
