"""KernelShap public API tests (reference C2 parity)."""
import numpy as np
import pytest

from distributedkernelshap_amd import KernelShap
from distributedkernelshap_amd.models import LinearPredictor, make_adult_like


@pytest.fixture(scope="module")
def fitted(request):
    data = make_adult_like(n_instances=6, n_background=30, seed=1)
    pred = LinearPredictor.random(data.X.shape[1], 2, seed=1)
    ks = KernelShap(pred, link="logit", feature_names=data.group_names, device="cpu")
    ks.fit(data.background, groups=data.groups, group_names=data.group_names)
    return ks, data, pred


def test_fit_metadata(fitted):
    ks, data, _ = fitted
    assert ks.meta["name"] == "KernelShap"
    assert ks.meta["params"]["grouped"] is True
    assert ks.meta["params"]["link"] == "logit"
    assert ks.expected_value.shape == (2,)


def test_explain_output_layout(fitted):
    ks, data, pred = fitted
    exp = ks.explain(data.X[:3])
    assert len(exp.shap_values) == 2
    assert exp.shap_values[0].shape == (3, 12)
    assert exp.data["link"] == "logit"
    assert exp.feature_names == data.group_names
    raw = exp.data["raw"]
    assert raw["raw_prediction"].shape == (3, 2)
    assert raw["prediction"].shape == (3,)
    assert raw["instances"].shape == data.X[:3].shape
    assert "aggregated" in raw["importances"]
    # json round-trip of the full explanation
    from distributedkernelshap_amd.interface import Explanation

    exp2 = Explanation.from_json(exp.to_json())
    assert np.allclose(exp2.shap_values[0], exp.shap_values[0])


def test_explain_unfitted_raises():
    pred = LinearPredictor.random(5, 2)
    ks = KernelShap(pred, device="cpu")
    with pytest.raises(TypeError):
        ks.explain(np.zeros((1, 5)))


def test_groups_validation():
    pred = LinearPredictor.random(5, 2)
    ks = KernelShap(pred, device="cpu")
    with pytest.raises(ValueError):
        ks.fit(np.zeros((4, 5)), groups=[[0, 1], [2]])  # misses columns 3,4


def test_background_summarisation_subsample():
    d = 6
    data = np.random.default_rng(0).normal(size=(500, d))
    pred = LinearPredictor.random(d, 2)
    ks = KernelShap(pred, device="cpu")
    ks.fit(data, summarise_background=True, n_background_samples=50)
    assert ks.background_data.shape == (50, d)
    assert ks.meta["params"]["summarise_background"] is True


def test_background_summarisation_kmeans():
    d = 4
    data = np.random.default_rng(0).normal(size=(400, d))
    pred = LinearPredictor.random(d, 2)
    ks = KernelShap(pred, device="cpu")
    ks.fit(data, summarise_background="kmeans", n_background_samples=10)
    assert ks.background_data.shape == (10, d)
    assert ks.bg_weights is not None and ks.bg_weights.shape == (10,)
    # centroids snapped to actually-occurring values per column
    for j in range(d):
        assert np.all(np.isin(ks.background_data[:, j], data[:, j]))


def test_summarise_result(fitted):
    ks, data, _ = fitted
    # one-hot groups start after 4 numeric cols; widths from the generator
    widths = [7, 8, 6, 5, 4, 2, 6, 8]
    starts = list(np.cumsum([4] + widths[:-1]))
    # explain in ungrouped space: fit a second explainer without groups
    pred = ks.predictor
    ks2 = KernelShap(pred, link="logit", device="cpu")
    ks2.fit(data.background)
    exp = ks2.explain(
        data.X[:2],
        summarise_result=True,
        cat_vars_start_idx=starts,
        cat_vars_enc_dim=widths,
    )
    assert exp.shap_values[0].shape == (2, 12)


def test_weights_validation():
    pred = LinearPredictor.random(4, 2)
    ks = KernelShap(pred, device="cpu")
    with pytest.raises(ValueError):
        ks.fit(np.zeros((5, 4)), weights=np.ones(3))


def test_regression_task_prediction():
    from distributedkernelshap_amd.models import LinearPredictor

    rng = np.random.default_rng(0)
    d = 6
    pred = LinearPredictor(rng.normal(size=(1, d)), np.zeros(1), activation="none")
    bg = rng.normal(size=(20, d))
    ks = KernelShap(pred, link="identity", task="regression", device="cpu")
    ks.fit(bg)
    exp = ks.explain(rng.normal(size=(3, d)))
    raw = exp.data["raw"]
    # regression keeps raw predictions, no argmax/thresholding
    assert raw["prediction"].shape == (3, 1)
    assert np.allclose(raw["prediction"].ravel(), raw["raw_prediction"].ravel())
