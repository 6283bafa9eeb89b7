"""Interface / Explanation / helper tests (reference C1 parity)."""
import numpy as np
import pytest

from distributedkernelshap_amd.interface import Explanation, NumpyEncoder
from distributedkernelshap_amd.explainers.kernel_shap import (
    rank_by_importance,
    sum_categories,
)


def test_explanation_roundtrip():
    meta = {"name": "KernelShap", "params": {"link": "logit"}}
    data = {
        "shap_values": [np.arange(6, dtype=float).reshape(2, 3)],
        "expected_value": np.array([0.1, 0.9]),
        "link": "logit",
        "feature_names": ["a", "b", "c"],
        "raw": {"raw_prediction": np.ones((2, 2)), "prediction": np.array([1, 0]),
                "instances": np.zeros((2, 3)), "importances": {}},
    }
    exp = Explanation(meta, data)
    exp2 = Explanation.from_json(exp.to_json())
    assert exp2.meta["name"] == "KernelShap"
    assert np.allclose(exp2.shap_values[0], data["shap_values"][0])
    assert np.allclose(exp2.expected_value, data["expected_value"])
    assert exp2.feature_names == ["a", "b", "c"]


def test_explanation_attribute_access():
    exp = Explanation({"name": "x"}, {"link": "identity"})
    assert exp.link == "identity"
    assert exp.name == "x"
    with pytest.raises(AttributeError):
        exp.nope
    with pytest.warns(DeprecationWarning):
        assert exp["link"] == "identity"


def test_rank_by_importance():
    sv = [np.array([[1.0, -3.0, 2.0], [1.0, -3.0, 2.0]]),
          np.array([[0.5, 0.1, -4.0], [0.5, 0.1, -4.0]])]
    imp = rank_by_importance(sv, feature_names=["a", "b", "c"])
    assert imp["0"]["names"] == ["b", "c", "a"]
    assert imp["1"]["names"] == ["c", "a", "b"]
    assert imp["aggregated"]["names"] == ["c", "b", "a"]
    assert np.isclose(imp["aggregated"]["ranked_effect"][0], 6.0)


def test_sum_categories_2d():
    v = np.arange(12, dtype=float).reshape(2, 6)
    # columns: [num, cat(3 cols), num, num] -> starts [1], dims [3]
    out = sum_categories(v, [1], [3])
    assert out.shape == (2, 4)
    assert np.allclose(out[:, 1], v[:, 1:4].sum(axis=1))
    assert np.allclose(out[:, 0], v[:, 0])
    assert np.allclose(out[:, 2], v[:, 4])


def test_sum_categories_3d():
    v = np.arange(2 * 5 * 5, dtype=float).reshape(2, 5, 5)
    out = sum_categories(v, [0], [2])
    assert out.shape == (2, 4, 4)
    assert np.allclose(out[:, 0, 0], v[:, :2, :2].sum(axis=(1, 2)))


def test_sum_categories_validation():
    with pytest.raises(ValueError):
        sum_categories(np.zeros((2, 4)), [0], [2, 2])
    with pytest.raises(ValueError):
        sum_categories(np.zeros((2, 4)), None, None)
