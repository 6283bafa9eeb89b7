"""Explainer base interfaces and the Explanation result object.

Native re-design of the reference's ``explainers/interface.py`` (C1 in
SURVEY.md §2.1): same public surface — ``Explainer``/``FitMixin`` ABCs, an
``Explanation`` whose meta/data keys are attribute-accessible, JSON
round-tripping with numpy coercion — built on plain dataclasses instead of
``attr``.
"""
from __future__ import annotations

import abc
import json
from typing import Any, Dict

import numpy as np

__all__ = [
    "DEFAULT_META_KERNEL_SHAP",
    "DEFAULT_DATA_KERNEL_SHAP",
    "Explainer",
    "FitMixin",
    "Explanation",
    "NumpyEncoder",
]

# schemas mirroring reference explainers/interface.py:14-37
DEFAULT_META_KERNEL_SHAP: Dict[str, Any] = {
    "name": None,
    "type": ["blackbox"],
    "task": None,
    "explanations": ["local"],
    "params": {},
}

DEFAULT_DATA_KERNEL_SHAP: Dict[str, Any] = {
    "shap_values": [],
    "expected_value": [],
    "link": "identity",
    "categorical_names": {},
    "feature_names": [],
    "raw": {
        "raw_prediction": [],
        "prediction": [],
        "instances": [],
        "importances": {},
    },
}


class NumpyEncoder(json.JSONEncoder):
    """JSON encoder coercing numpy scalars/arrays (reference interface.py:140-163)."""

    def default(self, obj):  # noqa: D102
        if isinstance(
            obj,
            (
                np.int_,
                np.intc,
                np.intp,
                np.int8,
                np.int16,
                np.int32,
                np.int64,
                np.uint8,
                np.uint16,
                np.uint32,
                np.uint64,
            ),
        ):
            return int(obj)
        if isinstance(obj, (np.float16, np.float32, np.float64)):
            return float(obj)
        if isinstance(obj, np.bool_):
            return bool(obj)
        if isinstance(obj, np.ndarray):
            return obj.tolist()
        return super().default(obj)


class Explainer(abc.ABC):
    """Base class for explainers (reference interface.py:54-72)."""

    def __init__(self, meta: Dict[str, Any] = None):
        self.meta = dict(meta) if meta is not None else {}

    @abc.abstractmethod
    def explain(self, X: Any, **kwargs) -> "Explanation":
        ...


class FitMixin(abc.ABC):
    """Mixin for explainers that require a fit step (reference interface.py:75-78)."""

    @abc.abstractmethod
    def fit(self, background_data: Any, **kwargs) -> "Explainer":
        ...


class Explanation:
    """Explanation result: ``meta`` and ``data`` dicts, keys readable as
    attributes; JSON (de)serialisable (reference interface.py:81-137).
    """

    def __init__(self, meta: Dict[str, Any], data: Dict[str, Any]):
        self.meta = meta
        self.data = data

    def __getattr__(self, item):
        # only called when normal lookup fails
        meta = self.__dict__.get("meta", {})
        data = self.__dict__.get("data", {})
        if item in data:
            return data[item]
        if item in meta:
            return meta[item]
        raise AttributeError(item)

    def __getitem__(self, item):
        # deprecated dict-style access kept for parity (interface.py:128-137)
        import warnings

        warnings.warn(
            "Explanation['key'] is deprecated; use attribute access",
            DeprecationWarning,
            stacklevel=2,
        )
        if item in self.data:
            return self.data[item]
        return self.meta[item]

    def __repr__(self):
        return f"Explanation(meta={self.meta!r})"

    def to_json(self) -> str:
        return json.dumps({"meta": self.meta, "data": self.data}, cls=NumpyEncoder)

    @classmethod
    def from_json(cls, jsonrepr: str) -> "Explanation":
        obj = json.loads(jsonrepr)
        meta, data = obj.get("meta", {}), obj.get("data", {})
        # re-materialise the well-known array fields
        if data.get("shap_values"):
            data["shap_values"] = [np.asarray(a) for a in data["shap_values"]]
        if data.get("expected_value") is not None:
            data["expected_value"] = np.asarray(data["expected_value"])
        raw = data.get("raw")
        if isinstance(raw, dict):
            for k in ("raw_prediction", "prediction", "instances"):
                if raw.get(k) is not None:
                    raw[k] = np.asarray(raw[k])
        return cls(meta, data)
