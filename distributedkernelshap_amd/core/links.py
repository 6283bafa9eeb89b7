"""Link functions (identity / logit) as used by the reference.

Reference: link handling configured at ``benchmarks/ray_pool.py:36`` and
consumed via ``shap.common.convert_to_link`` at
``explainers/kernel_shap.py:14,229,949``.
"""
from __future__ import annotations

import numpy as np

_EPS = 1e-15


def identity(x):
    return x


def identity_inv(x):
    return x


def logit(x):
    x = np.clip(x, _EPS, 1.0 - _EPS)
    return np.log(x / (1.0 - x))


def logit_inv(x):
    return 1.0 / (1.0 + np.exp(-x))


_LINKS = {
    "identity": (identity, identity_inv),
    "logit": (logit, logit_inv),
}


def convert_to_link(name: str):
    """Return (f, finv) for a link name; mirrors shap's convert_to_link."""
    if callable(name):
        return name, None
    try:
        return _LINKS[name]
    except KeyError:
        raise ValueError(
            f"Unknown link '{name}'; expected one of {sorted(_LINKS)}"
        ) from None
