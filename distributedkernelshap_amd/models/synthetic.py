"""Synthetic dataset generators (no network: datasets are generated, not
downloaded — replaces the reference's GCS download-and-cache,
``explainers/utils.py:14-19,160-188`` and the Adult pipeline
``scripts/process_adult_data.py``).

``make_adult_like`` reproduces the Adult benchmark *shape*: 12 feature groups
(4 numeric singletons + 8 one-hot categorical blocks, drop-first encoding,
``scripts/process_adult_data.py:180-217``), 100 background rows, 2,560 test
instances.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List

import numpy as np

__all__ = ["SyntheticData", "make_adult_like", "make_tabular"]


@dataclass
class SyntheticData:
    X: np.ndarray  # (n_instances, D)
    background: np.ndarray  # (n_background, D)
    groups: List[List[int]]
    group_names: List[str]
    feature_names: List[str]
    category_map: dict = field(default_factory=dict)


def _build(
    rng: np.random.Generator,
    n_numeric: int,
    cat_widths: List[int],
    n_instances: int,
    n_background: int,
) -> SyntheticData:
    d = n_numeric + sum(cat_widths)
    groups: List[List[int]] = [[j] for j in range(n_numeric)]
    names = [f"num_{j}" for j in range(n_numeric)]
    feature_names = list(names)
    col = n_numeric
    category_map = {}
    for ci, w in enumerate(cat_widths):
        groups.append(list(range(col, col + w)))
        names.append(f"cat_{ci}")
        feature_names += [f"cat_{ci}_{l}" for l in range(w)]
        category_map[n_numeric + ci] = [f"lvl_{l}" for l in range(w + 1)]
        col += w

    def sample(n: int) -> np.ndarray:
        out = np.zeros((n, d), dtype=np.float64)
        out[:, :n_numeric] = rng.normal(0.0, 1.0, size=(n, n_numeric))
        c = n_numeric
        for w in cat_widths:
            # drop-first one-hot: level 0 encodes to all-zeros
            lvl = rng.integers(0, w + 1, size=n)
            hot = lvl > 0
            out[np.nonzero(hot)[0], c + lvl[hot] - 1] = 1.0
            c += w
        return out

    return SyntheticData(
        X=sample(n_instances),
        background=sample(n_background),
        groups=groups,
        group_names=names,
        feature_names=feature_names,
        category_map=category_map,
    )


def make_adult_like(
    n_instances: int = 2560, n_background: int = 100, seed: int = 0
) -> SyntheticData:
    """Adult-shaped synthetic data: 12 groups, D=50 one-hot columns."""
    rng = np.random.Generator(np.random.Philox(key=[seed, 0xADA17]))
    cat_widths = [7, 8, 6, 5, 4, 2, 6, 8]  # 46 one-hot cols + 4 numeric = 50
    return _build(rng, 4, cat_widths, n_instances, n_background)


def make_tabular(
    n_features: int,
    n_instances: int,
    n_background: int,
    seed: int = 0,
) -> SyntheticData:
    """Plain continuous tabular data, one group per feature (stress configs)."""
    rng = np.random.Generator(np.random.Philox(key=[seed, 0x7AB]))
    return _build(rng, n_features, [], n_instances, n_background)
