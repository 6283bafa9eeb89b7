"""Shared utilities (native counterpart of reference ``explainers/utils.py``)."""
from __future__ import annotations

import functools
import os
from typing import Optional

import numpy as np

__all__ = ["Bunch", "batch", "get_filename", "methdispatch"]


class Bunch(dict):
    """Dict with attribute access (reference utils.py:22-40)."""

    def __init__(self, **kwargs):
        super().__init__(kwargs)

    def __getattr__(self, key):
        try:
            return self[key]
        except KeyError:
            raise AttributeError(key) from None

    def __setattr__(self, key, value):
        self[key] = value

    def __dir__(self):
        return list(self.keys())


def batch(X: np.ndarray, batch_size: Optional[int] = None, n_batches: int = 4):
    """Split X into minibatches (reference utils.py:89-121 semantics).

    If ``batch_size`` is given, emit ceil(n/batch_size) batches of at most
    ``batch_size`` rows; otherwise split as evenly as possible into
    ``n_batches`` parts.
    """
    n = X.shape[0]
    if batch_size is not None and batch_size > 0:
        n_batches = (n + batch_size - 1) // batch_size
        cuts = [batch_size * i for i in range(1, n_batches)]
        return np.array_split(X, cuts)
    return np.array_split(X, n_batches)


def get_filename(
    workers: int,
    batch_size: int,
    actor_cpu_fraction: float = 1.0,
    serve: bool = False,
    max_batch_size: Optional[int] = None,
    results_dir: str = "results",
) -> str:
    """Result file naming compatible with the reference's scheme
    (utils.py:67-86; parsed back by Analysis.ipynb cell 2)."""
    os.makedirs(results_dir, exist_ok=True)
    if serve:
        return os.path.join(
            results_dir,
            f"ray_replicas_{workers}_maxbatch_{max_batch_size}.pkl",
        )
    return os.path.join(
        results_dir,
        f"ray_workers_{workers}_bsize_{batch_size}_actorfr_{actor_cpu_fraction}.pkl",
    )


def methdispatch(func):
    """singledispatch on a method's first non-self argument (utils.py:43-64)."""
    dispatcher = functools.singledispatch(func)

    @functools.wraps(func)
    def wrapper(*args, **kw):
        return dispatcher.dispatch(args[1].__class__)(*args, **kw)

    wrapper.register = dispatcher.register
    return wrapper
