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


# --------------------------------------------------------------------- #
# dataset / model persistence (reference utils.py:124-188 downloaded from
# GCS; here datasets are synthetic-generated and cached locally — no network)

ASSETS_DIR = os.environ.get("KSHAP_ASSETS_DIR", "assets")


def load_data(assets_dir: str = None):
    """Return the benchmark dataset Bunch (X_train, y_train, X_test,
    background, groups, group_names), generating + caching it on first use
    (reference ``load_data``, utils.py:160-188)."""
    import pickle

    assets_dir = assets_dir or ASSETS_DIR
    path = os.path.join(assets_dir, "data.pkl")
    if os.path.exists(path):
        with open(path, "rb") as f:
            return pickle.load(f)
    from ..models.synthetic import make_adult_like

    data = make_adult_like(n_instances=2560, n_background=100, seed=0)
    rng = np.random.Generator(np.random.Philox(key=[0, 0xDA7A]))
    n_train = 30000
    d = data.X.shape[1]
    X_train = np.concatenate(
        [make_adult_like(n_instances=n_train - 2560, n_background=1, seed=7).X,
         data.X],
        axis=0,
    )
    w_true = rng.normal(0.0, 0.7, size=d)
    logits = X_train @ w_true + rng.normal(0, 0.5, size=n_train)
    y_train = (logits > 0).astype(np.int64)
    bunch = Bunch(
        X_train=X_train,
        y_train=y_train,
        X_test=data.X,
        y_test=(data.X @ w_true > 0).astype(np.int64),
        background=data.background,
        groups=data.groups,
        group_names=data.group_names,
        category_map=data.category_map,
    )
    os.makedirs(assets_dir, exist_ok=True)
    with open(path, "wb") as f:
        pickle.dump(bunch, f)
    return bunch


def load_model(path: str = None):
    """Unpickle a predictor; sklearn LogisticRegression is wrapped into the
    native LinearPredictor so the fused GPU path applies
    (reference ``load_model``, utils.py:137-157)."""
    import pickle

    path = path or os.path.join(ASSETS_DIR, "predictor.pkl")
    with open(path, "rb") as f:
        obj = pickle.load(f)
    from ..models.predictors import LinearPredictor

    if isinstance(obj, LinearPredictor) or callable(getattr(obj, "linear_params", None)):
        return obj
    coef = getattr(obj, "coef_", None)
    if coef is not None:
        intercept = np.atleast_1d(obj.intercept_)
        if coef.shape[0] == 1:  # binary sigmoid == softmax([0, z])
            w = np.vstack([np.zeros_like(coef[0]), coef[0]])
            b = np.array([0.0, float(intercept[0])])
        else:
            w, b = coef, intercept
        return LinearPredictor(w, b, activation="softmax")
    if not callable(obj):
        raise TypeError(f"loaded object {type(obj)} is not a predictor")
    return obj
