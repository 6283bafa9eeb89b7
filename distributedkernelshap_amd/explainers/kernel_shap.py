"""KernelShap public explainer — MI355X-native.

API-compatible rebuild of the reference's ``explainers/kernel_shap.py`` (C2/C3
in SURVEY.md §2.1): categorical grouping, background summarisation,
sequential-vs-distributed dispatch, link functions, result assembly into an
``Explanation``. The algorithm core is
``distributedkernelshap_amd.core.KernelShapEngine`` (native CPU oracle + HIP
CDNA4 kernels) instead of ``shap.KernelExplainer``.
"""
from __future__ import annotations

import logging
import warnings
from typing import Any, Callable, Dict, List, Optional, Sequence, Tuple, Union

import numpy as np
import pandas as pd
import scipy.sparse as sparse

from ..core.engine import KernelShapEngine
from ..utils import methdispatch
from ..interface import (
    DEFAULT_DATA_KERNEL_SHAP,
    DEFAULT_META_KERNEL_SHAP,
    Explainer,
    Explanation,
    FitMixin,
)

logger = logging.getLogger(__name__)

__all__ = [
    "KernelShap",
    "KernelExplainerWrapper",
    "rank_by_importance",
    "sum_categories",
    "KERNEL_SHAP_BACKGROUND_THRESHOLD",
    "DISTRIBUTED_OPTS",
]

# reference explainers/kernel_shap.py:23-33
KERNEL_SHAP_PARAMS = [
    "link",
    "group_names",
    "grouped",
    "groups",
    "weights",
    "summarise_background",
    "summarise_result",
    "transpose",
    "kwargs",
]
KERNEL_SHAP_BACKGROUND_THRESHOLD = 300

# reference explainers/kernel_shap.py:210-214 (n_cpus -> n_workers here:
# workers are GPU ranks / processes, not ray cpu actors)
DISTRIBUTED_OPTS: Dict[str, Any] = {
    "n_workers": None,
    "batch_size": 1,
    "actor_cpu_fraction": 1.0,
}


def rank_by_importance(
    shap_values: List[np.ndarray], feature_names: Optional[Sequence[str]] = None
) -> Dict[str, Any]:
    """Rank features by mean(|shap|) per class + aggregated over classes.

    Reference ``explainers/kernel_shap.py:36-109``.
    """
    if len(shap_values[0].shape) == 1:
        shap_values = [s.reshape(1, -1) for s in shap_values]
    n_features = shap_values[0].shape[1]
    if feature_names is None:
        feature_names = [f"feature_{i}" for i in range(n_features)]
    if len(feature_names) != n_features:
        logger.warning(
            "feature_names length (%d) != n_features (%d); using defaults",
            len(feature_names),
            n_features,
        )
        feature_names = [f"feature_{i}" for i in range(n_features)]

    importances: Dict[str, Dict[str, Any]] = {}
    avg_mag = []
    for class_idx, values in enumerate(shap_values):
        avg_class = np.abs(values).mean(axis=0)
        avg_mag.append(avg_class)
        order = np.argsort(avg_class)[::-1]
        importances[str(class_idx)] = {
            "ranked_effect": avg_class[order].tolist(),
            "names": [feature_names[i] for i in order],
        }
    combined = np.sum(np.stack(avg_mag), axis=0)
    order = np.argsort(combined)[::-1]
    importances["aggregated"] = {
        "ranked_effect": combined[order].tolist(),
        "names": [feature_names[i] for i in order],
    }
    return importances


def sum_categories(
    values: np.ndarray, start_idx: Sequence[int], enc_feat_dim: Sequence[int]
) -> np.ndarray:
    """Collapse one-hot encoded blocks of a shap-value matrix by summation.

    Reference ``explainers/kernel_shap.py:112-207`` (np.add.reduceat based).
    Supports 2-D (B, D) rank-1 values and 3-D (B, D, D) interaction values.
    """
    if start_idx is None or enc_feat_dim is None:
        raise ValueError("start_idx and enc_feat_dim must both be specified")
    if len(start_idx) != len(enc_feat_dim):
        raise ValueError("start_idx and enc_feat_dim must have equal length")

    def _slices(d: int) -> List[int]:
        out, i = [], 0
        starts = dict(zip(start_idx, enc_feat_dim))
        while i < d:
            out.append(i)
            i += starts.get(i, 1)
        return out

    if values.ndim == 2:
        slices = _slices(values.shape[1])
        return np.add.reduceat(values, slices, axis=1)
    if values.ndim == 3:
        slices = _slices(values.shape[1])
        out = np.add.reduceat(values, slices, axis=1)
        out = np.add.reduceat(out, slices, axis=2)
        return out
    raise ValueError("values must be 2-D or 3-D")


class KernelExplainerWrapper:
    """Worker-side explainer replica: per-process seeding and the
    ``(batch_idx, batch)`` tagging protocol so an unordered pool can reorder
    results (reference ``explainers/kernel_shap.py:217-261``).

    Wraps :class:`KernelShapEngine` directly: construction == the reference's
    actor-constructor broadcast of (predictor, background, link, seed).
    """

    def __init__(
        self,
        predictor: Callable,
        background_data: np.ndarray,
        bg_weights: Optional[np.ndarray] = None,
        groups: Optional[Sequence[Sequence[int]]] = None,
        link: str = "identity",
        seed: Optional[int] = None,
        device: str = "auto",
    ):
        self._engine = KernelShapEngine(
            predictor,
            background_data,
            bg_weights=bg_weights,
            groups=groups,
            link=link,
            seed=seed if seed is not None else 0,
            device=device,
        )
        self.expected_value = self._engine.expected_value
        self.vector_out = self._engine.vector_out

    def get_explanation(
        self, X: Union[Tuple[int, np.ndarray], np.ndarray], **kwargs
    ) -> Union[Tuple[int, List[np.ndarray]], List[np.ndarray]]:
        if isinstance(X, tuple):
            batch_idx, batch = X
            with warnings.catch_warnings():
                warnings.simplefilter("ignore")
                sv = self._engine.shap_values(batch, **kwargs)
            return batch_idx, sv
        return self._engine.shap_values(X, **kwargs)

    def shap_values(self, X, **kwargs):
        return self._engine.shap_values(X, **kwargs)

    def return_attribute(self, name: str):
        return getattr(self, name)


class KernelShap(Explainer, FitMixin):
    """Black-box KernelSHAP explainer over an MI355X-native engine.

    Constructor signature mirrors the reference
    (``explainers/kernel_shap.py:264-367``); ``distributed_opts`` selects the
    data-parallel engine (static instance sharding over GPU ranks / spawned
    workers; reference used a ray actor pool).
    """

    def __init__(
        self,
        predictor: Callable,
        link: str = "identity",
        feature_names: Optional[Sequence[str]] = None,
        categorical_names: Optional[Dict[int, List[str]]] = None,
        task: str = "classification",
        seed: Optional[int] = None,
        distributed_opts: Optional[Dict[str, Any]] = None,
        device: str = "auto",
    ):
        meta = dict(DEFAULT_META_KERNEL_SHAP)
        meta["name"] = self.__class__.__name__
        meta["task"] = task
        super().__init__(meta=meta)
        self.predictor = predictor
        self.link = link
        self.feature_names = list(feature_names) if feature_names is not None else []
        self.categorical_names = dict(categorical_names) if categorical_names else {}
        self.task = task
        self.seed = seed if seed is not None else 0
        self.device = device
        self._fitted = False
        self._explainer: Optional[Any] = None
        self.expected_value = None

        self.distribute = False
        self.distributed_opts = dict(DISTRIBUTED_OPTS)
        if distributed_opts is not None:
            distributed_opts = dict(distributed_opts)
            # reference spelling (explainers/kernel_shap.py:210-214) used
            # 'n_cpus' for the worker count; accept it as an alias
            if "n_cpus" in distributed_opts and "n_workers" not in distributed_opts:
                distributed_opts["n_workers"] = distributed_opts.pop("n_cpus")
            self.distributed_opts.update(distributed_opts)
            workers = self.distributed_opts.get("n_workers")
            # n_workers=1 is a pool of one replica (reference parity);
            # only None / 0 / -1 mean in-process sequential
            if workers is not None and workers >= 1:
                self.distribute = True
        self.distributed_opts["algorithm"] = "kernel_shap"
        self._update_metadata(
            {"task": task, "link": link, "distributed_opts": dict(self.distributed_opts)},
            params=True,
        )

    # ------------------------------------------------------------------ #

    def _update_metadata(self, data: Dict[str, Any], params: bool = False) -> None:
        """Reference ``explainers/kernel_shap.py:673-695``."""
        if params:
            self.meta.setdefault("params", {}).update(data)
        else:
            self.meta.update(data)

    def _summarise_background(
        self, background_data: np.ndarray, n_background_samples: int, use_kmeans: bool
    ):
        """Subsample or k-means summarise the background
        (reference ``explainers/kernel_shap.py:503-542``).

        Returns (data, weights). k-means centroids are snapped to the nearest
        actually-occurring value per column (shap.kmeans behavior) and weights
        are cluster occupancies.
        """
        n = background_data.shape[0]
        if n <= n_background_samples:
            return background_data, None
        if not use_kmeans:
            rng = np.random.Generator(np.random.Philox(key=[self.seed, 0xB6]))
            idx = rng.choice(n, size=n_background_samples, replace=False)
            return background_data[idx], None
        from sklearn.cluster import KMeans

        km = KMeans(n_clusters=n_background_samples, random_state=self.seed, n_init=10)
        labels = km.fit_predict(background_data)
        centers = km.cluster_centers_
        # snap each centroid coordinate to the nearest actual data value
        for j in range(background_data.shape[1]):
            col = np.sort(np.unique(background_data[:, j]))
            pos = np.searchsorted(col, centers[:, j])
            pos = np.clip(pos, 0, len(col) - 1)
            left = col[np.maximum(pos - 1, 0)]
            right = col[pos]
            centers[:, j] = np.where(
                np.abs(centers[:, j] - left) <= np.abs(right - centers[:, j]), left, right
            )
        weights = np.bincount(labels, minlength=n_background_samples).astype(np.float64)
        return centers, weights

    def _check_inputs(
        self,
        background_data: np.ndarray,
        group_names: Optional[Sequence[str]],
        groups: Optional[Sequence[Sequence[int]]],
        weights: Optional[np.ndarray],
    ) -> None:
        """Defensive validation with warnings, reference
        ``explainers/kernel_shap.py:369-501`` (graceful degradation, no raise
        unless structurally impossible)."""
        n, d = background_data.shape
        if n > KERNEL_SHAP_BACKGROUND_THRESHOLD:
            logger.warning(
                "Large background dataset (%d > %d rows): consider "
                "summarise_background=True to reduce explanation cost.",
                n,
                KERNEL_SHAP_BACKGROUND_THRESHOLD,
            )
        if groups is not None:
            ncols = sum(len(g) for g in groups)
            if ncols != d:
                raise ValueError(
                    f"groups cover {ncols} columns but background has {d}"
                )
            if group_names is not None and len(group_names) != len(groups):
                logger.warning(
                    "group_names length (%d) != groups length (%d); "
                    "auto-generating group names.",
                    len(group_names),
                    len(groups),
                )
        if group_names is not None and groups is None and len(group_names) != d:
            # transposed-data style mismatch detection (reference :443-449)
            logger.warning(
                "group_names length (%d) != n columns (%d) and no groups "
                "given; names will be auto-generated.",
                len(group_names),
                d,
            )
        if weights is not None:
            w = np.asarray(weights)
            if w.ndim != 1 or w.shape[0] != n:
                raise ValueError(
                    f"weights must be 1-D of length {n}, got shape {w.shape}"
                )
            if not np.all(w >= 0) or w.sum() <= 0:
                raise ValueError("weights must be non-negative and sum > 0")

    # ------------------------------------------------------------------ #

    @methdispatch
    def _coerce_data(self, data, want_names: bool = True):
        """Normalise supported input containers to (ndarray, names).

        Single-dispatch on the data type via ``utils.methdispatch``, mirroring
        the reference ``_get_data`` (``explainers/kernel_shap.py:544-671``,
        dispatch helper ``utils.py:43-64``): registered overloads for
        DataFrame, Series and scipy sparse below; the base case covers
        ndarray and any array-like.
        """
        arr = np.asarray(data, dtype=np.float64)
        if arr.ndim == 1:
            arr = arr.reshape(1, -1)
        return (arr, None) if want_names else arr

    @_coerce_data.register(pd.Series)
    def _coerce_series(self, data, want_names: bool = True):
        names = [str(data.name)] if data.name is not None else None
        arr = np.asarray(data.to_numpy(), dtype=np.float64).reshape(1, -1)
        return (arr, names) if want_names else arr

    @_coerce_data.register(pd.DataFrame)
    def _coerce_frame(self, data, want_names: bool = True):
        names = [str(c) for c in data.columns]
        arr = np.asarray(data.to_numpy(), dtype=np.float64)
        return (arr, names) if want_names else arr

    @_coerce_data.register(sparse.spmatrix)
    def _coerce_sparse(self, data, want_names: bool = True):
        logger.warning(
            "Sparse background/input densified; KernelSHAP perturbation "
            "synthesis operates on dense rows."
        )
        arr = np.asarray(data.toarray(), dtype=np.float64)
        if arr.ndim == 1:
            arr = arr.reshape(1, -1)
        return (arr, None) if want_names else arr

    def fit(
        self,
        background_data: np.ndarray,
        summarise_background: Union[bool, str] = False,
        n_background_samples: int = KERNEL_SHAP_BACKGROUND_THRESHOLD,
        group_names: Optional[Sequence[str]] = None,
        groups: Optional[Sequence[Sequence[int]]] = None,
        weights: Optional[np.ndarray] = None,
        **kwargs,
    ) -> "KernelShap":
        """Fit the explainer on a background dataset
        (reference ``explainers/kernel_shap.py:697-808``).

        ``summarise_background``: False | True (subsample) | 'kmeans'.
        Accepts ndarray, DataFrame, Series or scipy sparse input.
        """
        background_data, inferred_names = self._coerce_data(background_data)
        if group_names is None and groups is None and inferred_names is not None:
            group_names = inferred_names

        bg_weights = None
        summarised = False
        if summarise_background:
            use_kmeans = summarise_background == "kmeans"
            background_data, bg_weights = self._summarise_background(
                background_data, n_background_samples, use_kmeans
            )
            summarised = bg_weights is not None or background_data.shape[0] <= n_background_samples
        if weights is not None:
            bg_weights = np.asarray(weights, dtype=np.float64)

        self._check_inputs(background_data, group_names, groups, weights)

        d = background_data.shape[1]
        grouped = groups is not None
        if groups is None and group_names is not None and len(group_names) == d:
            groups = [[j] for j in range(d)]
            grouped = False
        if groups is not None and group_names is None:
            group_names = [f"group_{i}" for i in range(len(groups))]
        if group_names is None:
            group_names = self.feature_names or [f"feature_{j}" for j in range(d)]
            if len(group_names) != (len(groups) if groups else d):
                group_names = [f"feature_{j}" for j in range(d)]
        self.group_names = list(group_names)
        self.groups = groups
        self.background_data = background_data
        self.bg_weights = bg_weights

        init_kwargs = dict(
            bg_weights=bg_weights,
            groups=groups,
            link=self.link,
            seed=self.seed,
            device=self.device,
        )
        if self.distribute:
            from .distributed import DistributedExplainer

            self._explainer = DistributedExplainer(
                dict(self.distributed_opts),
                KernelExplainerWrapper,
                (self.predictor, background_data),
                init_kwargs,
            )
        else:
            self._explainer = KernelExplainerWrapper(
                self.predictor, background_data, **init_kwargs
            )
        self.expected_value = self._explainer.expected_value
        self._fitted = True
        self._update_metadata(
            {
                "grouped": grouped,
                "group_names": list(self.group_names),
                "summarise_background": bool(summarised),
                "weights": weights is not None,
            },
            params=True,
        )
        return self

    # ------------------------------------------------------------------ #

    def explain(
        self,
        X: np.ndarray,
        summarise_result: bool = False,
        cat_vars_start_idx: Optional[Sequence[int]] = None,
        cat_vars_enc_dim: Optional[Sequence[int]] = None,
        **kwargs,
    ) -> Explanation:
        """Explain a batch of instances
        (reference ``explainers/kernel_shap.py:810-898``).

        kwargs ride through to the engine: ``nsamples``, ``l1_reg``, ``silent``.
        """
        if not self._fitted:
            raise TypeError(
                "KernelShap explainer has not been fitted; call fit() first"
            )
        X = self._coerce_data(X, want_names=False)
        kwargs.pop("silent", None)
        shap_values = self._explainer.get_explanation(X, **kwargs)
        self.expected_value = (
            self._explainer.expected_value
            if not isinstance(self._explainer.expected_value, type(None))
            else self.expected_value
        )
        return self.build_explanation(
            X,
            shap_values,
            self.expected_value,
            summarise_result=summarise_result,
            cat_vars_start_idx=cat_vars_start_idx,
            cat_vars_enc_dim=cat_vars_enc_dim,
        )

    def build_explanation(
        self,
        X: np.ndarray,
        shap_values: List[np.ndarray],
        expected_value: np.ndarray,
        summarise_result: bool = False,
        cat_vars_start_idx: Optional[Sequence[int]] = None,
        cat_vars_enc_dim: Optional[Sequence[int]] = None,
    ) -> Explanation:
        """Assemble the Explanation object
        (reference ``explainers/kernel_shap.py:900-980``); the output layout
        is the compatibility contract (BASELINE.json north star)."""
        summarised = False
        if summarise_result:
            if cat_vars_start_idx is None or cat_vars_enc_dim is None:
                logger.warning(
                    "summarise_result=True requires cat_vars_start_idx and "
                    "cat_vars_enc_dim; skipping summarisation."
                )
            else:
                shap_values = [
                    sum_categories(sv, cat_vars_start_idx, cat_vars_enc_dim)
                    for sv in shap_values
                ]
                summarised = True

        model_out = np.asarray(self.predictor(X))
        # raw_prediction is exposed on the explainer's link scale — the
        # reference applies linkfv(self.predictor(X)) (kernel_shap.py:948-950)
        from ..core.links import convert_to_link

        linkf, _ = convert_to_link(self.link if isinstance(self.link, str) else "identity")
        raw_pred = linkf(model_out)
        if self.task == "regression":
            prediction = raw_pred.reshape(raw_pred.shape[0], -1)
        elif raw_pred.ndim > 1 and raw_pred.shape[1] > 1:
            # link functions are monotone: argmax is link-invariant
            prediction = np.argmax(raw_pred, axis=1)
        else:
            # single-output classifier: threshold in model (probability) space
            prediction = (model_out > 0.5).astype(int).reshape(-1)

        importances = rank_by_importance(
            shap_values,
            feature_names=self.group_names if not summarised else None,
        )

        data = {
            "shap_values": shap_values,
            "expected_value": np.atleast_1d(np.asarray(expected_value)),
            "link": self.link if isinstance(self.link, str) else "custom",
            "categorical_names": self.categorical_names,
            "feature_names": list(self.group_names),
            "raw": {
                "raw_prediction": raw_pred,
                "prediction": prediction,
                "instances": X,
                "importances": importances,
            },
        }
        full = dict(DEFAULT_DATA_KERNEL_SHAP)
        full.update(data)
        self._update_metadata({"summarise_result": summarised}, params=True)
        return Explanation(dict(self.meta), full)

    def reset_predictor(self, predictor: Callable) -> None:
        self.predictor = predictor
